"""corev1/appsv1-shaped workload types for the lws_amd cluster substrate.

The reference orchestrates stock Kubernetes objects (Pod, StatefulSet,
Service, ControllerRevision).  lws_amd ships its own control plane, so the
same object shapes are defined here; field names follow the k8s wire format
so manifests written for the reference port over directly.

Covers the subset the reference controllers actually read/write:
 - Pod/PodTemplateSpec incl. env, resources, nodeSelector, subdomain,
   affinity (pkg/webhooks/pod_webhook.go, pkg/utils/pod/pod_utils.go)
 - StatefulSet incl. ordinals.start, partition rolling update, PVC
   retention (pkg/controllers/pod_controller.go:381-461)
 - headless Service with publishNotReadyAddresses
   (pkg/utils/controller/controller_utils.go:33-65)
 - ControllerRevision (pkg/utils/revision/revision_utils.go)
 - PodGroup for gang scheduling (pkg/schedulerprovider/volcano_provider.go)
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional

from .meta import Condition, IntOrString, LabelSelector, ObjectMeta

# Resource name for AMD GPUs (replaces the reference's google.com/tpu /
# nvidia.com/gpu paths).
AMD_GPU_RESOURCE = "amd.com/gpu"


@dataclass
class EnvVarSource:
    field_path: str = ""


@dataclass
class EnvVar:
    name: str = ""
    value: str = ""
    value_from: Optional[EnvVarSource] = None


@dataclass
class ContainerPort:
    name: str = ""
    container_port: int = 0


@dataclass
class ResourceRequirements:
    limits: dict[str, IntOrString] = field(default_factory=dict)
    requests: dict[str, IntOrString] = field(default_factory=dict)


@dataclass
class VolumeMount:
    name: str = ""
    mount_path: str = ""


@dataclass
class Container:
    name: str = ""
    image: str = ""
    command: list[str] = field(default_factory=list)
    args: list[str] = field(default_factory=list)
    env: list[EnvVar] = field(default_factory=list)
    ports: list[ContainerPort] = field(default_factory=list)
    resources: ResourceRequirements = field(default_factory=ResourceRequirements)
    volume_mounts: list[VolumeMount] = field(default_factory=list)


@dataclass
class PodAffinityTerm:
    label_selector: Optional[LabelSelector] = None
    topology_key: str = ""


@dataclass
class PodAffinity:
    required_during_scheduling_ignored_during_execution: list[PodAffinityTerm] = field(
        default_factory=list)


@dataclass
class PodAntiAffinity:
    required_during_scheduling_ignored_during_execution: list[PodAffinityTerm] = field(
        default_factory=list)


@dataclass
class Affinity:
    pod_affinity: Optional[PodAffinity] = None
    pod_anti_affinity: Optional[PodAntiAffinity] = None


@dataclass
class PodSpec:
    containers: list[Container] = field(default_factory=list)
    init_containers: list[Container] = field(default_factory=list)
    node_selector: dict[str, str] = field(default_factory=dict)
    subdomain: str = ""
    hostname: str = ""
    affinity: Optional[Affinity] = None
    scheduler_name: str = ""
    priority_class_name: str = ""


@dataclass
class PodTemplateSpec:
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: PodSpec = field(default_factory=PodSpec)


@dataclass
class ContainerState:
    # "waiting" | "running" | "terminated"
    state: str = "waiting"
    exit_code: int = 0


@dataclass
class ContainerStatus:
    name: str = ""
    ready: bool = False
    restart_count: int = 0
    started: bool = False
    state: ContainerState = field(default_factory=ContainerState)


@dataclass
class PodStatus:
    phase: str = "Pending"  # Pending | Running | Succeeded | Failed
    conditions: list[Condition] = field(default_factory=list)
    container_statuses: list[ContainerStatus] = field(default_factory=list)
    init_container_statuses: list[ContainerStatus] = field(default_factory=list)
    pod_ip: str = ""
    host_ip: str = ""
    node_name: str = ""  # convenience mirror of spec.nodeName


@dataclass
class Pod:
    api_version: str = "v1"
    kind: str = "Pod"
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: PodSpec = field(default_factory=PodSpec)
    status: PodStatus = field(default_factory=PodStatus)
    # scheduling binding (spec.nodeName in k8s)
    node_name: str = ""


@dataclass
class PersistentVolumeClaimSpec:
    access_modes: list[str] = field(default_factory=list)
    storage_class_name: str = ""
    resources: ResourceRequirements = field(default_factory=ResourceRequirements)


@dataclass
class PersistentVolumeClaim:
    api_version: str = "v1"
    kind: str = "PersistentVolumeClaim"
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: PersistentVolumeClaimSpec = field(default_factory=PersistentVolumeClaimSpec)


@dataclass
class StatefulSetPersistentVolumeClaimRetentionPolicy:
    when_deleted: str = ""  # Retain | Delete
    when_scaled: str = ""


@dataclass
class StatefulSetOrdinals:
    start: int = 0


@dataclass
class RollingUpdateStatefulSetStrategy:
    partition: int = 0
    max_unavailable: Optional[IntOrString] = None


@dataclass
class StatefulSetUpdateStrategy:
    type: str = "RollingUpdate"
    rolling_update: Optional[RollingUpdateStatefulSetStrategy] = None


@dataclass
class StatefulSetSpec:
    replicas: int = 1
    selector: Optional[LabelSelector] = None
    template: PodTemplateSpec = field(default_factory=PodTemplateSpec)
    service_name: str = ""
    pod_management_policy: str = "OrderedReady"  # or "Parallel"
    update_strategy: StatefulSetUpdateStrategy = field(
        default_factory=StatefulSetUpdateStrategy)
    ordinals: Optional[StatefulSetOrdinals] = None
    volume_claim_templates: list[PersistentVolumeClaim] = field(default_factory=list)
    persistent_volume_claim_retention_policy: Optional[
        StatefulSetPersistentVolumeClaimRetentionPolicy] = None


@dataclass
class StatefulSetStatus:
    observed_generation: int = 0
    replicas: int = 0
    ready_replicas: int = 0
    current_replicas: int = 0
    updated_replicas: int = 0
    available_replicas: int = 0
    current_revision: str = ""
    update_revision: str = ""


@dataclass
class StatefulSet:
    api_version: str = "apps/v1"
    kind: str = "StatefulSet"
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: StatefulSetSpec = field(default_factory=StatefulSetSpec)
    status: StatefulSetStatus = field(default_factory=StatefulSetStatus)


@dataclass
class ServiceSpec:
    cluster_ip: str = ""  # "None" => headless
    selector: dict[str, str] = field(default_factory=dict)
    publish_not_ready_addresses: bool = False
    ports: list[ContainerPort] = field(default_factory=list)


@dataclass
class Service:
    api_version: str = "v1"
    kind: str = "Service"
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: ServiceSpec = field(default_factory=ServiceSpec)


@dataclass
class ControllerRevision:
    api_version: str = "apps/v1"
    kind: str = "ControllerRevision"
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    data: dict = field(default_factory=dict)  # raw patch snapshot
    revision: int = 0


@dataclass
class PodGroupSpec:
    min_member: int = 0
    min_resources: dict[str, IntOrString] = field(default_factory=dict)
    queue: str = ""


@dataclass
class PodGroupStatus:
    phase: str = ""


@dataclass
class PodGroup:
    """Gang-scheduling unit (volcano.sh PodGroup equivalent;
    pkg/schedulerprovider/volcano_provider.go:49-101)."""

    api_version: str = "scheduling.lws.amd.com/v1"
    kind: str = "PodGroup"
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: PodGroupSpec = field(default_factory=PodGroupSpec)
    status: PodGroupStatus = field(default_factory=PodGroupStatus)


# ---------------------------------------------------------------------------
# helpers shared by controllers

def pod_requests_amd_gpus(spec: PodSpec) -> int:
    """Number of amd.com/gpu requested by the pod (max of containers'
    requests+limits, as k8s treats limits as requests for extended
    resources).  Replaces PodRequestsTPUs (pkg/utils/accelerators/tpu.go:44)."""
    total = 0
    for c in list(spec.containers) + list(spec.init_containers):
        r = c.resources
        v = r.requests.get(AMD_GPU_RESOURCE, r.limits.get(AMD_GPU_RESOURCE, 0))
        try:
            total += int(v)
        except (TypeError, ValueError):
            pass
    return total
