"""LeaderWorkerSet v1 API types.

Byte-compatible (field names, constants, env vars, defaults, validation
envelope) with the reference API group `leaderworkerset.x-k8s.io/v1`
(/root/reference/api/leaderworkerset/v1/leaderworkerset_types.go), built
for the lws_amd standalone control plane.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional

from .core import (PersistentVolumeClaim, PodTemplateSpec,
                   StatefulSetPersistentVolumeClaimRetentionPolicy)
from .meta import Condition, IntOrString, ObjectMeta

GROUP = "leaderworkerset.x-k8s.io"
VERSION = "v1"
API_VERSION = f"{GROUP}/{VERSION}"
KIND = "LeaderWorkerSet"

# --- well-known annotation/label keys (leaderworkerset_types.go:26-99) ---
EXCLUSIVE_KEY_ANNOTATION_KEY = "leaderworkerset.sigs.k8s.io/exclusive-topology"
SUBGROUP_EXCLUSIVE_KEY_ANNOTATION_KEY = "leaderworkerset.sigs.k8s.io/subgroup-exclusive-topology"
SET_NAME_LABEL_KEY = "leaderworkerset.sigs.k8s.io/name"
GROUP_INDEX_LABEL_KEY = "leaderworkerset.sigs.k8s.io/group-index"
WORKER_INDEX_LABEL_KEY = "leaderworkerset.sigs.k8s.io/worker-index"
SIZE_ANNOTATION_KEY = "leaderworkerset.sigs.k8s.io/size"
REPLICAS_ANNOTATION_KEY = "leaderworkerset.sigs.k8s.io/replicas"
GROUP_UNIQUE_HASH_LABEL_KEY = "leaderworkerset.sigs.k8s.io/group-key"
LEADER_POD_NAME_ANNOTATION_KEY = "leaderworkerset.sigs.k8s.io/leader-name"
REVISION_KEY = "leaderworkerset.sigs.k8s.io/template-revision-hash"
SUBGROUP_INDEX_LABEL_KEY = "leaderworkerset.sigs.k8s.io/subgroup-index"
SUBGROUP_SIZE_ANNOTATION_KEY = "leaderworkerset.sigs.k8s.io/subgroup-size"
SUBGROUP_UNIQUE_HASH_LABEL_KEY = "leaderworkerset.sigs.k8s.io/subgroup-key"
SUBGROUP_POLICY_TYPE_ANNOTATION_KEY = "leaderworkerset.sigs.k8s.io/subgroup-policy-type"
SUBDOMAIN_POLICY_ANNOTATION_KEY = "leaderworkerset.sigs.k8s.io/subdomainPolicy"
RECREATE_GROUP_AFTER_START_ANNOTATION_KEY = (
    "leaderworkerset.sigs.k8s.io/experimental-recreate-group-after-start")

# --- injected env vars (leaderworkerset_types.go:67-75) ---
LWS_LEADER_ADDRESS = "LWS_LEADER_ADDRESS"
LWS_GROUP_SIZE = "LWS_GROUP_SIZE"
LWS_WORKER_INDEX = "LWS_WORKER_INDEX"

# --- scale envelope (leaderworkerset_types.go:122-123 + webhook) ---
MAX_REPLICAS = 1_000_000
MAX_INT32 = 2**31 - 1

# --- enums ---
class RestartPolicyType:
    RecreateGroupOnPodRestart = "RecreateGroupOnPodRestart"
    RecreateGroupAfterStart = "RecreateGroupAfterStart"
    DeprecatedDefault = "Default"
    NoneRestart = "None"


class StartupPolicyType:
    LeaderReady = "LeaderReady"
    LeaderCreated = "LeaderCreated"


class SubGroupPolicyType:
    LeaderWorker = "LeaderWorker"
    LeaderExcluded = "LeaderExcluded"


class SubdomainPolicy:
    Shared = "Shared"
    UniquePerReplica = "UniquePerReplica"


class RolloutStrategyType:
    RollingUpdate = "RollingUpdate"


# --- condition types (leaderworkerset_types.go:395-413) ---
class LeaderWorkerSetConditionType:
    Available = "Available"
    Progressing = "Progressing"
    UpdateInProgress = "UpdateInProgress"


@dataclass
class RollingUpdateConfiguration:
    """leaderworkerset_types.go:269-314."""

    partition: Optional[int] = None          # default 0
    max_unavailable: Optional[IntOrString] = None  # default 1
    max_surge: Optional[IntOrString] = None        # default 0


@dataclass
class RolloutStrategy:
    type: str = ""
    rolling_update_configuration: Optional[RollingUpdateConfiguration] = None


@dataclass
class SubGroupPolicy:
    type: Optional[str] = None  # SubGroupPolicyType, default LeaderWorker
    sub_group_size: Optional[int] = None


@dataclass
class NetworkConfig:
    subdomain_policy: Optional[str] = None  # SubdomainPolicy


@dataclass
class LeaderWorkerTemplate:
    """leaderworkerset_types.go:151-191."""

    leader_template: Optional[PodTemplateSpec] = None
    worker_template: PodTemplateSpec = field(default_factory=PodTemplateSpec)
    size: Optional[int] = None               # default 1
    restart_policy: str = ""                 # default RecreateGroupOnPodRestart
    sub_group_policy: Optional[SubGroupPolicy] = None
    volume_claim_templates: list[PersistentVolumeClaim] = field(default_factory=list)
    persistent_volume_claim_retention_policy: Optional[
        StatefulSetPersistentVolumeClaimRetentionPolicy] = None


@dataclass
class LeaderWorkerSetSpec:
    """leaderworkerset_types.go:111-143."""

    replicas: Optional[int] = None           # default 1
    leader_worker_template: LeaderWorkerTemplate = field(
        default_factory=LeaderWorkerTemplate)
    rollout_strategy: RolloutStrategy = field(default_factory=RolloutStrategy)
    startup_policy: str = ""                 # default LeaderCreated
    network_config: Optional[NetworkConfig] = None


@dataclass
class LeaderWorkerSetStatus:
    """leaderworkerset_types.go:363-393."""

    conditions: list[Condition] = field(default_factory=list)
    ready_replicas: int = 0
    updated_replicas: int = 0
    replicas: int = 0
    hpa_pod_selector: str = ""
    observed_generation: int = 0


@dataclass
class LeaderWorkerSet:
    api_version: str = API_VERSION
    kind: str = KIND
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: LeaderWorkerSetSpec = field(default_factory=LeaderWorkerSetSpec)
    status: LeaderWorkerSetStatus = field(default_factory=LeaderWorkerSetStatus)


@dataclass
class LeaderWorkerSetTemplateSpec:
    """leaderworkerset_types.go:447-455 — embedded by DisaggregatedSet roles."""

    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: LeaderWorkerSetSpec = field(default_factory=LeaderWorkerSetSpec)
