"""amd.com/gpu + RCCL rendezvous env injector.

Replaces the reference's TPU/JAX injector (pkg/utils/accelerators/tpu.go)
with the MI355X-native equivalent: for pods requesting
``amd.com/gpu``, derive the torch.distributed-over-RCCL bootstrap env from
the group topology the same way tpu.go derives TPU hostlists:

  MASTER_ADDR       = leader address (reuses the LWS_LEADER_ADDRESS value)
  MASTER_PORT       = annotation lws.amd.com/rccl-port (default 29500)
  WORLD_SIZE        = group size x GPUs per pod (or subgroup size x GPUs)
  NODE_RANK         = worker index within the group (or subgroup)
  LOCAL_WORLD_SIZE  = GPUs per pod

With one process per GPU launched from NODE_RANK * LOCAL_WORLD_SIZE, every
rank computes RANK = NODE_RANK*LOCAL_WORLD_SIZE + LOCAL_RANK, and RCCL
rides xGMI intra-node.  Subgroup handling mirrors
tpu.go:99-199 (addTPUVariablesSubGroup): each subgroup is its own
rendezvous domain whose master is the subgroup's first pod.
"""
from __future__ import annotations

from ..api import leaderworkerset as lwsapi
from ..api.core import EnvVar, Pod, pod_requests_amd_gpus
from ..utils.podutils import add_env_vars_if_not_exists

RCCL_PORT_ANNOTATION = "lws.amd.com/rccl-port"
DEFAULT_RCCL_PORT = 29500

ENV_MASTER_ADDR = "MASTER_ADDR"
ENV_MASTER_PORT = "MASTER_PORT"
ENV_WORLD_SIZE = "WORLD_SIZE"
ENV_NODE_RANK = "NODE_RANK"
ENV_LOCAL_WORLD_SIZE = "LOCAL_WORLD_SIZE"


def pod_requests_gpus(pod: Pod) -> bool:
    return pod_requests_amd_gpus(pod.spec) > 0


def _pod_dns(name: str, subdomain: str, namespace: str) -> str:
    return f"{name}.{subdomain}.{namespace}"


def add_rccl_variables(pod: Pod, pod_count: int) -> None:
    """Inject RCCL rendezvous env (tpu.go:202-300 analogue)."""
    labels = pod.metadata.labels or {}
    annotations = pod.metadata.annotations or {}
    if not pod_requests_gpus(pod):
        return
    lws_name = labels.get(lwsapi.SET_NAME_LABEL_KEY)
    group_index = labels.get(lwsapi.GROUP_INDEX_LABEL_KEY)
    worker_index = int(labels.get(lwsapi.WORKER_INDEX_LABEL_KEY, "0"))
    if lws_name is None or group_index is None:
        raise ValueError(f"missing LWS identity labels on pod {pod.metadata.name}")
    gpus_per_pod = pod_requests_amd_gpus(pod.spec)
    port = int(annotations.get(RCCL_PORT_ANNOTATION, DEFAULT_RCCL_PORT))
    leader_name = f"{lws_name}-{group_index}"
    subdomain = pod.spec.subdomain or lws_name
    namespace = pod.metadata.namespace

    sub_group_size = annotations.get(lwsapi.SUBGROUP_SIZE_ANNOTATION_KEY)
    if sub_group_size is not None:
        sgs = int(sub_group_size)
        policy = annotations.get(lwsapi.SUBGROUP_POLICY_TYPE_ANNOTATION_KEY,
                                 lwsapi.SubGroupPolicyType.LeaderWorker)
        leader_even = (pod_count % sgs == 0)
        if policy == lwsapi.SubGroupPolicyType.LeaderExcluded:
            # subgroups over workers 1..N-1 only
            sub_idx = (worker_index - 1) // sgs if worker_index >= 1 else 0
            first = 1 + sub_idx * sgs
            node_rank = (worker_index - first)
            world = sgs * gpus_per_pod
        elif leader_even:
            sub_idx = worker_index // sgs
            first = sub_idx * sgs
            node_rank = worker_index - first
            world = sgs * gpus_per_pod
        else:
            # leader is the extra pod of subgroup 0 ((size-1) % sgs == 0)
            if worker_index == 0:
                sub_idx, first = 0, 0
                node_rank = 0
                world = (sgs + 1) * gpus_per_pod
            else:
                sub_idx = (worker_index - 1) // sgs
                first = 1 + sub_idx * sgs
                node_rank = (worker_index - first) + (1 if sub_idx == 0 else 0)
                world = (sgs + (1 if sub_idx == 0 else 0)) * gpus_per_pod
        master_pod = leader_name if first == 0 else f"{leader_name}-{first}"
        master = _pod_dns(master_pod, subdomain, namespace) \
            if master_pod != leader_name else _pod_dns(leader_name, subdomain, namespace)
    else:
        node_rank = worker_index
        world = pod_count * gpus_per_pod
        master = _pod_dns(leader_name, subdomain, namespace)

    envs = [
        EnvVar(name=ENV_MASTER_ADDR, value=master),
        EnvVar(name=ENV_MASTER_PORT, value=str(port)),
        EnvVar(name=ENV_WORLD_SIZE, value=str(world)),
        EnvVar(name=ENV_NODE_RANK, value=str(node_rank)),
        EnvVar(name=ENV_LOCAL_WORLD_SIZE, value=str(gpus_per_pod)),
    ]
    for c in pod.spec.containers:
        add_env_vars_if_not_exists(c, envs[0], *envs[1:])
    for c in pod.spec.init_containers:
        add_env_vars_if_not_exists(c, envs[0], *envs[1:])
