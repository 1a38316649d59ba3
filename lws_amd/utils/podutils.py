"""Pod-level domain utilities.

Behavioral port of reference pkg/utils/pod/pod_utils.go: readiness/restart
predicates and the LWS_* rendezvous env injection contract that every
served container (including lws_amd's own MI355X engine) reads.
"""
from __future__ import annotations

from typing import Optional

from ..api import leaderworkerset as lwsapi
from ..api.core import Container, EnvVar, Pod


def leader_pod(pod: Pod) -> bool:
    """pod_utils.go:53-55 — leader iff worker-index label == "0"."""
    return (pod.metadata.labels or {}).get(lwsapi.WORKER_INDEX_LABEL_KEY) == "0"


def container_restarted(pod: Pod) -> bool:
    """pod_utils.go:29-45 — any container/init-container restartCount > 0."""
    if pod.status.phase in ("Running", "Succeeded", "Failed"):
        for cs in list(pod.status.init_container_statuses) + \
                list(pod.status.container_statuses):
            if cs.restart_count > 0:
                return True
    return False


def pod_deleted(pod: Pod) -> bool:
    return pod.metadata.deletion_timestamp is not None


def pod_running_and_ready(pod: Pod) -> bool:
    """pod_utils.go:58-60."""
    if pod.status.phase != "Running":
        return False
    for cond in pod.status.conditions:
        if cond.type == "Ready" and cond.status == "True":
            return True
    return False


def add_env_vars_if_not_exists(c: Container, first_env: EnvVar,
                               *envs: EnvVar) -> None:
    """pod_utils.go:108-129 — prepend, preserving injected order, keeping
    user-specified vars that don't collide."""
    new_env = [first_env, *envs]
    seen = {e.name for e in new_env}
    for env in c.env:
        if env.name not in seen:
            new_env.append(env)
            seen.add(env.name)
    c.env = new_env


def add_lws_variables(pod: Pod) -> None:
    """pod_utils.go:132-179 — inject LWS_LEADER_ADDRESS / LWS_GROUP_SIZE /
    LWS_WORKER_INDEX into every container and init-container."""
    labels = pod.metadata.labels or {}
    annotations = pod.metadata.annotations or {}
    lws_name = labels.get(lwsapi.SET_NAME_LABEL_KEY)
    if not lws_name:
        raise ValueError(f"no name label found for pod {pod.metadata.name}")
    group_index = labels.get(lwsapi.GROUP_INDEX_LABEL_KEY)
    if group_index is None:
        raise ValueError(f"no group index label found for pod {pod.metadata.name}")
    size = annotations.get(lwsapi.SIZE_ANNOTATION_KEY)
    if size is None:
        raise ValueError(f"no size annotation found for pod {pod.metadata.name}")
    worker_index = labels.get(lwsapi.WORKER_INDEX_LABEL_KEY)
    if worker_index is None:
        raise ValueError(f"no worker index label found for pod {pod.metadata.name}")

    leader_address = EnvVar(
        name=lwsapi.LWS_LEADER_ADDRESS,
        value=f"{lws_name}-{group_index}.{pod.spec.subdomain}.{pod.metadata.namespace}")
    size_env = EnvVar(name=lwsapi.LWS_GROUP_SIZE, value=size)
    worker_env = EnvVar(name=lwsapi.LWS_WORKER_INDEX, value=worker_index)

    for c in pod.spec.containers:
        add_env_vars_if_not_exists(c, leader_address, size_env, worker_env)
    for c in pod.spec.init_containers:
        add_env_vars_if_not_exists(c, leader_address, size_env, worker_env)
