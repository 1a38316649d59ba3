"""Pod mutating admission — the identity-injection engine.

Behavioral port of reference pkg/webhooks/pod_webhook.go:
 - derive group/worker index from the pod-name ordinal
 - stamp group-index / worker-index / group-key (sha1) / subgroup labels
 - set subdomain for UniquePerReplica
 - inject exclusive-placement pod (anti-)affinity
 - inject gang-scheduling metadata via the SchedulerProvider seam
 - inject accelerator env (amd.com/gpu -> RCCL; replaces the TPU path)
 - inject LWS_* env into every container
"""
from __future__ import annotations

from typing import Optional

from ..accelerators.rccl import add_rccl_variables, pod_requests_gpus
from ..api import leaderworkerset as lwsapi
from ..api.core import (Affinity, Pod, PodAffinity, PodAffinityTerm,
                        PodAntiAffinity)
from ..api.meta import LabelSelector, LabelSelectorRequirement
from ..cluster.statefulset_controller import parse_parent_and_ordinal
from ..utils.hashutil import sha1_hex
from ..utils.podutils import add_lws_variables, leader_pod


def gen_group_unique_key(ns: str, pod_name: str) -> str:
    """pod_webhook.go:180-182."""
    return sha1_hex(f"{ns}/{pod_name}")


def exclusive_affinity_applied(pod: Pod, topology_key: str) -> bool:
    """pod_webhook.go:230-247."""
    aff = pod.spec.affinity
    if aff is None or aff.pod_affinity is None or aff.pod_anti_affinity is None:
        return False
    has_aff = any(t.topology_key == topology_key for t in
                  aff.pod_affinity.required_during_scheduling_ignored_during_execution)
    has_anti = any(t.topology_key == topology_key for t in
                   aff.pod_anti_affinity.required_during_scheduling_ignored_during_execution)
    return has_aff and has_anti


def set_exclusive_affinities(pod: Pod, group_unique_key: str,
                             topology_key: str, pod_affinity_key: str) -> None:
    """pod_webhook.go:185-227 — podAffinity on key IN hash, podAntiAffinity
    on key Exists && NotIn hash, both at topologyKey."""
    if exclusive_affinity_applied(pod, topology_key):
        return
    if pod.spec.affinity is None:
        pod.spec.affinity = Affinity()
    if pod.spec.affinity.pod_affinity is None:
        pod.spec.affinity.pod_affinity = PodAffinity()
    if pod.spec.affinity.pod_anti_affinity is None:
        pod.spec.affinity.pod_anti_affinity = PodAntiAffinity()
    pod.spec.affinity.pod_affinity.required_during_scheduling_ignored_during_execution.append(
        PodAffinityTerm(
            label_selector=LabelSelector(match_expressions=[
                LabelSelectorRequirement(key=pod_affinity_key, operator="In",
                                         values=[group_unique_key])]),
            topology_key=topology_key))
    pod.spec.affinity.pod_anti_affinity.required_during_scheduling_ignored_during_execution.append(
        PodAffinityTerm(
            label_selector=LabelSelector(match_expressions=[
                LabelSelectorRequirement(key=pod_affinity_key, operator="Exists"),
                LabelSelectorRequirement(key=pod_affinity_key, operator="NotIn",
                                         values=[group_unique_key])]),
            topology_key=topology_key))


def get_sub_group_index(pod_count: int, sub_group_size: int,
                        worker_index: int) -> str:
    """pod_webhook.go:249-255."""
    if (pod_count - 1) % sub_group_size == 0 and pod_count % sub_group_size != 0:
        return str((worker_index - 1) // sub_group_size)
    return str(worker_index // sub_group_size)


class PodWebhook:
    """Registered as a mutator on Pod create (pod_webhook.go:83-178)."""

    def __init__(self, scheduler_provider=None):
        self.scheduler_provider = scheduler_provider

    def default(self, pod: Pod) -> None:
        labels = pod.metadata.labels
        annotations = pod.metadata.annotations
        if lwsapi.SET_NAME_LABEL_KEY not in labels:
            return  # not part of a leaderworkerset
        size = annotations.get(lwsapi.SIZE_ANNOTATION_KEY)
        if size is None:
            raise ValueError(
                f"size annotation is unexpectedly missing for pod {pod.metadata.name}")
        pod_count = int(size)

        if leader_pod(pod):
            if lwsapi.GROUP_INDEX_LABEL_KEY not in labels:
                _, group_index = parse_parent_and_ordinal(pod.metadata.name)
                if group_index == -1:
                    raise ValueError(f"parsing pod ordinal for pod {pod.metadata.name}")
                labels[lwsapi.GROUP_INDEX_LABEL_KEY] = str(group_index)
            if annotations.get(lwsapi.SUBDOMAIN_POLICY_ANNOTATION_KEY) == \
                    lwsapi.SubdomainPolicy.UniquePerReplica:
                pod.spec.subdomain = pod.metadata.name
            if lwsapi.GROUP_UNIQUE_HASH_LABEL_KEY not in labels:
                group_unique_key = gen_group_unique_key(
                    pod.metadata.namespace, pod.metadata.name)
                labels[lwsapi.GROUP_UNIQUE_HASH_LABEL_KEY] = group_unique_key
            else:
                group_unique_key = labels[lwsapi.GROUP_UNIQUE_HASH_LABEL_KEY]
            ep_key = annotations.get(lwsapi.EXCLUSIVE_KEY_ANNOTATION_KEY)
            if ep_key is not None:
                set_exclusive_affinities(pod, group_unique_key, ep_key,
                                         lwsapi.GROUP_UNIQUE_HASH_LABEL_KEY)
            sub_group_size = annotations.get(lwsapi.SUBGROUP_SIZE_ANNOTATION_KEY)
            policy_type = annotations.get(
                lwsapi.SUBGROUP_POLICY_TYPE_ANNOTATION_KEY, "")
            if sub_group_size is not None and \
                    not labels.get(lwsapi.SUBGROUP_INDEX_LABEL_KEY) and \
                    policy_type != lwsapi.SubGroupPolicyType.LeaderExcluded:
                labels[lwsapi.SUBGROUP_INDEX_LABEL_KEY] = "0"
                sub_key = gen_group_unique_key(pod.metadata.name, "0")
                labels[lwsapi.SUBGROUP_UNIQUE_HASH_LABEL_KEY] = sub_key
                sub_ep_key = annotations.get(
                    lwsapi.SUBGROUP_EXCLUSIVE_KEY_ANNOTATION_KEY)
                if sub_ep_key is not None:
                    set_exclusive_affinities(pod, sub_key, sub_ep_key,
                                             lwsapi.SUBGROUP_UNIQUE_HASH_LABEL_KEY)
        else:
            _, worker_index = parse_parent_and_ordinal(pod.metadata.name)
            if worker_index == -1:
                raise ValueError(f"parsing pod ordinal for pod {pod.metadata.name}")
            labels[lwsapi.WORKER_INDEX_LABEL_KEY] = str(worker_index)
            sub_group_size = annotations.get(lwsapi.SUBGROUP_SIZE_ANNOTATION_KEY)
            if sub_group_size is not None and \
                    not labels.get(lwsapi.SUBGROUP_INDEX_LABEL_KEY):
                leader_name = annotations.get(
                    lwsapi.LEADER_POD_NAME_ANNOTATION_KEY, "")
                sub_group_index = get_sub_group_index(
                    pod_count, int(sub_group_size), worker_index)
                labels[lwsapi.SUBGROUP_INDEX_LABEL_KEY] = sub_group_index
                sub_key = gen_group_unique_key(leader_name, sub_group_index)
                labels[lwsapi.SUBGROUP_UNIQUE_HASH_LABEL_KEY] = sub_key
                sub_ep_key = annotations.get(
                    lwsapi.SUBGROUP_EXCLUSIVE_KEY_ANNOTATION_KEY)
                if sub_ep_key is not None:
                    set_exclusive_affinities(pod, sub_key, sub_ep_key,
                                             lwsapi.SUBGROUP_UNIQUE_HASH_LABEL_KEY)

        if self.scheduler_provider is not None:
            self.scheduler_provider.inject_pod_group_metadata(pod)

        if pod_requests_gpus(pod):
            add_rccl_variables(pod, pod_count)

        add_lws_variables(pod)


def register(store, scheduler_provider=None) -> PodWebhook:
    wh = PodWebhook(scheduler_provider)
    store.add_mutator("Pod", wh.default)
    return wh
