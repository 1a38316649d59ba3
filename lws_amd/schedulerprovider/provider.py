"""Gang-scheduling provider seam.

Behavioral port of reference pkg/schedulerprovider (interface.go +
volcano_provider.go): a pluggable provider with two hooks — webhook-side
pod-group metadata injection and pod-controller-side PodGroup creation.
The in-repo GangProvider creates a PodGroup per (group, revision) with
MinMember=size (1 for LeaderReady) and MinResources summed over the whole
group, owned by the leader pod so it is GC'd with the group; the lws_amd
scheduler then binds the gang all-or-nothing.
"""
from __future__ import annotations

from typing import Optional

from ..api import leaderworkerset as lwsapi
from ..api.core import Pod, PodGroup, PodGroupSpec
from ..api.meta import IntOrString, OwnerReference
from ..cluster.scheduler import POD_GROUP_ANNOTATION
from ..cluster.store import AlreadyExistsError, Store
from ..utils import revision as revisionutils

SUPPORTED_PROVIDERS = ("gang", "volcano")
QUEUE_ANNOTATION = "scheduling.lws.amd.com/queue"


def pod_group_name(pod: Pod) -> str:
    """interface.go:47-51 — ``<lws>-<groupIdx>-<revision>``."""
    labels = pod.metadata.labels or {}
    return (f"{labels.get(lwsapi.SET_NAME_LABEL_KEY, '')}-"
            f"{labels.get(lwsapi.GROUP_INDEX_LABEL_KEY, '')}-"
            f"{revisionutils.get_revision_key(pod)}")


def calculate_pg_min_resources(lws) -> dict[str, IntOrString]:
    """pkg/utils/utils.go:84-103 — leader + (size-1) x worker resource sum
    (amd.com/gpu and any other countable resources)."""
    t = lws.spec.leader_worker_template
    size = t.size or 1
    leader_tmpl = t.leader_template if t.leader_template is not None \
        else t.worker_template
    totals: dict[str, float] = {}

    def add(template, factor):
        for c in template.spec.containers:
            for res, val in (c.resources.requests or
                             c.resources.limits or {}).items():
                try:
                    totals[res] = totals.get(res, 0) + float(val) * factor
                except (TypeError, ValueError):
                    pass
    add(leader_tmpl, 1)
    if size > 1:
        add(t.worker_template, size - 1)
    return {k: int(v) if float(v).is_integer() else v
            for k, v in totals.items()}


class GangProvider:
    """Volcano-provider equivalent running against the in-repo scheduler
    (volcano_provider.go:49-109)."""

    def __init__(self, store: Store):
        self.store = store

    def inject_pod_group_metadata(self, pod: Pod) -> None:
        """Webhook hook: stamp the pod-group annotation."""
        if lwsapi.SET_NAME_LABEL_KEY not in (pod.metadata.labels or {}):
            return
        pod.metadata.annotations[POD_GROUP_ANNOTATION] = pod_group_name(pod)

    def create_pod_group_if_not_exists(self, lws, leader_pod: Pod) -> None:
        """Pod-controller hook: one PodGroup per group per revision, owned
        by the leader pod (GC'd with the group)."""
        name = pod_group_name(leader_pod)
        if self.store.try_get("PodGroup", leader_pod.metadata.namespace, name):
            return
        size = lws.spec.leader_worker_template.size or 1
        min_member = size
        if lws.spec.startup_policy == lwsapi.StartupPolicyType.LeaderReady:
            # workers only exist after the leader is ready — gang of 1
            min_member = 1
        pg = PodGroup(spec=PodGroupSpec(
            min_member=min_member,
            min_resources=calculate_pg_min_resources(lws),
            queue=(lws.metadata.annotations or {}).get(QUEUE_ANNOTATION, "")))
        pg.metadata.name = name
        pg.metadata.namespace = leader_pod.metadata.namespace
        pg.metadata.labels = {
            lwsapi.SET_NAME_LABEL_KEY:
                (lws.metadata.labels or {}).get(lwsapi.SET_NAME_LABEL_KEY,
                                                lws.metadata.name)}
        pg.metadata.owner_references = [OwnerReference(
            api_version="v1", kind="Pod", name=leader_pod.metadata.name,
            uid=leader_pod.metadata.uid, controller=True,
            block_owner_deletion=True)]
        try:
            self.store.create(pg)
        except AlreadyExistsError:
            pass


def new_scheduler_provider(name: str, store: Store) -> Optional[GangProvider]:
    """interface.go:57-64 factory."""
    if not name:
        return None
    if name not in SUPPORTED_PROVIDERS:
        raise ValueError(f"unsupported scheduler provider {name!r}; "
                         f"supported: {SUPPORTED_PROVIDERS}")
    return GangProvider(store)
