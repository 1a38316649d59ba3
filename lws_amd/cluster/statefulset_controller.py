"""StatefulSet controller for the lws_amd cluster substrate.

The reference delegates per-group pod lifecycle to Kubernetes' StatefulSet
controller; lws_amd ships its own.  Semantics covered (the subset the LWS
controllers depend on — SURVEY.md §2.2):

 - stable pod identity ``<sts>-<ordinal>`` with configurable ordinal start
   (worker STS uses start=1, reference pod_controller.go:436-446)
 - Parallel pod management (leader STS, leaderworkerset_controller.go:850)
 - template-hash revisions: currentRevision / updateRevision in status,
   pods labeled ``controller-revision-hash``
 - RollingUpdate with ``partition`` + ``maxUnavailable``: pods with
   ordinal >= partition are replaced with the new template, at most
   maxUnavailable simultaneously-unready; pods below partition are
   recreated from the *current* (old) revision
 - scale up/down, pod recreation on deletion, status accounting
   (AvailableReplicas/ReadyReplicas/Current/Updated revisions; readiness
   definition used by pkg/utils/statefulset/statefulset_utils.go:48-51)
"""
from __future__ import annotations

import re
from typing import Optional

from ..api import serde
from ..api.core import ControllerRevision, Pod, StatefulSet
from ..api.meta import OwnerReference, get_int_or_percent
from ..utils.hashutil import hash_object
from .controller import Controller, Manager
from .store import AlreadyExistsError, NotFoundError, Store

POD_NAME_RE = re.compile(r"^(.*)-([0-9]+)$")
STS_REVISION_LABEL = "controller-revision-hash"
STS_POD_NAME_LABEL = "statefulset.kubernetes.io/pod-name"
STS_OWNER_LABEL = "lws.amd.com/statefulset-name"


def parse_parent_and_ordinal(name: str) -> tuple[Optional[str], int]:
    """GetParentNameAndOrdinal equivalent
    (pkg/utils/statefulset/statefulset_utils.go:27-45)."""
    m = POD_NAME_RE.match(name)
    if not m:
        return None, -1
    return m.group(1), int(m.group(2))


def pod_is_ready(pod: Pod) -> bool:
    if pod.status.phase != "Running":
        return False
    for cond in pod.status.conditions:
        if cond.type == "Ready":
            return cond.status == "True"
    return False


def template_hash(template) -> str:
    return hash_object(serde.to_dict(template))


class StatefulSetController:
    def __init__(self, manager: Manager) -> None:
        self.store: Store = manager.store
        self.store.add_label_index(STS_OWNER_LABEL)
        self.ctrl = Controller("statefulset", self.reconcile)
        manager.add_controller(self.ctrl)
        manager.watch("StatefulSet", self.ctrl)
        manager.watch("Pod", self.ctrl, self._map_pod)

    def _map_pod(self, event: str, pod) -> list[tuple[str, str]]:
        parent, _ = parse_parent_and_ordinal(pod.metadata.name)
        owner_sts = pod.metadata.labels.get(STS_OWNER_LABEL)
        if owner_sts:
            return [(pod.metadata.namespace, owner_sts)]
        if parent:
            return [(pod.metadata.namespace, parent)]
        return []

    # ------------------------------------------------------------------
    def reconcile(self, namespace: str, name: str) -> Optional[float]:
        sts = self.store.try_get("StatefulSet", namespace, name)
        if sts is None or sts.metadata.deletion_timestamp is not None:
            return None

        start = sts.spec.ordinals.start if sts.spec.ordinals else 0
        replicas = max(0, sts.spec.replicas)
        want_ordinals = list(range(start, start + replicas))

        update_rev = template_hash(sts.spec.template)
        self._ensure_revision_snapshot(sts, update_rev)

        pods = {ord_: p for ord_, p in self._owned_pods(sts).items()}

        # --- scale down: delete pods outside [start, start+replicas) ---
        for ord_, pod in list(pods.items()):
            if ord_ not in want_ordinals:
                self._delete_pod(pod)
                del pods[ord_]

        # --- determine current revision -------------------------------
        current_rev = sts.status.current_revision or update_rev
        if current_rev != update_rev:
            live = [p for p in pods.values() if p.metadata.deletion_timestamp is None]
            if live and all(p.metadata.labels.get(STS_REVISION_LABEL) == update_rev
                            for p in live) and len(live) == replicas:
                current_rev = update_rev

        partition = 0
        max_unavailable = 1
        if sts.spec.update_strategy.rolling_update is not None:
            partition = sts.spec.update_strategy.rolling_update.partition or 0
            mu = sts.spec.update_strategy.rolling_update.max_unavailable
            if mu is not None:
                max_unavailable = max(1, get_int_or_percent(mu, replicas, False))

        # --- create missing pods --------------------------------------
        for ord_ in want_ordinals:
            if ord_ not in pods:
                rev = update_rev if ord_ >= partition else current_rev
                self._create_pod(sts, ord_, rev)

        # --- rolling update: replace stale pods >= partition -----------
        unready = sum(1 for ord_ in want_ordinals
                      if ord_ not in pods
                      or pods[ord_].metadata.deletion_timestamp is not None
                      or not pod_is_ready(pods[ord_]))
        budget = max_unavailable - unready
        for ord_ in sorted((o for o in want_ordinals if o >= partition), reverse=True):
            pod = pods.get(ord_)
            if pod is None or pod.metadata.deletion_timestamp is not None:
                continue
            if pod.metadata.labels.get(STS_REVISION_LABEL) != update_rev:
                if budget <= 0:
                    break
                self._delete_pod(pod)
                budget -= 1

        # --- status -----------------------------------------------------
        self._update_status(sts, want_ordinals, current_rev, update_rev)
        return None

    # ------------------------------------------------------------------
    def _owned_pods(self, sts: StatefulSet) -> dict[int, Pod]:
        out: dict[int, Pod] = {}
        # indexed lookup + shared refs (read-only: every pod mutation in
        # this controller goes through store verbs on fresh objects)
        for pod in self.store.list(
                "Pod", sts.metadata.namespace,
                label_selector={STS_OWNER_LABEL: sts.metadata.name},
                copy=False):
            parent, ord_ = parse_parent_and_ordinal(pod.metadata.name)
            if parent != sts.metadata.name or ord_ < 0:
                continue
            out[ord_] = pod
        return out

    def _ensure_revision_snapshot(self, sts: StatefulSet, rev: str) -> None:
        name = f"{sts.metadata.name}-{rev}"
        if self.store.try_get("ControllerRevision", sts.metadata.namespace, name):
            return
        cr = ControllerRevision(
            data={"template": serde.to_dict(sts.spec.template)})
        cr.metadata.name = name
        cr.metadata.namespace = sts.metadata.namespace
        cr.metadata.labels = {STS_OWNER_LABEL: sts.metadata.name,
                              STS_REVISION_LABEL: rev}
        cr.metadata.owner_references = [OwnerReference(
            api_version=sts.api_version, kind=sts.kind, name=sts.metadata.name,
            uid=sts.metadata.uid, controller=True, block_owner_deletion=False)]
        try:
            self.store.create(cr)
        except AlreadyExistsError:
            pass

    def _template_for_revision(self, sts: StatefulSet, rev: str):
        name = f"{sts.metadata.name}-{rev}"
        cr = self.store.try_get("ControllerRevision", sts.metadata.namespace, name)
        if cr is None:
            return serde.deep_copy(sts.spec.template)
        from ..api.core import PodTemplateSpec
        return serde.from_dict(PodTemplateSpec, cr.data["template"])

    def _create_pod(self, sts: StatefulSet, ordinal: int, rev: str) -> None:
        template = self._template_for_revision(sts, rev)
        pod = Pod(metadata=serde.deep_copy(template.metadata),
                  spec=serde.deep_copy(template.spec))
        pod.metadata.name = f"{sts.metadata.name}-{ordinal}"
        pod.metadata.namespace = sts.metadata.namespace
        pod.metadata.labels = dict(template.metadata.labels or {})
        pod.metadata.labels[STS_REVISION_LABEL] = rev
        pod.metadata.labels[STS_POD_NAME_LABEL] = pod.metadata.name
        pod.metadata.labels[STS_OWNER_LABEL] = sts.metadata.name
        pod.metadata.annotations = dict(template.metadata.annotations or {})
        pod.metadata.owner_references = [OwnerReference(
            api_version=sts.api_version, kind=sts.kind, name=sts.metadata.name,
            uid=sts.metadata.uid, controller=True, block_owner_deletion=True)]
        pod.spec.hostname = pod.metadata.name
        if not pod.spec.subdomain:
            pod.spec.subdomain = sts.spec.service_name
        try:
            self.store.create(pod)
        except AlreadyExistsError:
            return
        self._ensure_pvcs(sts, pod)

    def _ensure_pvcs(self, sts: StatefulSet, pod: Pod) -> None:
        """PVCs from volumeClaimTemplates, named <claim>-<pod> (k8s STS
        convention; reference controller_utils.go:67-96 builds the same
        claims from lws.volumeClaimTemplates).  Retention policy: Delete
        whenScaled -> owned by the pod; Delete whenDeleted -> owned by the
        StatefulSet; Retain (default) -> standalone, survives both."""
        from ..api import serde as _serde

        created = self.store.try_get("Pod", pod.metadata.namespace,
                                     pod.metadata.name)
        if created is None:
            return
        policy = sts.spec.persistent_volume_claim_retention_policy
        for vct in sts.spec.volume_claim_templates or []:
            pvc = _serde.deep_copy(vct)
            pvc.metadata.name = f"{vct.metadata.name}-{pod.metadata.name}"
            pvc.metadata.namespace = sts.metadata.namespace
            pvc.metadata.labels = dict(vct.metadata.labels or {})
            pvc.metadata.labels[STS_OWNER_LABEL] = sts.metadata.name
            pvc.metadata.owner_references = []
            if policy is not None and policy.when_scaled == "Delete":
                pvc.metadata.owner_references = [OwnerReference(
                    api_version="v1", kind="Pod", name=created.metadata.name,
                    uid=created.metadata.uid, controller=False)]
            elif policy is not None and policy.when_deleted == "Delete":
                pvc.metadata.owner_references = [OwnerReference(
                    api_version=sts.api_version, kind=sts.kind,
                    name=sts.metadata.name, uid=sts.metadata.uid,
                    controller=False)]
            try:
                self.store.create(pvc)
            except AlreadyExistsError:
                pass

    def _delete_pod(self, pod: Pod) -> None:
        if pod.metadata.deletion_timestamp is not None:
            return
        try:
            self.store.delete("Pod", pod.metadata.namespace, pod.metadata.name,
                              propagation="Background")
        except NotFoundError:
            pass

    def _update_status(self, sts: StatefulSet, want_ordinals: list[int],
                       current_rev: str, update_rev: str) -> None:
        pods = self._owned_pods(sts)
        live = {o: p for o, p in pods.items()
                if p.metadata.deletion_timestamp is None and o in want_ordinals}
        fresh = self.store.try_get("StatefulSet", sts.metadata.namespace,
                                   sts.metadata.name)
        if fresh is None:
            return
        st = fresh.status
        st.observed_generation = fresh.metadata.generation
        st.replicas = len(live)
        st.ready_replicas = sum(1 for p in live.values() if pod_is_ready(p))
        st.available_replicas = st.ready_replicas
        st.updated_replicas = sum(
            1 for p in live.values()
            if p.metadata.labels.get(STS_REVISION_LABEL) == update_rev)
        st.current_replicas = sum(
            1 for p in live.values()
            if p.metadata.labels.get(STS_REVISION_LABEL) == current_rev)
        st.current_revision = current_rev
        st.update_revision = update_rev
        try:
            self.store.update_status(fresh)
        except NotFoundError:
            pass  # sts deleted concurrently


def statefulset_ready(sts: StatefulSet) -> bool:
    """StatefulsetReady equivalent (statefulset_utils.go:48-51)."""
    return (sts.status.observed_generation >= sts.metadata.generation
            and sts.status.available_replicas == sts.spec.replicas
            and sts.status.current_revision == sts.status.update_revision)
