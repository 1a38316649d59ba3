"""Ownership-guarded CRUD of child LeaderWorkerSets.

Behavioral port of reference
pkg/controllers/disaggregatedset/lws_manager.go, including the #981
foreign-object protection: an LWS occupying an expected name that is not
controller-owned by this DisaggregatedSet is refused (Create/Scale) or
treated as absent (Get/List).
"""
from __future__ import annotations

from typing import Optional

from ...api import disaggregatedset as dsapi
from ...api import leaderworkerset as lwsapi
from ...api import serde
from ...api.leaderworkerset import LeaderWorkerSet
from ...api.meta import OwnerReference
from ...cluster.store import AlreadyExistsError, NotFoundError, Store
from ...utils import dsutils


def _controlled_by(obj, ds) -> bool:
    ref = next((r for r in obj.metadata.owner_references if r.controller),
               None)
    return ref is not None and ref.uid == ds.metadata.uid


class ForeignObjectError(RuntimeError):
    pass


class LeaderWorkerSetManager:
    def __init__(self, store: Store):
        self.store = store

    def create(self, ds, role: str, slice_: int, config, revision: str,
               labels: dict[str, str], replicas: int) -> None:
        """lws_manager.go:59-125 — label injection + placement affinity +
        adopt-or-refuse."""
        lws_name = dsutils.generate_name(ds.metadata.name, slice_, revision,
                                         role)
        spec = serde.deep_copy(config.spec)
        spec.replicas = replicas
        wt = spec.leader_worker_template.worker_template
        wt.metadata.labels = {**(wt.metadata.labels or {}), **labels}
        dsutils.set_placement_affinities(wt.spec, ds.metadata.name, slice_,
                                         ds.spec.placement_policy)
        lt = spec.leader_worker_template.leader_template
        if lt is not None:
            lt.metadata.labels = {**(lt.metadata.labels or {}), **labels}
            dsutils.set_placement_affinities(lt.spec, ds.metadata.name,
                                             slice_, ds.spec.placement_policy)

        lws = LeaderWorkerSet()
        lws.metadata.name = lws_name
        lws.metadata.namespace = ds.metadata.namespace
        lws.metadata.labels = {**(config.metadata.labels or {}), **labels}
        lws.metadata.annotations = dict(config.metadata.annotations or {})
        lws.metadata.owner_references = [OwnerReference(
            api_version=dsapi.API_VERSION, kind=dsapi.KIND,
            name=ds.metadata.name, uid=ds.metadata.uid, controller=True)]
        lws.spec = spec
        try:
            self.store.create(lws)
        except AlreadyExistsError:
            existing = self.store.try_get(lwsapi.KIND, ds.metadata.namespace,
                                          lws_name)
            if existing is not None and not _controlled_by(existing, ds):
                raise ForeignObjectError(
                    f"LeaderWorkerSet {lws_name} exists but is not controlled "
                    f"by DisaggregatedSet {ds.metadata.name}; refusing to "
                    "adopt it")

    def scale(self, ds, name: str, replicas: int) -> None:
        """lws_manager.go:132-153."""
        lws = self.store.try_get(lwsapi.KIND, ds.metadata.namespace, name)
        if lws is None:
            raise NotFoundError(f"LeaderWorkerSet {name}")
        if not _controlled_by(lws, ds):
            raise ForeignObjectError(
                f"LeaderWorkerSet {name} exists but is not controlled by "
                f"DisaggregatedSet {ds.metadata.name}; refusing to scale it")
        if dsutils.get_lws_replicas(lws) == replicas:
            return
        lws.spec.replicas = replicas
        self.store.update(lws)

    def get(self, ds, name: str) -> Optional[LeaderWorkerSet]:
        """lws_manager.go:161-178 — foreign-owned treated as absent."""
        lws = self.store.try_get(lwsapi.KIND, ds.metadata.namespace, name)
        if lws is None or not _controlled_by(lws, ds):
            return None
        return lws

    def list(self, ds, slice_: int, role: str = "") -> list[LeaderWorkerSet]:
        """lws_manager.go:184-205."""
        selector = {dsapi.SET_NAME_LABEL_KEY: ds.metadata.name}
        if role:
            selector[dsapi.ROLE_LABEL_KEY] = role
        out = []
        for lws in self.store.list(lwsapi.KIND, ds.metadata.namespace,
                                   label_selector=selector):
            if _controlled_by(lws, ds) and \
                    dsutils.slice_label_matches(lws.metadata.labels, slice_):
                out.append(lws)
        return out

    def get_for_role(self, ds, slice_: int, revision: str,
                     role: str) -> Optional[LeaderWorkerSet]:
        """lws_manager.go:213-222 — legacy-name fallback for slice 0."""
        lws = self.get(ds, dsutils.generate_name(ds.metadata.name, slice_,
                                                 revision, role))
        if lws is not None or slice_ != 0:
            return lws
        return self.get(ds, dsutils.generate_legacy_name(ds.metadata.name,
                                                         revision, role))

    def delete(self, namespace: str, name: str) -> None:
        try:
            self.store.delete(lwsapi.KIND, namespace, name,
                              propagation="Background")
        except NotFoundError:
            pass

    def get_revision_roles_list(self, ds, slice_: int, revision: str):
        """lws_manager.go:253-284 — (oldRevisions, newRevision)."""
        lws_list = self.list(ds, slice_)
        old = [l for l in lws_list
               if (l.metadata.labels or {}).get(dsapi.REVISION_LABEL_KEY) != revision]
        new = [l for l in lws_list
               if (l.metadata.labels or {}).get(dsapi.REVISION_LABEL_KEY) == revision]
        old_revisions = dsutils.group_by_revision(old)
        new_grouped = dsutils.group_by_revision(new)
        new_revision = new_grouped[0] if new_grouped else None
        return old_revisions, new_revision

    def set_initial_replicas(self, namespace: str, name: str,
                             replicas: int) -> Optional[int]:
        """lws_manager.go:314-344."""
        lws = self.store.try_get(lwsapi.KIND, namespace, name)
        if lws is None:
            raise NotFoundError(f"LeaderWorkerSet {name}")
        old_value = dsutils.get_initial_replicas(lws)
        if old_value is not None and old_value == replicas:
            return old_value
        dsutils.set_initial_replicas(lws, replicas)
        self.store.update(lws)
        return old_value
