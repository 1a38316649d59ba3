"""Per-(slice, revision, role) headless Services with revision-aware routing.

Behavioral port of reference
pkg/controllers/disaggregatedset/service_manager.go: the ``<lws>-prv``
service is created only once the revision is ready on ALL roles; drained
revisions' services are cleaned up; removed slices' services deleted;
legacy slice-agnostic services handled during migration.
"""
from __future__ import annotations

from ...api import disaggregatedset as dsapi
from ...api.core import Service, ServiceSpec
from ...api.meta import OwnerReference
from ...cluster.store import AlreadyExistsError, NotFoundError, Store
from ...utils import dsutils

SERVICE_SUFFIX = "-prv"


def revision_ready_on_all_roles(group: dsutils.RevisionRoles,
                                role_names: list[str]) -> bool:
    """service_manager.go:96-104 — every role has ReadyReplicas >= 1."""
    for name in role_names:
        lws = group.roles.get(name)
        if lws is None or lws.status.ready_replicas < 1:
            return False
    return True


class ServiceManager:
    def __init__(self, store: Store):
        self.store = store

    def reconcile_services(self, ds, slice_: int,
                           revision_roles: dsutils.RevisionRolesList,
                           target_revision: str) -> None:
        """service_manager.go:50-92."""
        role_names = dsutils.get_role_names(ds)
        target_group = next((g for g in revision_roles
                             if g.revision == target_revision), None)
        if target_group is None or \
                not revision_ready_on_all_roles(target_group, role_names):
            return
        for role_name in role_names:
            lws = target_group.roles.get(role_name)
            if lws is None:
                continue
            self._ensure_service(ds, lws)
        self._cleanup_drained_services(ds, slice_, revision_roles,
                                       target_revision, role_names)

    def _ensure_service(self, ds, lws) -> None:
        svc = self._build_service(ds, lws)
        try:
            self.store.create(svc)
        except AlreadyExistsError:
            pass

    def _build_service(self, ds, lws) -> Service:
        """service_manager.go:132-163 — selector mirrors the LWS's DS labels
        ({name, role, revision}[, slice])."""
        labels = lws.metadata.labels or {}
        selector = {
            dsapi.SET_NAME_LABEL_KEY: ds.metadata.name,
            dsapi.ROLE_LABEL_KEY: labels.get(dsapi.ROLE_LABEL_KEY, ""),
            dsapi.REVISION_LABEL_KEY: labels.get(dsapi.REVISION_LABEL_KEY, ""),
        }
        if dsutils.has_slice_label(labels):
            selector[dsapi.SLICE_LABEL_KEY] = labels[dsapi.SLICE_LABEL_KEY]
        svc = Service()
        svc.metadata.name = lws.metadata.name + SERVICE_SUFFIX
        svc.metadata.namespace = ds.metadata.namespace
        svc.metadata.labels = dict(selector)
        svc.metadata.owner_references = [OwnerReference(
            api_version=dsapi.API_VERSION, kind=dsapi.KIND,
            name=ds.metadata.name, uid=ds.metadata.uid, controller=True)]
        svc.spec = ServiceSpec(cluster_ip="None", selector=selector)
        return svc

    def _cleanup_drained_services(self, ds, slice_, revision_roles,
                                  target_revision, role_names) -> None:
        """service_manager.go:165-214."""
        ready = {g.revision for g in revision_roles
                 if revision_ready_on_all_roles(g, role_names)}
        ready.add(target_revision)
        for svc in self.store.list("Service", ds.metadata.namespace,
                                   label_selector={
                                       dsapi.SET_NAME_LABEL_KEY:
                                           ds.metadata.name}):
            if not dsutils.slice_label_matches(svc.metadata.labels, slice_):
                continue
            rev = (svc.metadata.labels or {}).get(dsapi.REVISION_LABEL_KEY, "")
            if rev not in ready:
                self._delete(svc.metadata.namespace, svc.metadata.name)

    def cleanup_removed_slices(self, ds, desired_slices: int) -> None:
        """service_manager.go:218-249."""
        for svc in self.store.list("Service", ds.metadata.namespace,
                                   label_selector={
                                       dsapi.SET_NAME_LABEL_KEY:
                                           ds.metadata.name}):
            v = (svc.metadata.labels or {}).get(dsapi.SLICE_LABEL_KEY, "")
            try:
                idx = int(v)
            except ValueError:
                continue
            if idx >= desired_slices:
                self._delete(svc.metadata.namespace, svc.metadata.name)

    def delete_legacy_service(self, ds, revision: str, role: str) -> None:
        """service_manager.go:254-265."""
        name = dsutils.generate_legacy_name(ds.metadata.name, revision,
                                            role) + SERVICE_SUFFIX
        self._delete(ds.metadata.namespace, name)

    def _delete(self, namespace: str, name: str) -> None:
        try:
            self.store.delete("Service", namespace, name,
                              propagation="Background")
        except NotFoundError:
            pass
