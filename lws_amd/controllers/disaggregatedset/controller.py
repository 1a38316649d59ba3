"""DisaggregatedSet reconciler — orchestrates per-slice reconciles.

Behavioral port of reference
pkg/controllers/disaggregatedset/disaggregatedset_controller.go:
revision compute, removed-slice cleanup, scaler reconcile (with seeding),
legacy slice-0 migration, per-slice rolling-vs-simple dispatch, drained
revision cleanup, service reconcile, scaler status and DS status.
"""
from __future__ import annotations

import time
from typing import Optional

from ...api import disaggregatedset as dsapi
from ...api import leaderworkerset as lwsapi
from ...api.meta import new_condition
from ...cluster.controller import Controller, Manager
from ...cluster.store import ConflictError, NotFoundError, Store
from ...utils import dsutils
from .executor import RollingUpdateExecutor, get_target_replicas, _is_external
from .lws_manager import LeaderWorkerSetManager
from .scaler_manager import ScalerManager
from .service_manager import ServiceManager


class DisaggregatedSetReconciler:
    def __init__(self, manager: Manager, recorder=None) -> None:
        self.store: Store = manager.store
        self.lws_manager = LeaderWorkerSetManager(self.store)
        self.service_manager = ServiceManager(self.store)
        self.scaler_manager = ScalerManager(self.store)
        self.executor = RollingUpdateExecutor(self.lws_manager,
                                              record=recorder)
        self.ctrl = Controller("disaggregatedset", self.reconcile)
        manager.add_controller(self.ctrl)
        manager.watch(dsapi.KIND, self.ctrl)
        manager.watch(lwsapi.KIND, self.ctrl, self._map_owned)
        manager.watch(dsapi.SCALER_KIND, self.ctrl, self._map_owned)

    def _map_owned(self, event: str, obj) -> list[tuple[str, str]]:
        name = (obj.metadata.labels or {}).get(dsapi.SET_NAME_LABEL_KEY)
        if name:
            return [(obj.metadata.namespace, name)]
        return []

    # ------------------------------------------------------------------
    def reconcile(self, namespace: str, name: str) -> Optional[float]:
        ds = self.store.try_get(dsapi.KIND, namespace, name)
        if ds is None or ds.metadata.deletion_timestamp is not None:
            return None

        # Step 1: target revision (disaggregatedset_controller.go:81)
        revision = dsutils.compute_revision(ds.spec.roles)
        slice_count = dsutils.get_slices(ds)

        # Step 2: removed-slice cleanup (:86)
        self._cleanup_removed_slices(ds, slice_count)

        # scalers (seeded; :98-105)
        seed_for = self._seed_for_role(ds)
        scalers = self.scaler_manager.reconcile(ds, seed_for)

        role_names = dsutils.get_role_names(ds)

        # legacy slice-0 migration (:121-125)
        if slice_count > 1:
            self._recreate_legacy_slice0(ds, revision, role_names)

        # Step 3: per-slice reconcile (:131)
        requeue: Optional[float] = None
        errors = []
        for slice_ in range(slice_count):
            try:
                r = self._reconcile_slice(ds, slice_, revision, role_names,
                                          scalers)
                if r is not None and (requeue is None or r < requeue):
                    requeue = r
            except Exception as e:  # noqa: BLE001 — per-slice isolation
                errors.append(e)

        self._update_scaler_status(ds, scalers)
        self._update_status(ds, role_names, revision, scalers)
        if errors:
            raise errors[0]
        return requeue

    # ------------------------------------------------------------------
    def _reconcile_slice(self, ds, slice_: int, revision: str,
                         role_names: list[str], scalers) -> Optional[float]:
        """disaggregatedset_controller.go:383-430."""
        self._cleanup_drained_lws(ds, slice_, revision)
        old_revisions, _ = self.lws_manager.get_revision_roles_list(
            ds, slice_, revision)
        total_old = sum(old_revisions.total_replicas(r) for r in role_names)
        if old_revisions and total_old > 0:
            result = self.executor.reconcile_rolling_update_new(
                ds, slice_, revision, scalers)
        else:
            result = self._reconcile_simple(ds, slice_, revision, scalers)
        all_lws = self.lws_manager.list(ds, slice_)
        revision_roles = dsutils.group_by_revision(all_lws)
        self.service_manager.reconcile_services(ds, slice_, revision_roles,
                                                revision)
        return result

    def _reconcile_simple(self, ds, slice_: int, revision: str,
                          scalers) -> Optional[float]:
        """disaggregatedset_controller.go:475-531."""
        for role, config in dsutils.get_role_configs(ds).items():
            existing = self.lws_manager.get_for_role(ds, slice_, revision,
                                                     role)
            current = dsutils.get_lws_replicas(existing) if existing else 0
            desired = get_target_replicas(ds, role, scalers, current)
            if existing is None:
                self.lws_manager.create(
                    ds, role=role, slice_=slice_, config=config,
                    revision=revision,
                    labels=dsutils.generate_labels(ds.metadata.name, slice_,
                                                   revision, role),
                    replicas=desired)
            elif dsutils.get_lws_replicas(existing) != desired:
                self.lws_manager.scale(ds, existing.metadata.name, desired)
        return None

    def _cleanup_drained_lws(self, ds, slice_: int, revision: str) -> None:
        """disaggregatedset_controller.go:537-589 — delete an old revision's
        LWS only when ALL its roles drained to 0."""
        by_revision: dict[str, list] = {}
        for lws in self.lws_manager.list(ds, slice_):
            rev = (lws.metadata.labels or {}).get(dsapi.REVISION_LABEL_KEY, "")
            if rev == revision:
                continue
            by_revision.setdefault(rev, []).append(lws)
        for rev, lws_list in by_revision.items():
            if all(dsutils.get_lws_replicas(l) == 0 for l in lws_list):
                for lws in lws_list:
                    self.lws_manager.delete(ds.metadata.namespace,
                                            lws.metadata.name)

    def _cleanup_removed_slices(self, ds, desired_slices: int) -> None:
        """disaggregatedset_controller.go:443-464."""
        for lws in self.lws_manager.list(ds, -1):
            v = (lws.metadata.labels or {}).get(dsapi.SLICE_LABEL_KEY, "")
            try:
                idx = int(v)
            except ValueError:
                continue
            if idx >= desired_slices:
                self.lws_manager.delete(ds.metadata.namespace,
                                        lws.metadata.name)
        self.service_manager.cleanup_removed_slices(ds, desired_slices)

    def _recreate_legacy_slice0(self, ds, revision: str,
                                role_names: list[str]) -> None:
        """disaggregatedset_controller.go:605-636 — service-first delete."""
        for role in role_names:
            lws = self.lws_manager.get(
                ds, dsutils.generate_legacy_name(ds.metadata.name, revision,
                                                 role))
            if lws is None or dsutils.has_slice_label(lws.metadata.labels):
                continue
            self.service_manager.delete_legacy_service(ds, revision, role)
            self.lws_manager.delete(ds.metadata.namespace, lws.metadata.name)

    def _seed_for_role(self, ds):
        """disaggregatedset_controller.go:329-351."""
        all_lws = self.lws_manager.list(ds, -1)
        sums: dict[str, int] = {}
        seen: set[str] = set()
        for lws in all_lws:
            role = (lws.metadata.labels or {}).get(dsapi.ROLE_LABEL_KEY, "")
            seen.add(role)
            sums[role] = sums.get(role, 0) + dsutils.get_lws_replicas(lws)

        def seed(role: str) -> int:
            if role not in seen:
                return 1
            return sums.get(role, 0)
        return seed

    def _update_scaler_status(self, ds, scalers) -> None:
        """disaggregatedset_controller.go:357-378."""
        if not scalers:
            return
        observed: dict[str, int] = {}
        for lws in self.lws_manager.list(ds, -1):
            role = (lws.metadata.labels or {}).get(dsapi.ROLE_LABEL_KEY, "")
            if role in scalers:
                observed[role] = observed.get(role, 0) + lws.status.replicas
        self.scaler_manager.write_status(ds, scalers, observed)

    # ------------------------------------------------------------------
    def _update_status(self, ds, role_names: list[str], revision: str,
                       scalers) -> None:
        """disaggregatedset_controller.go:162-225."""
        from ...api.disaggregatedset import RoleStatus

        slice_count = dsutils.get_slices(ds)
        role_statuses = []
        available = True
        for role in role_names:
            lws_list = self.lws_manager.list(ds, -1, role)
            rs = RoleStatus(name=role)
            for lws in lws_list:
                rs.replicas += lws.status.replicas
                rs.ready_replicas += lws.status.ready_replicas
                if (lws.metadata.labels or {}).get(
                        dsapi.REVISION_LABEL_KEY) == revision:
                    rs.updated_replicas += lws.status.updated_replicas
            role_statuses.append(rs)
            if _is_external(ds, role) and scalers.get(role) is None:
                available = False
                continue
            desired = get_target_replicas(ds, role, scalers, 0) * slice_count
            if rs.replicas != desired or rs.ready_replicas != desired or \
                    rs.updated_replicas != desired:
                available = False

        cur = self.store.try_get(dsapi.KIND, ds.metadata.namespace,
                                 ds.metadata.name)
        if cur is None:
            return
        changed = False
        from ...api import serde
        if serde.to_dict(cur.status.role_statuses) != \
                serde.to_dict(role_statuses):
            cur.status.role_statuses = role_statuses
            changed = True
        if self._set_condition(cur, available):
            changed = True
        if cur.status.observed_generation != cur.metadata.generation:
            cur.status.observed_generation = cur.metadata.generation
            changed = True
        if changed:
            try:
                self.store.update_status(cur)
            except (NotFoundError, ConflictError):
                pass

    @staticmethod
    def _set_condition(ds, available: bool) -> bool:
        """setDisaggregatedSetCondition: Available/Progressing exclusive
        pair; LastTransitionTime only on real flips."""
        if available:
            cond_type, reason, message = (
                dsapi.DisaggregatedSetConditionType.Available, "AllRolesReady",
                "All roles have reached their desired replica count, ready "
                "and updated to the current revision")
        else:
            cond_type, reason, message = (
                dsapi.DisaggregatedSetConditionType.Progressing,
                "RolloutInProgress",
                "Not all roles have reached their desired replica count, "
                "ready and updated to the current revision")
        other = (dsapi.DisaggregatedSetConditionType.Progressing if available
                 else dsapi.DisaggregatedSetConditionType.Available)
        changed = False
        found = False
        now = time.time()
        for c in ds.status.conditions:
            if c.type == cond_type:
                found = True
                if c.status != "True":
                    c.status = "True"
                    c.last_transition_time = now
                    c.reason, c.message = reason, message
                    c.observed_generation = ds.metadata.generation
                    changed = True
                elif c.observed_generation != ds.metadata.generation or \
                        c.reason != reason:
                    c.observed_generation = ds.metadata.generation
                    c.reason, c.message = reason, message
                    changed = True
            elif c.type == other:
                if c.status == "True":
                    c.status = "False"
                    c.last_transition_time = now
                    c.reason, c.message = reason, message
                    c.observed_generation = ds.metadata.generation
                    changed = True
                elif c.observed_generation != ds.metadata.generation:
                    c.observed_generation = ds.metadata.generation
                    changed = True
        if not found:
            ds.status.conditions.append(new_condition(
                cond_type, "True", reason, message, ds.metadata.generation))
            changed = True
        return changed
