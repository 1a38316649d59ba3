"""Per-role /scale scalers for External roles.

Behavioral port of reference
pkg/controllers/disaggregatedset/scaler_manager.go: auto-create one
DisaggregatedSetRoleScaler per External role (seeded so a Static->External
flip doesn't drain a running role), GC scalers for roles no longer
External, adopt-or-warn on name conflicts, and write /scale status with a
leader-only selector so HPA's per-pod averaging divides by group count.
"""
from __future__ import annotations

from typing import Callable, Optional

from ...api import disaggregatedset as dsapi
from ...api import leaderworkerset as lwsapi
from ...api.disaggregatedset import (DisaggregatedSetRoleScaler,
                                     DisaggregatedSetRoleScalerSpec)
from ...api.meta import OwnerReference, new_condition
from ...cluster.store import AlreadyExistsError, ConflictError, NotFoundError, Store


def scaler_name(ds_name: str, role: str) -> str:
    return f"{ds_name}-{role}"


def _is_controlled_by(obj, uid: str) -> bool:
    return any(r.controller and r.uid == uid
               for r in obj.metadata.owner_references)


class ScalerManager:
    def __init__(self, store: Store):
        self.store = store

    def reconcile(self, ds, seed_for: Optional[Callable[[str], int]] = None
                  ) -> dict[str, DisaggregatedSetRoleScaler]:
        """scaler_manager.go:64-120."""
        external = {r.name for r in ds.spec.roles
                    if dsapi.role_scaling_mode(r) ==
                    dsapi.RoleScalingMode.External}
        existing: dict[str, DisaggregatedSetRoleScaler] = {}
        for s in self.store.list(dsapi.SCALER_KIND, ds.metadata.namespace,
                                 label_selector={dsapi.SET_NAME_LABEL_KEY:
                                                 ds.metadata.name}):
            if not _is_controlled_by(s, ds.metadata.uid):
                continue
            role = (s.metadata.labels or {}).get(dsapi.ROLE_LABEL_KEY, "")
            if role not in external:
                try:
                    self.store.delete(dsapi.SCALER_KIND, ds.metadata.namespace,
                                      s.metadata.name)
                except NotFoundError:
                    pass
                continue
            existing[role] = s
        for role in external:
            if role in existing:
                continue
            seed = seed_for(role) if seed_for is not None else 0
            s = self._create(ds, role, seed)
            if s is not None:
                existing[role] = s
        return existing

    def _create(self, ds, role: str, seed: int):
        """scaler_manager.go:122-163 — adopt-or-warn."""
        name = scaler_name(ds.metadata.name, role)
        scaler = DisaggregatedSetRoleScaler(
            spec=DisaggregatedSetRoleScalerSpec(replicas=seed))
        scaler.metadata.name = name
        scaler.metadata.namespace = ds.metadata.namespace
        scaler.metadata.labels = {dsapi.SET_NAME_LABEL_KEY: ds.metadata.name,
                                  dsapi.ROLE_LABEL_KEY: role}
        scaler.metadata.owner_references = [OwnerReference(
            api_version=dsapi.API_VERSION, kind=dsapi.KIND,
            name=ds.metadata.name, uid=ds.metadata.uid, controller=True,
            block_owner_deletion=True)]
        try:
            return self.store.create(scaler)
        except AlreadyExistsError:
            cur = self.store.try_get(dsapi.SCALER_KIND, ds.metadata.namespace,
                                     name)
            if cur is None:
                return None
            if not _is_controlled_by(cur, ds.metadata.uid):
                return None  # foreign object: warn and stay out (#981)
            return cur

    def write_status(self, ds, scalers: dict, observed: dict[str, int]) -> None:
        """scaler_manager.go:168-199."""
        for role, s in scalers.items():
            cur = self.store.try_get(dsapi.SCALER_KIND, ds.metadata.namespace,
                                     s.metadata.name)
            if cur is None:
                continue
            cur.status.replicas = observed.get(role, 0)
            cur.status.selector = (
                f"{dsapi.SET_NAME_LABEL_KEY}={ds.metadata.name},"
                f"{dsapi.ROLE_LABEL_KEY}={role},"
                f"{lwsapi.WORKER_INDEX_LABEL_KEY}=0")
            cur.status.observed_generation = cur.metadata.generation
            conds = [c for c in cur.status.conditions
                     if c.type != dsapi.DISAGGREGATED_SET_ROLE_SCALER_READY]
            conds.append(new_condition(
                dsapi.DISAGGREGATED_SET_ROLE_SCALER_READY, "True", "Bound",
                "Scaler bound to a live DisaggregatedSet role",
                cur.metadata.generation))
            cur.status.conditions = conds
            try:
                self.store.update_status(cur)
            except (NotFoundError, ConflictError):
                pass
