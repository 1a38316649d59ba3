"""Owner-reference garbage collector.

Closes the orphan race the cascade alone cannot: a reconciler holding a
stale copy of a just-deleted owner can re-create a dependent (e.g. the LWS
reconciler re-applying the leader StatefulSet right after the LWS was
deleted).  Kubernetes' garbage collector handles this by deleting objects
whose controller ownerReference points at a uid that no longer exists;
this controller does the same, event-driven on deletions plus a periodic
sweep.
"""
from __future__ import annotations

from typing import Optional

from .controller import Controller, Manager
from .store import NotFoundError, Store


class GarbageCollector:
    def __init__(self, manager: Manager, resync_seconds: float = 2.0):
        self.store: Store = manager.store
        self.resync = resync_seconds
        self.ctrl = Controller("garbage-collector", self.reconcile)
        manager.add_controller(self.ctrl)
        # any deletion can orphan dependents created concurrently
        self.store.add_handler(None, self._on_event)
        self.ctrl.enqueue("", "sweep")

    def _on_event(self, event: str, obj) -> None:
        if event == "DELETED":
            self.ctrl.enqueue("", "sweep")

    def reconcile(self, namespace: str, name: str) -> Optional[float]:
        live = {}
        objects = []
        for kind, ns, obj_name in self.store.snapshot_keys():
            obj = self.store.try_get(kind, ns, obj_name)
            if obj is None:
                continue
            live[obj.metadata.uid] = obj
            objects.append(obj)
        for obj in objects:
            ctrl_ref = next((r for r in obj.metadata.owner_references
                             if r.controller), None)
            if ctrl_ref is None:
                continue
            if obj.metadata.deletion_timestamp is not None:
                continue
            owner = live.get(ctrl_ref.uid)
            # collect when the controller owner is gone, OR is itself being
            # deleted (kube foregroundDeletion GC semantics): a dependent
            # created in the TOCTOU window after the cascade pass — e.g. a
            # StatefulSet reconcile re-creating a pod for a just-marked-
            # deleting STS — would otherwise block the foreground chain
            # forever
            if owner is not None and owner.metadata.deletion_timestamp is None:
                continue
            try:
                self.store.delete(type(obj).__name__
                                  if not hasattr(obj, "kind") else obj.kind,
                                  obj.metadata.namespace, obj.metadata.name,
                                  propagation="Background")
            except NotFoundError:
                pass
        return self.resync  # periodic safety-net sweep
