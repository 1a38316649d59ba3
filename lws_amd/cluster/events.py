"""Kubernetes-style Events — the primary observability channel.

The reference emits rich Events from every controller (SURVEY.md §5:
GroupsProgressing/GroupsUpdating/CreatingRevision/FailedCreate/
RollingUpdateStarted/ScalingUp/RecreateGroup/...).  lws_amd records the
same reasons as Event objects in the store, deduplicated with a count like
kube's event aggregation, listable via the API server / client.
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field

from ..api.meta import ObjectMeta


@dataclass
class ObjectReference:
    kind: str = ""
    namespace: str = ""
    name: str = ""
    uid: str = ""


@dataclass
class Event:
    api_version: str = "v1"
    kind: str = "Event"
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    involved_object: ObjectReference = field(default_factory=ObjectReference)
    reason: str = ""
    message: str = ""
    type: str = "Normal"  # Normal | Warning
    count: int = 1
    first_timestamp: float = 0.0
    last_timestamp: float = 0.0


class EventRecorder:
    """events.EventRecorder equivalent backed by the object store.

    Events expire after ``ttl_seconds`` (kube's default event TTL is 1 h)
    and the total is bounded by ``max_events`` — pruning runs inline on
    record, oldest-last-timestamp first, so an event storm cannot grow
    the store without bound."""

    def __init__(self, store, ttl_seconds: float = 3600.0,
                 max_events: int = 1000):
        self.store = store
        self.ttl = ttl_seconds
        self.max_events = max_events
        self._last_prune = 0.0

    def _prune(self, now: float) -> None:
        if now - self._last_prune < 5.0:
            return
        self._last_prune = now
        from .store import ApiError

        events = self.store.list("Event")
        expired = [e for e in events if now - e.last_timestamp > self.ttl]
        keep = [e for e in events if now - e.last_timestamp <= self.ttl]
        if len(keep) > self.max_events:
            keep.sort(key=lambda e: e.last_timestamp)
            expired.extend(keep[:len(keep) - self.max_events])
        for e in expired:
            try:
                self.store.delete("Event", e.metadata.namespace,
                                  e.metadata.name)
            except ApiError:
                pass

    def eventf(self, obj, event_type: str, reason: str, message: str) -> None:
        from .store import AlreadyExistsError, ConflictError, NotFoundError

        ref = ObjectReference(kind=getattr(obj, "kind", ""),
                              namespace=obj.metadata.namespace,
                              name=obj.metadata.name, uid=obj.metadata.uid)
        # aggregate by (object, reason) like kube's event correlation
        name = f"{ref.name}.{reason.lower()}"
        now = time.time()
        self._prune(now)
        for _ in range(5):
            existing = self.store.try_get("Event", ref.namespace, name)
            if existing is None:
                ev = Event(involved_object=ref, reason=reason, message=message,
                           type=event_type, count=1, first_timestamp=now,
                           last_timestamp=now)
                ev.metadata.name = name
                ev.metadata.namespace = ref.namespace
                try:
                    self.store.create(ev)
                    return
                except AlreadyExistsError:
                    continue
            existing.count += 1
            existing.message = message
            existing.last_timestamp = now
            try:
                self.store.update(existing)
                return
            except (ConflictError, NotFoundError):
                continue


class NullRecorder:
    def eventf(self, obj, event_type, reason, message) -> None:
        pass
