"""Controller runtime: workqueues, reconcilers, manager.

The reference leans on controller-runtime v0.24 (workqueues, informer-fed
enqueue, single-reconcile-per-key, requeue-after).  This module provides the
same execution model in-process: each Controller owns a deduplicating
workqueue fed by store watch events and a worker thread that invokes
``reconcile(namespace, name)``; returning a float requeues after that many
seconds; raising requeues with backoff (matching conflict → requeue,
reference leaderworkerset_controller.go:198-200).
"""
from __future__ import annotations

import heapq
import logging
import threading
import time
import traceback
from typing import Callable, Optional

from .metrics import GLOBAL as metrics
from .store import ConflictError, Store

log = logging.getLogger("lws_amd")


class WorkQueue:
    """Deduplicating work queue with delayed adds (client-go workqueue)."""

    def __init__(self) -> None:
        self._cond = threading.Condition()
        self._queue: list[tuple[str, str]] = []
        self._dirty: set[tuple[str, str]] = set()
        self._processing: set[tuple[str, str]] = set()
        self._delayed: list[tuple[float, tuple[str, str]]] = []
        self._shutdown = False

    def add(self, key: tuple[str, str]) -> None:
        with self._cond:
            if key in self._dirty:
                return
            self._dirty.add(key)
            if key not in self._processing:
                self._queue.append(key)
                self._cond.notify()

    def add_after(self, key: tuple[str, str], delay: float) -> None:
        with self._cond:
            heapq.heappush(self._delayed, (time.monotonic() + delay, key))
            self._cond.notify()

    def get(self, timeout: Optional[float] = None) -> Optional[tuple[str, str]]:
        deadline = None if timeout is None else time.monotonic() + timeout
        with self._cond:
            while True:
                now = time.monotonic()
                while self._delayed and self._delayed[0][0] <= now:
                    _, key = heapq.heappop(self._delayed)
                    if key not in self._dirty:
                        self._dirty.add(key)
                        if key not in self._processing:
                            self._queue.append(key)
                if self._shutdown:
                    return None
                if self._queue:
                    key = self._queue.pop(0)
                    self._dirty.discard(key)
                    self._processing.add(key)
                    return key
                wait = None
                if self._delayed:
                    wait = max(0.0, self._delayed[0][0] - now)
                if deadline is not None:
                    rem = deadline - now
                    if rem <= 0:
                        return None
                    wait = rem if wait is None else min(wait, rem)
                self._cond.wait(wait if wait is not None else 0.2)

    def done(self, key: tuple[str, str]) -> None:
        with self._cond:
            self._processing.discard(key)
            if key in self._dirty:
                self._queue.append(key)
                self._cond.notify()

    def shutdown(self) -> None:
        with self._cond:
            self._shutdown = True
            self._cond.notify_all()

    def empty(self) -> bool:
        with self._cond:
            return not self._queue and not self._processing and not self._delayed


class Controller:
    """One reconcile loop over (namespace, name) keys."""

    def __init__(self, name: str,
                 reconcile: Callable[[str, str], Optional[float]]) -> None:
        self.name = name
        self.reconcile = reconcile
        self.queue = WorkQueue()
        self._threads: list[threading.Thread] = []
        self._stop = threading.Event()

    def enqueue(self, namespace: str, name: str) -> None:
        self.queue.add((namespace, name))

    def enqueue_after(self, namespace: str, name: str, delay: float) -> None:
        self.queue.add_after((namespace, name), delay)

    def start(self, workers: int = 1) -> None:
        for i in range(workers):
            t = threading.Thread(target=self._run, name=f"{self.name}-{i}",
                                 daemon=True)
            t.start()
            self._threads.append(t)

    def _run(self) -> None:
        while not self._stop.is_set():
            key = self.queue.get(timeout=0.5)
            if key is None:
                if self.queue._shutdown:
                    return
                continue
            ns, name = key
            t0 = time.perf_counter()
            try:
                result = self.reconcile(ns, name)
                if isinstance(result, (int, float)) and result > 0:
                    self.queue.add_after(key, float(result))
                    outcome = "requeue_after"
                else:
                    outcome = "success"
            except ConflictError:
                self.queue.add_after(key, 0.01)
                outcome = "conflict_requeue"
            except Exception:  # noqa: BLE001 — reconcile errors retry w/ backoff
                log.error("reconcile %s %s/%s failed:\n%s", self.name, ns, name,
                          traceback.format_exc())
                self.queue.add_after(key, 0.2)
                outcome = "error"
            finally:
                self.queue.done(key)
            metrics.observe(self.name, time.perf_counter() - t0, outcome)

    def stop(self) -> None:
        self._stop.set()
        self.queue.shutdown()
        for t in self._threads:
            t.join(timeout=5)


class Manager:
    """Holds the store and all controllers; wires watch→enqueue mappings
    (controller-runtime manager + SetupWithManager equivalent)."""

    def __init__(self, store: Optional[Store] = None) -> None:
        self.store = store or Store()
        self.controllers: list[Controller] = []
        self._watches: list[tuple[str, Callable[[str, object], None]]] = []

    def add_controller(self, ctrl: Controller) -> Controller:
        self.controllers.append(ctrl)
        return ctrl

    def watch(self, kind: str, ctrl: Controller,
              map_fn: Optional[Callable[[str, object], list[tuple[str, str]]]] = None
              ) -> None:
        def handler(event: str, obj: object) -> None:
            if map_fn is None:
                ctrl.enqueue(obj.metadata.namespace, obj.metadata.name)  # type: ignore[attr-defined]
            else:
                for ns, name in map_fn(event, obj) or []:
                    ctrl.enqueue(ns, name)
        self.store.add_handler(kind, handler)
        self._watches.append((kind, handler))

    def start(self) -> None:
        for c in self.controllers:
            c.start()
        # initial informer sync: replay ADDED for every existing object so a
        # manager started over a pre-populated store (controller restart)
        # reconciles everything — this is what lets a restarted manager
        # resume a rollout mid-flight (controller-runtime cache sync)
        for kind, handler in self._watches:
            for obj in self.store.list(kind):
                handler("ADDED", obj)

    def stop(self) -> None:
        for c in self.controllers:
            c.stop()

    def wait_idle(self, timeout: float = 30.0, settle: float = 0.05) -> bool:
        """Wait until every controller queue is empty and stays empty for
        ``settle`` seconds (test helper; the real manager runs forever)."""
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            if all(c.queue.empty() for c in self.controllers):
                t0 = time.monotonic()
                stable = True
                while time.monotonic() - t0 < settle:
                    if not all(c.queue.empty() for c in self.controllers):
                        stable = False
                        break
                    time.sleep(0.005)
                if stable:
                    return True
            time.sleep(0.01)
        return False
