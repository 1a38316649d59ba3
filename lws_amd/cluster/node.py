"""Nodes and node agents ("nodelets") for the lws_amd cluster substrate.

A Node models one host (for MI355X deployments: one 8-GPU xGMI island node,
or one GPU exposed as a schedulable unit).  The NodeAgent drives bound pods
through their lifecycle via a pluggable PodRuntime:

 - FakeRuntime: instant (or delayed) readiness — used by the CPU test suite
   the way the reference's integration tests manually simulate kubelet
   behavior (reference test/testutils/util.go:58-140)
 - external runtimes (see lws_amd.serving.agent) launch real engine
   processes on GPUs
"""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Optional

from ..api.core import ContainerState, ContainerStatus, Pod, pod_requests_amd_gpus
from ..api.meta import IntOrString, ObjectMeta, new_condition
from .controller import Controller, Manager
from .store import ConflictError, NotFoundError

RUNTIME_FINALIZER = "lws.amd.com/pod-runtime"


@dataclass
class Node:
    api_version: str = "v1"
    kind: str = "Node"
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    capacity: dict[str, IntOrString] = field(default_factory=dict)
    address: str = "127.0.0.1"


class PodRuntime:
    """Interface for executing pod workloads on a node."""

    def start(self, pod: Pod, agent: "NodeAgent") -> None:
        raise NotImplementedError

    def stop(self, pod: Pod, agent: "NodeAgent") -> None:
        raise NotImplementedError


class FakeRuntime(PodRuntime):
    """Marks containers running+ready after ``ready_delay`` seconds."""

    def __init__(self, ready_delay: float = 0.0):
        self.ready_delay = ready_delay
        self.started: set[str] = set()
        self._lock = threading.Lock()

    def start(self, pod: Pod, agent: "NodeAgent") -> None:
        with self._lock:
            if pod.metadata.uid in self.started:
                return
            self.started.add(pod.metadata.uid)
        if self.ready_delay > 0:
            timer = threading.Timer(self.ready_delay, agent.mark_pod_ready, [pod])
            timer.daemon = True
            timer.start()
        else:
            agent.mark_pod_ready(pod)

    def stop(self, pod: Pod, agent: "NodeAgent") -> None:
        with self._lock:
            self.started.discard(pod.metadata.uid)
        agent.finish_pod_teardown(pod)


class NodeAgent:
    """Watches pods bound to its node; drives status via the runtime."""

    def __init__(self, manager: Manager, node: Node,
                 runtime: Optional[PodRuntime] = None) -> None:
        self.store = manager.store
        self.node = node
        self.runtime = runtime or FakeRuntime()
        self.ctrl = Controller(f"nodelet-{node.metadata.name}", self.reconcile)
        manager.add_controller(self.ctrl)
        manager.watch("Pod", self.ctrl, self._map_pod)

    def _map_pod(self, event: str, pod) -> list[tuple[str, str]]:
        if pod.node_name == self.node.metadata.name:
            return [(pod.metadata.namespace, pod.metadata.name)]
        return []

    def reconcile(self, namespace: str, name: str) -> Optional[float]:
        pod = self.store.try_get("Pod", namespace, name)
        if pod is None or pod.node_name != self.node.metadata.name:
            return None
        if pod.metadata.deletion_timestamp is not None:
            self.runtime.stop(pod, self)
            return None
        if RUNTIME_FINALIZER not in pod.metadata.finalizers:
            self.store.add_finalizer("Pod", namespace, name, RUNTIME_FINALIZER)
        self.runtime.start(pod, self)
        return None

    # -- callbacks for runtimes -----------------------------------------
    def mark_pod_ready(self, pod: Pod) -> None:
        self._set_status(pod, phase="Running", ready=True)

    def mark_pod_running(self, pod: Pod) -> None:
        self._set_status(pod, phase="Running", ready=False)

    def mark_container_restarted(self, pod: Pod) -> None:
        """Simulate a container restart (failure-detection path)."""
        for _ in range(20):
            cur = self.store.try_get("Pod", pod.metadata.namespace,
                                     pod.metadata.name)
            if cur is None:
                return
            if not cur.status.container_statuses:
                self._set_status(cur, phase="Running", ready=False)
                continue
            for cs in cur.status.container_statuses:
                cs.restart_count += 1
                cs.ready = False
            try:
                self.store.update_status(cur)
                return
            except ConflictError:
                continue

    def finish_pod_teardown(self, pod: Pod) -> None:
        self.store.remove_finalizer("Pod", pod.metadata.namespace,
                                    pod.metadata.name, RUNTIME_FINALIZER)

    def _set_status(self, pod: Pod, phase: str, ready: bool) -> None:
        for _ in range(20):
            cur = self.store.try_get("Pod", pod.metadata.namespace,
                                     pod.metadata.name)
            if cur is None or cur.node_name != self.node.metadata.name:
                return
            cur.status.phase = phase
            cur.status.host_ip = self.node.address
            cur.status.pod_ip = self.node.address
            cur.status.node_name = self.node.metadata.name
            names = [c.name for c in cur.spec.containers] or ["main"]
            cur.status.container_statuses = [
                ContainerStatus(name=n, ready=ready, restart_count=rc,
                                started=True,
                                state=ContainerState(state="running"))
                for n, rc in ((n, self._restart_count(cur, n)) for n in names)]
            conds = [c for c in cur.status.conditions if c.type != "Ready"]
            conds.append(new_condition("Ready", "True" if ready else "False",
                                       "KubeletReady", ""))
            cur.status.conditions = conds
            try:
                self.store.update_status(cur)
                return
            except ConflictError:
                time.sleep(0.002)

    @staticmethod
    def _restart_count(pod: Pod, container_name: str) -> int:
        for cs in pod.status.container_statuses:
            if cs.name == container_name:
                return cs.restart_count
        return 0


def node_gpu_capacity(node: Node) -> int:
    try:
        return int(node.capacity.get("amd.com/gpu", 0))
    except (TypeError, ValueError):
        return 0


def pod_gpu_request(pod: Pod) -> int:
    return pod_requests_amd_gpus(pod.spec)
