"""Reconcile metrics — controller-runtime's Prometheus surface, in-process.

The reference exposes controller-runtime's built-in registry
(controller_runtime_reconcile_total / _errors_total / _time_seconds) on
the metrics endpoint (cmd/main.go:341-348) and registers no custom
metrics.  This module provides the same three families for the lws_amd
controllers; the API server renders them on /metrics.
"""
from __future__ import annotations

import threading


class ReconcileMetrics:
    BUCKETS = (0.0005, 0.001, 0.0025, 0.005, 0.01, 0.025, 0.05, 0.1,
               0.25, 0.5, 1.0, 2.5, 5.0, 10.0)

    def __init__(self) -> None:
        self._lock = threading.Lock()
        # controller -> {result -> count}
        self.total: dict[str, dict[str, int]] = {}
        self.errors: dict[str, int] = {}
        # controller -> (bucket_counts, sum, count)
        self.latency: dict[str, tuple[list[int], float, int]] = {}

    def observe(self, controller: str, seconds: float, result: str) -> None:
        with self._lock:
            self.total.setdefault(controller, {})
            self.total[controller][result] = \
                self.total[controller].get(result, 0) + 1
            if result == "error":
                self.errors[controller] = self.errors.get(controller, 0) + 1
            buckets, s, c = self.latency.get(
                controller, ([0] * len(self.BUCKETS), 0.0, 0))
            buckets = list(buckets)
            for i, b in enumerate(self.BUCKETS):
                if seconds <= b:
                    buckets[i] += 1
            self.latency[controller] = (buckets, s + seconds, c + 1)

    def reset(self) -> None:
        with self._lock:
            self.total.clear()
            self.errors.clear()
            self.latency.clear()

    def render(self) -> list[str]:
        with self._lock:
            lines = ["# TYPE lws_amd_reconcile_total counter"]
            for ctrl in sorted(self.total):
                for result, n in sorted(self.total[ctrl].items()):
                    lines.append(
                        f'lws_amd_reconcile_total{{controller="{ctrl}",'
                        f'result="{result}"}} {n}')
            lines.append("# TYPE lws_amd_reconcile_errors_total counter")
            for ctrl, n in sorted(self.errors.items()):
                lines.append(
                    f'lws_amd_reconcile_errors_total{{controller="{ctrl}"}}'
                    f' {n}')
            lines.append("# TYPE lws_amd_reconcile_time_seconds histogram")
            for ctrl in sorted(self.latency):
                buckets, s, c = self.latency[ctrl]
                for b, n in zip(self.BUCKETS, buckets):
                    lines.append(
                        f'lws_amd_reconcile_time_seconds_bucket{{'
                        f'controller="{ctrl}",le="{b}"}} {n}')
                lines.append(
                    f'lws_amd_reconcile_time_seconds_bucket{{'
                    f'controller="{ctrl}",le="+Inf"}} {c}')
                lines.append(
                    f'lws_amd_reconcile_time_seconds_sum{{'
                    f'controller="{ctrl}"}} {s:.6f}')
                lines.append(
                    f'lws_amd_reconcile_time_seconds_count{{'
                    f'controller="{ctrl}"}} {c}')
            return lines


GLOBAL = ReconcileMetrics()
