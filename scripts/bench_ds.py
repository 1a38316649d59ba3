#!/usr/bin/env python3
"""DisaggregatedSet bench — BASELINE.json config #4 on MI355X.

Measures the DS lifecycle end to end through the real control plane with
real engines per role (bf16, random-init, synthetic):

    create DS (prefill + decode roles) -> all roles Ready   (t_ready)
    template update -> coordinated lockstep rollout          (t_rollout)
    delete -> drained

The named config (prefill size=2 / decode size=6 Mixtral-8x7B) spans an
8-GPU node; on the 1-GPU gpurun box this runs the scaled-down 1-pod-per-
role variant (both engines share the GPU) with a model chosen to fit the
rollout's transitional engine count in 288 GB HBM.  The rollout metric is
control-plane + engine-bring-up bound, which this variant measures
faithfully; run with --model mixtral-8x7b --roles 1 on a multi-GPU box
for the full-size config.

    python scripts/bench_ds.py --model llama-3-8b --steps 3 --warmup 1
"""
from __future__ import annotations

import argparse
import json
import os
import statistics
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import torch  # noqa: E402

from lws_amd.api import disaggregatedset as dsapi  # noqa: E402
from lws_amd.cluster.node import NodeAgent, PodRuntime  # noqa: E402
from lws_amd.serving.engine import Engine, EngineConfig  # noqa: E402
from lws_amd.serving.runtime import (BENCH_KV_PAGES_ANNOTATION,  # noqa: E402
                                     BENCH_MODEL_ANNOTATION,
                                     BENCH_SEED_ANNOTATION)


class EnginePodRuntime(PodRuntime):
    """One TP=1 engine per pod, built synchronously at pod start.

    The DS roles are independent single-pod groups, so no collective
    rendezvous is needed: the node agent builds the shard (weights + KV
    pool + warmup) and marks the pod Ready — the same definition of Ready
    the flagship bench uses.
    """

    def __init__(self, device: str):
        self.device = device
        self.engines: dict[str, Engine] = {}

    def start(self, pod, agent: NodeAgent) -> None:
        ann = pod.metadata.annotations or {}
        cfg = EngineConfig(
            model=ann.get(BENCH_MODEL_ANNOTATION, "llama-tiny"),
            kv_pages=int(ann.get(BENCH_KV_PAGES_ANNOTATION, "64")),
            seed=int(ann.get(BENCH_SEED_ANNOTATION, "0")),
            device=self.device)
        eng = Engine(cfg)
        eng.load()
        # one warm decode proves the serving path end to end
        eng.generate([[1, 2, 3]], max_new_tokens=1)
        self.engines[pod.metadata.uid] = eng
        agent.mark_pod_ready(pod)

    def stop(self, pod, agent: NodeAgent) -> None:
        eng = self.engines.pop(pod.metadata.uid, None)
        if eng is not None:
            eng.unload()
        agent.finish_pod_teardown(pod)


def build_ds(args):
    from lws_amd.api.core import (Container, PodSpec, PodTemplateSpec,
                                  ResourceRequirements)
    from lws_amd.api.disaggregatedset import (DisaggregatedRoleSpec,
                                              DisaggregatedSet,
                                              DisaggregatedSetSpec)
    from lws_amd.api.leaderworkerset import (LeaderWorkerSetSpec,
                                             LeaderWorkerTemplate)
    from lws_amd.api.meta import ObjectMeta

    def role(name):
        tmpl = PodTemplateSpec(
            metadata=ObjectMeta(annotations={
                BENCH_MODEL_ANNOTATION: args.model,
                BENCH_KV_PAGES_ANNOTATION: str(args.kv_pages),
                BENCH_SEED_ANNOTATION: str(args.seed),
            }),
            spec=PodSpec(containers=[Container(
                name="engine", image="lws-amd-engine:bench",
                resources=ResourceRequirements(
                    requests={"amd.com/gpu": 1}))]))
        return DisaggregatedRoleSpec(
            name=name,
            spec=LeaderWorkerSetSpec(
                replicas=args.roles,
                leader_worker_template=LeaderWorkerTemplate(
                    size=args.size, worker_template=tmpl)))

    ds = DisaggregatedSet()
    ds.metadata = ObjectMeta(name="bench-ds", namespace="default")
    ds.spec = DisaggregatedSetSpec(roles=[role("prefill"), role("decode")])
    return ds


def wait_until(fn, timeout, desc, poll=0.002):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        r = fn()
        if r:
            return r
        time.sleep(poll)
    raise TimeoutError(f"bench_ds: timed out waiting for {desc}")


def ds_available(store):
    cur = store.try_get(dsapi.KIND, "default", "bench-ds")
    if cur is None:
        return None
    cond = next((c for c in cur.status.conditions if c.type == "Available"),
                None)
    return cur if cond is not None and cond.status == "True" else None


def run_cycle(cluster, args, seed_extra) -> dict:
    from lws_amd.utils import dsutils

    store = cluster.store
    ds = build_ds(args)
    for r in ds.spec.roles:
        r.spec.leader_worker_template.worker_template.metadata.annotations[
            BENCH_SEED_ANNOTATION] = str(args.seed + seed_extra)
    old_rev = dsutils.compute_revision(ds.spec.roles)

    t0 = time.perf_counter()
    store.create(ds)
    wait_until(lambda: ds_available(store), 3600, "DS Available")
    t_ready = time.perf_counter() - t0

    # lockstep rollout: bump a template annotation on BOTH roles -> new
    # revision -> executor drains old and brings up new in planner order
    def bump(o):
        for r in o.spec.roles:
            r.spec.leader_worker_template.worker_template.metadata \
                .annotations["bench.lws.amd.com/generation"] = \
                str(seed_extra + 1)
    from tests.conftest import retry_update
    t1 = time.perf_counter()
    retry_update(store, dsapi.KIND, "default", "bench-ds", bump)
    cur = store.try_get(dsapi.KIND, "default", "bench-ds")
    new_rev = dsutils.compute_revision(cur.spec.roles)
    assert new_rev != old_rev

    def rolled():
        c = ds_available(store)
        if c is None:
            return None
        lws_list = [o for o in store.list("LeaderWorkerSet", "default")
                    if o.metadata.deletion_timestamp is None]
        # exactly the 2 new-revision children remain, each fully ready
        if len(lws_list) != 2:
            return None
        for o in lws_list:
            if (o.metadata.labels or {}).get(
                    dsapi.REVISION_LABEL_KEY) != new_rev:
                return None
            if (o.status.ready_replicas or 0) < args.roles:
                return None
        return c
    wait_until(rolled, 3600, "lockstep rollout complete")
    t_rollout = time.perf_counter() - t1

    store.delete(dsapi.KIND, "default", "bench-ds", propagation="Background")
    wait_until(lambda: not store.list("Pod", "default"), 600, "pods drained")
    return {"time_to_ready_s": t_ready, "rollout_s": t_rollout}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama-3-8b")
    p.add_argument("--steps", type=int, default=3)
    p.add_argument("--warmup", type=int, default=1)
    p.add_argument("--roles", type=int, default=1,
                   help="replicas per role")
    p.add_argument("--size", type=int, default=1)
    p.add_argument("--kv-pages", type=int, default=64)
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--device", default=None)
    args = p.parse_args()
    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")

    from lws_amd.cluster.cluster import LwsCluster, make_nodes
    runtime = EnginePodRuntime(device)
    nodes = make_nodes(1, gpus_per_node=64, topology_per_node=False)
    cluster = LwsCluster(nodes=nodes,
                         runtime_factory=lambda n: runtime).start()
    try:
        for i in range(args.warmup):
            run_cycle(cluster, args, seed_extra=i)
        results = [run_cycle(cluster, args, seed_extra=100 + i)
                   for i in range(args.steps)]
    finally:
        cluster.stop()

    ready = sorted(r["time_to_ready_s"] for r in results)
    roll = sorted(r["rollout_s"] for r in results)
    out = {
        "metric": "DS 2-role time-to-ready + lockstep rollout p50",
        "value": round(statistics.median(ready) * 1e3, 2),
        "unit": "ms",
        "steps": args.steps,
        "warmup": args.warmup,
        "higher_is_better": False,
        "dtype": "bf16",
        "data": "synthetic",
        "config": {
            "model": args.model,
            "roles": {"prefill": args.roles, "decode": args.roles},
            "size": args.size,
            "device": device,
            "time_to_ready_ms_all": [round(v * 1e3, 2) for v in ready],
            "rollout_ms_p50": round(statistics.median(roll) * 1e3, 2),
            "rollout_ms_all": [round(v * 1e3, 2) for v in roll],
        },
    }
    print(json.dumps(out), flush=True)


if __name__ == "__main__":
    main()
