"""DS bench harness dry-run on CPU (the driver-facing scripts/bench_ds.py
path: EnginePodRuntime + lockstep rollout measurement with tiny models)."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_ds_cpu_cycle():
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "scripts", "bench_ds.py"),
         "--model", "llama-tiny", "--steps", "1", "--warmup", "0",
         "--device", "cpu"],
        cwd=REPO, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["metric"].startswith("DS 2-role")
    assert d["value"] > 0
    assert d["config"]["rollout_ms_p50"] > 0
