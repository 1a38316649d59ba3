#!/usr/bin/env python3
"""Probe decode-shape GEMM throughput: hipBLASLt heuristic vs TunableOp."""
import argparse
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch

SHAPES_8B = [("qkv", 6144, 4096), ("o", 4096, 4096),
             ("gate_up", 28672, 4096), ("down", 4096, 14336),
             ("lm_head", 128256, 4096)]
SHAPES_70B_TP1 = [("qkv", 10240, 8192), ("o", 8192, 8192),
                  ("gate_up", 57344, 8192), ("down", 8192, 28672),
                  ("lm_head", 128256, 8192)]


def bench_shape(M, N, K, iters=50):
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
    wt = w.t()
    for _ in range(5):
        _ = x @ wt
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        _ = x @ wt
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    gb = (M * K + N * K + M * N) * 2 / 1e9
    return dt * 1e6, gb / dt / 1e3  # us, TB/s effective


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--m", type=int, default=32)
    p.add_argument("--model", default="8b")
    p.add_argument("--tunable", action="store_true")
    args = p.parse_args()
    if args.tunable:
        torch.cuda.tunable.enable(True)
        torch.cuda.tunable.tuning_enable(True)
    shapes = SHAPES_8B if args.model == "8b" else SHAPES_70B_TP1
    mode = "tunable" if args.tunable else "heuristic"
    for name, N, K in shapes:
        us, tbs = bench_shape(args.m, N, K)
        print(f"[{mode}] M={args.m} {name:8s} N={N:6d} K={K:6d}: "
              f"{us:8.1f} us  {tbs:5.2f} TB/s", flush=True)
    if args.tunable:
        torch.cuda.tunable.write_file("gpurun_out/tunableop_results.csv")


if __name__ == "__main__":
    main()
