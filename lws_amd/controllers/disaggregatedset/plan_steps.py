"""Offline rollout-plan previewer over the DS planner.

Equivalent of reference hack/plan-steps/main.go: simulate a full
N-dimensional rollout without a cluster.

    python -m lws_amd.controllers.disaggregatedset.plan_steps \\
        --source '[2,6]' --target '[4,12]' --surge '[1,1]' --unavailable '[0,0]'
"""
from __future__ import annotations

import argparse
import json
import sys

from .planner import (compute_all_steps, compute_total_steps,
                      default_rolling_update_config)


def main(argv=None) -> int:
    p = argparse.ArgumentParser(prog="plan-steps")
    p.add_argument("--source", required=True,
                   help="JSON list of per-role current (old) replicas")
    p.add_argument("--target", required=True,
                   help="JSON list of per-role target replicas")
    p.add_argument("--surge", help="JSON list of per-role maxSurge (default 1)")
    p.add_argument("--unavailable",
                   help="JSON list of per-role maxUnavailable (default 0)")
    args = p.parse_args(argv)

    source = json.loads(args.source)
    target = json.loads(args.target)
    if len(source) != len(target):
        sys.exit("error: source and target must have the same number of roles")
    n = len(source)
    config = default_rolling_update_config(n)
    if args.surge:
        for i, v in enumerate(json.loads(args.surge)):
            config[i].max_surge = int(v)
    if args.unavailable:
        for i, v in enumerate(json.loads(args.unavailable)):
            config[i].max_unavailable = int(v)
    for i, c in enumerate(config):
        if c.max_surge == 0 and c.max_unavailable == 0:
            sys.exit(f"error: role {i}: maxSurge and maxUnavailable cannot "
                     "both be 0")

    steps = compute_all_steps(source, target, config)
    total = compute_total_steps(source, target, config)
    print(f"ideal batches: {total}; reconcile steps: {len(steps) - 1}")
    print(f"{'step':>4}  {'old':<20} {'new':<20}")
    for i, s in enumerate(steps):
        print(f"{i:>4}  {str(s.past):<20} {str(s.new):<20}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
