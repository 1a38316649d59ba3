"""DS planner tests: constraint properties over full simulated rollouts
(the reference validates via the ComputeAllSteps simulator —
planner_test.go — plus property assertions; hypothesis drives the space).
"""
import pytest
from hypothesis import given, settings, strategies as st

from lws_amd.controllers.disaggregatedset.planner import (
    RollingUpdateConfig, compute_all_steps, compute_next_step,
    compute_total_steps, default_rolling_update_config)


def check_rollout(initial_old, target, config):
    steps = compute_all_steps(initial_old, target, config)
    n = len(initial_old)
    # completes
    final = steps[-1]
    assert final.past == [0] * n, (steps, "old not drained")
    assert all(final.new[i] >= target[i] for i in range(n)), steps
    for prev, cur in zip(steps, steps[1:]):
        # (note: tryForceDrain legitimately changes old AND new in one step
        # — planner.go:300-322 — so no either-or assertion here)
        # monotonicity
        assert all(cur.past[i] <= prev.past[i] for i in range(n)), (prev, cur)
        assert all(cur.new[i] >= prev.new[i] for i in range(n)), (prev, cur)
        # surge bound (relative to the larger of initial/target during
        # scale-down rollouts, where old starts above target).  Orphan
        # prevention may deliberately pin a fully-drained role back at
        # old=1 (planner.go:294-297), exceeding the bound by one — allowed.
        for i in range(n):
            if target[i] > 0:
                slack = 1 if cur.past[i] == 1 else 0
                assert cur.past[i] + cur.new[i] <= \
                    max(initial_old[i], target[i]) + config[i].max_surge + \
                    slack, (prev, cur, i)
        # availability floor (roles not scaling up past initial)
        for i in range(n):
            if initial_old[i] >= target[i]:
                assert cur.past[i] + cur.new[i] >= \
                    target[i] - config[i].max_unavailable, (prev, cur, i)
        # orphan prevention: while any old-serving role is >0, no role that
        # started >0 may sit at 0 unless the new revision covers it
        serving = [cur.past[i] for i in range(n) if initial_old[i] > 0]
        if serving and any(v > 0 for v in serving):
            for i in range(n):
                if initial_old[i] > 0 and cur.past[i] == 0:
                    assert cur.new[i] >= target[i] - config[i].max_unavailable, \
                        (prev, cur, i)
    return steps


def test_basic_two_role():
    steps = check_rollout([2, 6], [2, 6], default_rolling_update_config(2))
    assert len(steps) > 2  # actually rolls, not a jump


def test_scale_changes_with_rollout():
    check_rollout([2, 6], [4, 12], default_rolling_update_config(2))
    check_rollout([4, 12], [2, 6], default_rolling_update_config(2))


def test_fresh_create():
    steps = check_rollout([0, 0], [3, 5], default_rolling_update_config(2))
    # pure scale-up: old never present
    assert all(s.past == [0, 0] for s in steps)


def test_drain_to_zero():
    steps = compute_all_steps([3, 5], [0, 0],
                              default_rolling_update_config(2))
    assert steps[-1].past == [0, 0]
    assert steps[-1].new == [0, 0]


def test_max_unavailable_mode():
    cfg = [RollingUpdateConfig(max_surge=0, max_unavailable=1)
           for _ in range(2)]
    check_rollout([2, 4], [2, 4], cfg)


def test_total_steps():
    cfg = default_rolling_update_config(2)
    assert compute_total_steps([2, 6], [2, 6], cfg) == 6
    cfg2 = [RollingUpdateConfig(max_surge=2, max_unavailable=0),
            RollingUpdateConfig(max_surge=1, max_unavailable=0)]
    assert compute_total_steps([4, 3], [4, 3], cfg2) == 3


def test_abnormal_state_corrected():
    # old above initial gets clamped before anything else
    step = compute_next_step([2, 2], [5, 2], [0, 0], [2, 2],
                             default_rolling_update_config(2))
    assert step.past == [2, 2]
    assert step.new == [0, 0]


@settings(max_examples=200, deadline=None)
@given(
    n=st.integers(min_value=1, max_value=4),
    data=st.data(),
)
def test_planner_properties(n, data):
    initial_old = [data.draw(st.integers(0, 12)) for _ in range(n)]
    target = [data.draw(st.integers(0, 12)) for _ in range(n)]
    config = [RollingUpdateConfig(
        max_surge=data.draw(st.integers(0, 3)),
        max_unavailable=data.draw(st.integers(0, 3)))
        for _ in range(n)]
    for c in config:
        if c.max_surge == 0 and c.max_unavailable == 0:
            c.max_surge = 1
    if all(t == 0 for t in target):
        steps = compute_all_steps(initial_old, target, config)
        assert steps[-1].past == [0] * n
        return
    check_rollout(initial_old, target, config)


def test_interleaved_drain_scale_up_completes():
    """Regression (hypothesis find): [12,12]->[1,12] with surge 1 takes 35
    steps — the old per-role-max simulator bound cut it off one short."""
    from lws_amd.controllers.disaggregatedset.planner import (
        RollingUpdateConfig, compute_all_steps)

    cfg = [RollingUpdateConfig(max_surge=1, max_unavailable=0)] * 2
    steps = compute_all_steps([12, 12], [1, 12], cfg)
    assert steps[-1].past == [0, 0]
    assert steps[-1].new == [1, 12]
