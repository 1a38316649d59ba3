"""DisaggregatedSet rolling-update planner — stateless N-dimensional math.

Behavioral port of reference pkg/controllers/disaggregatedset/planner.go.

Algorithm: a linear scaling function approximates discrete steps of a
linear interpolation between initialOld and target replica counts:

    newAtStep(i) = ceil(i * target / totalSteps)            # 0 -> target
    oldAtStep(i) = initialOld - floor(i * initialOld / totalSteps)

The controller is stateless: the current step index is derived from
observed replicas, then the next step's target is computed directly.
Constraints: each step changes EITHER old OR new (never both); surge bound
old + new <= target + maxSurge per role; orphan prevention (never leave
one role at 0 while others still serve); all roles stay proportional.
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Optional


@dataclass
class UpdateStep:
    past: list[int]
    new: list[int]

    def as_tuple(self):
        return (tuple(self.past), tuple(self.new))


@dataclass
class RollingUpdateConfig:
    max_surge: int = 1
    max_unavailable: int = 0


def default_rolling_update_config(num_roles: int) -> list[RollingUpdateConfig]:
    return [RollingUpdateConfig(max_surge=1, max_unavailable=0)
            for _ in range(num_roles)]


def batch_size(max_surge: int, max_unavailable: int) -> int:
    if max_surge > 0:
        return max_surge
    return max(1, max_unavailable)


def compute_total_steps(initial_old: list[int], target: list[int],
                        config: list[RollingUpdateConfig]) -> int:
    """planner.go:71-82 — max over roles of ceil(max(old, target)/batch)."""
    total = 0
    for i in range(len(initial_old)):
        max_replicas = max(initial_old[i], target[i], 0)
        b = batch_size(config[i].max_surge, config[i].max_unavailable)
        total = max(total, (max_replicas + b - 1) // b)
    return total


def compute_next_new_replicas(target: list[int], current_new: list[int],
                              total_steps: int) -> list[int]:
    """planner.go:84-118."""
    n = len(target)
    if total_steps == 0:
        return list(target)

    def step_index(current: int, target_val: int) -> int:
        if target_val == 0:
            return total_steps
        return int(current * total_steps / target_val)

    min_step = min([total_steps] + [step_index(current_new[i], target[i])
                                    for i in range(n)])
    next_step = min_step + 1

    out = []
    for i in range(n):
        progress = next_step * target[i] / total_steps
        computed = min(math.ceil(progress), target[i])
        out.append(max(computed, current_new[i]))
    return out


def compute_next_old_replicas(initial_old: list[int], current_old: list[int],
                              total_steps: int) -> list[int]:
    """planner.go:120-153."""
    n = len(initial_old)
    if total_steps == 0:
        return [0] * n

    def step_index(removed: int, source: int) -> int:
        if source == 0:
            return 0
        return int(removed * total_steps / source)

    max_step = 0
    for i in range(n):
        if initial_old[i] == 0:
            continue
        max_step = max(max_step,
                       step_index(initial_old[i] - current_old[i],
                                  initial_old[i]))
    next_step = max_step + 1

    out = []
    for i in range(n):
        progress = next_step * initial_old[i] / total_steps
        computed = max(0, initial_old[i] - math.floor(progress))
        out.append(min(computed, current_old[i]))
    return out


def correct_abnormal_state(current_old, current_new,
                           initial_old) -> Optional[UpdateStep]:
    """planner.go:157-177 — old above initial (e.g. external scale-up of a
    draining revision) is clamped back."""
    expected = [min(initial_old[i], current_old[i])
                for i in range(len(initial_old))]
    if any(current_old[i] > expected[i] for i in range(len(initial_old))):
        return UpdateStep(past=expected, new=list(current_new))
    return None


def is_complete(current_old, current_new, target_new) -> bool:
    return all(current_old[i] == 0 and current_new[i] >= target_new[i]
               for i in range(len(current_old)))


def is_new_at_target(current_new, target_new) -> bool:
    return all(current_new[i] >= target_new[i]
               for i in range(len(current_new)))


def can_scale_up(current_old, next_new, target_new, config) -> bool:
    for i in range(len(current_old)):
        if target_new[i] == 0:
            continue
        if current_old[i] + next_new[i] > target_new[i] + config[i].max_surge:
            return False
    return True


def compute_min_old(initial_old, current_new, target_new, config) -> list[int]:
    """planner.go:207-215 — availability floor per role."""
    min_old = [0] * len(initial_old)
    for i in range(len(initial_old)):
        if initial_old[i] >= target_new[i]:
            min_old[i] = max(
                0, target_new[i] - config[i].max_unavailable - current_new[i])
    return min_old


def try_scale_up(current_old, current_new, next_new, target_new,
                 config) -> Optional[UpdateStep]:
    if not any(next_new[i] > current_new[i] for i in range(len(current_new))):
        return None
    if not can_scale_up(current_old, next_new, target_new, config):
        return None
    return UpdateStep(past=list(current_old), new=list(next_new))


def can_drain_all_to_zero(next_new, initial_old, target, config) -> bool:
    for i in range(len(target)):
        if initial_old[i] >= target[i]:
            if next_new[i] < target[i] - config[i].max_unavailable:
                return False
    return True


def apply_orphan_prevention(next_old, current_new, initial_old, target,
                            config) -> None:
    """planner.go:268-298 — never leave one role at 0 while others serve."""
    any_zero = False
    all_zero = True
    for i in range(len(next_old)):
        if initial_old[i] == 0:
            continue
        if next_old[i] == 0:
            any_zero = True
        else:
            all_zero = False
    if not any_zero or all_zero:
        return
    if can_drain_all_to_zero(current_new, initial_old, target, config):
        for i in range(len(next_old)):
            next_old[i] = 0
        return
    for i in range(len(next_old)):
        if next_old[i] == 0 and initial_old[i] > 0:
            next_old[i] = 1


def try_proportional_drain(initial_old, current_old, current_new, target_new,
                           min_old, total_steps, config) -> Optional[UpdateStep]:
    next_old = compute_next_old_replicas(initial_old, current_old, total_steps)
    for i in range(len(next_old)):
        next_old[i] = max(next_old[i], min_old[i])
    apply_orphan_prevention(next_old, current_new, initial_old, target_new,
                            config)
    if not any(next_old[i] < current_old[i] for i in range(len(next_old))):
        return None
    return UpdateStep(past=next_old, new=list(current_new))


def try_force_drain(current_old, next_new, initial_old, target_new,
                    config) -> Optional[UpdateStep]:
    """planner.go:300-322 — when blocked on surge, drain just enough old."""
    drained = [0] * len(current_old)
    needs = False
    for i in range(len(current_old)):
        max_old = target_new[i] + config[i].max_surge - next_new[i]
        drained[i] = max(0, min(current_old[i], max_old))
        if initial_old[i] >= target_new[i]:
            floor_ = max(0,
                         target_new[i] - config[i].max_unavailable - next_new[i])
            drained[i] = max(drained[i], floor_)
        if drained[i] < current_old[i]:
            needs = True
    if not needs:
        return None
    apply_orphan_prevention(drained, next_new, initial_old, target_new, config)
    return UpdateStep(past=drained, new=list(next_new))


def compute_next_step(initial_old, current_old, current_new, target_new,
                      config) -> Optional[UpdateStep]:
    """planner.go:324-356 — ordering: complete -> correctAbnormal ->
    newAtTarget -> tryScaleUp -> tryProportionalDrain -> tryForceDrain."""
    if is_complete(current_old, current_new, target_new):
        return None
    total_steps = compute_total_steps(initial_old, target_new, config)
    if total_steps == 0:
        return None
    step = correct_abnormal_state(current_old, current_new, initial_old)
    if step is not None:
        return step
    if is_new_at_target(current_new, target_new):
        return UpdateStep(past=[0] * len(initial_old), new=list(current_new))
    next_new = compute_next_new_replicas(target_new, current_new, total_steps)
    min_old = compute_min_old(initial_old, current_new, target_new, config)
    step = try_scale_up(current_old, current_new, next_new, target_new, config)
    if step is not None:
        return step
    step = try_proportional_drain(initial_old, current_old, current_new,
                                  target_new, min_old, total_steps, config)
    if step is not None:
        return step
    step = try_force_drain(current_old, next_new, initial_old, target_new,
                           config)
    if step is not None:
        return step
    return None


def compute_all_steps(initial_old, target, config) -> list[UpdateStep]:
    """planner.go:359-389 — full-rollout simulator (used by tests)."""
    n = len(initial_old)
    current_old = list(initial_old)
    current_new = [0] * n
    # every step strictly drains old or grows new in at least one role, so
    # the worst case is bounded by the total unit changes across roles
    # (a per-role max bound undercounts interleaved up/down sequences,
    # e.g. [12,12]->[1,12] takes 35 steps)
    max_steps = sum(initial_old) + sum(target) + 10
    steps = [UpdateStep(past=list(initial_old), new=[0] * n)]
    for _ in range(max_steps):
        nxt = compute_next_step(initial_old, current_old, current_new, target,
                                config)
        if nxt is None:
            break
        steps.append(nxt)
        current_old = list(nxt.past)
        current_new = list(nxt.new)
    return steps
