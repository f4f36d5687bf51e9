"""LR schedules.

Parity with /root/reference/mlx_lm_utils.py:5-56 (linear_schedule,
cosine_decay, join_schedules) plus the trainer's cosine_with_warmup
construction (/root/reference/core/training.py:764-820). A schedule is a
callable step -> lr.
"""
from __future__ import annotations

import math
from typing import Callable, List

Schedule = Callable[[int], float]


def linear_schedule(init: float, end: float, steps: int) -> Schedule:
    def sched(step: int) -> float:
        if steps <= 0:
            return end
        t = min(max(step, 0), steps) / steps
        return init + (end - init) * t

    return sched


def cosine_decay(init: float, decay_steps: int, end: float = 0.0) -> Schedule:
    def sched(step: int) -> float:
        t = min(max(step, 0), decay_steps) / max(decay_steps, 1)
        return end + (init - end) * 0.5 * (1 + math.cos(math.pi * t))

    return sched


def join_schedules(schedules: List[Schedule], boundaries: List[int]) -> Schedule:
    def sched(step: int) -> float:
        offset = 0
        for i, b in enumerate(boundaries):
            if step < b:
                return schedules[i](step - offset)
            offset = b
        return schedules[-1](step - offset)

    return sched


def build_schedule(
    schedule_cfg: dict, learning_rate: float, total_steps: int
) -> Schedule:
    """Build from the YAML ``training.scheduler`` block."""
    stype = (schedule_cfg or {}).get("type", "cosine")
    min_lr_ratio = float((schedule_cfg or {}).get("min_lr_ratio", 0.0))
    end_lr = learning_rate * min_lr_ratio
    warmup_steps = int((schedule_cfg or {}).get("warmup_steps", 0))
    if stype == "linear":
        return linear_schedule(learning_rate, end_lr, total_steps)
    if stype == "cosine":
        return cosine_decay(learning_rate, total_steps, end_lr)
    if stype == "cosine_with_warmup":
        warmup_steps = warmup_steps or max(total_steps // 100, 1)
        warm = linear_schedule(learning_rate * 1e-2, learning_rate, warmup_steps)
        cos = cosine_decay(learning_rate, max(total_steps - warmup_steps, 1), end_lr)
        return join_schedules([warm, cos], [warmup_steps])
    if stype == "constant":
        return lambda step: learning_rate
    raise ValueError(f"Unknown scheduler type: {stype}")
