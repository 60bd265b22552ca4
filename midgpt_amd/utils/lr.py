"""Warmup + cosine decay LR schedule.

Matches optax.warmup_cosine_decay_schedule(0, peak, warmup_steps,
decay_steps, end_value=min_lr) as used by the reference
(src/train.py:147-149): linear 0 -> peak over [0, warmup], cosine
peak -> min_lr over [warmup, decay_steps], constant min_lr after.
``step`` is 0-indexed (the optax count at update time).
"""
from __future__ import annotations

import math


def warmup_cosine_lr(step: int, *, peak_lr: float, warmup_steps: int,
                     decay_steps: int, min_lr: float) -> float:
    if warmup_steps > 0 and step < warmup_steps:
        return peak_lr * step / warmup_steps
    if step >= decay_steps:
        return min_lr
    frac = (step - warmup_steps) / max(1, decay_steps - warmup_steps)
    return min_lr + 0.5 * (peak_lr - min_lr) * (1 + math.cos(math.pi * frac))
