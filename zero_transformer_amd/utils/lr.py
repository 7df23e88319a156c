"""Warmup-cosine LR schedule, matching optax.warmup_cosine_decay_schedule
as used by the reference (main_zero.py:207-213: init 0, peak, warmup_steps,
decay_steps=143000 total, end_value).

optax semantics: linear 0 -> peak over [0, warmup); cosine from peak to
end_value over [warmup, decay_steps); constant end_value after.
"""

from __future__ import annotations

import math


def warmup_cosine(
    peak_lr: float,
    warmup_steps: int,
    decay_steps: int,
    end_lr: float = 0.0,
    init_lr: float = 0.0,
):
    """Returns step -> lr (step is 1-indexed as counted by the optimizer)."""

    def schedule(step: int) -> float:
        s = max(step - 1, 0)
        if s < warmup_steps:
            if warmup_steps == 0:
                return peak_lr
            return init_lr + (peak_lr - init_lr) * s / warmup_steps
        n = max(decay_steps - warmup_steps, 1)
        frac = min((s - warmup_steps) / n, 1.0)
        cos = 0.5 * (1.0 + math.cos(math.pi * frac))
        alpha = end_lr / peak_lr if peak_lr else 0.0
        return peak_lr * ((1 - alpha) * cos + alpha)

    return schedule
