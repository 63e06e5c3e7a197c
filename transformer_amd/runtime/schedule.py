"""Noam LR schedule (C11, reference train.py:21-34).

lr(step) = d_model^-0.5 * min(step^-0.5, step * warmup^-1.5); default warmup
60000 — the reference's default, not the paper's 4000,
flag-overridable.
"""

from __future__ import annotations


class NoamSchedule:
    def __init__(self, d_model: int, warmup_steps: int = 60000):
        self.d_model = float(d_model)
        self.warmup_steps = int(warmup_steps)

    def __call__(self, step: int) -> float:
        step = float(max(int(step), 1))
        return (self.d_model ** -0.5) * min(step ** -0.5,
                                            step * self.warmup_steps ** -1.5)
