"""Temperature-scaled KL distillation (ref:criterions/kd_loss.py:10-27)."""

from __future__ import annotations

from flreid_amd import ops
from flreid_amd.modules.criterion import CriterionModule


class DistillKL(CriterionModule):
    def __init__(self, temperature: float = 1.0, **kwargs):
        super().__init__()
        for n, p in kwargs.items():
            setattr(self, n, p)
        self.temperature = temperature

    def forward(self, score, target, **kwargs):
        """`score` = student logits, `target` = teacher logits."""
        return ops.kd_loss(score, target, self.temperature)
