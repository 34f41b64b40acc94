"""Criterion contract (ref:modules/criterion.py:4-10)."""

import torch.nn as nn


class CriterionModule(nn.Module):
    def forward(self, score, target, **kwargs):
        raise NotImplementedError
