"""Model contract (ref:modules/model.py:6-32)."""

from __future__ import annotations

from typing import Any, Dict

import torch.nn as nn


class ModelModule(nn.Module):
    """nn.Module wrapper holding `self.net`; methods may subclass to add
    auxiliary state (Fisher matrices, exemplars, adaptive layers ...)."""

    def __init__(self, net: nn.Module):
        super().__init__()
        self.net = net

    def forward(self, *args, **kwargs):
        return self.net(*args, **kwargs)

    @property
    def device(self):
        return next(self.parameters()).device

    def model_state(self, *args, **kwargs) -> Dict:
        raise NotImplementedError

    def update_model(self, *args, **kwargs) -> Any:
        raise NotImplementedError
