"""Shared machinery for importance-regularised lifelong methods (EWC / MAS).

Importance accumulation replays remembered task loaders and accumulates
per-parameter grad statistics (K9 in SURVEY.md §2.9); on GPU the accumulation
runs through the fused multi-tensor importance kernel (flreid_amd.ops).
"""

from __future__ import annotations

from typing import Any, Dict, Optional, Union

import torch
from torch import nn
from torch.utils.data import DataLoader

from flreid_amd import ops
from flreid_amd.modules.model import ModelModule
from flreid_amd.runtime.precision import autocast


class ImportanceModel(ModelModule):
    """Net + per-parameter importance (Fisher) + parameter anchors.

    mode='sq'  -> EWC  (F += g² · |batch| / #batches; skips the current task —
                  ref:methods/ewc.py:56-78)
    mode='abs' -> MAS  (F += |g| · |batch| / #batches; includes every
                  remembered task — ref:methods/mas.py:55-76)
    """

    mode = "sq"
    skip_current = True

    def __init__(self, net: Union[nn.Sequential, nn.Module],
                 operator=None, lambda_penalty: float = 100.0, **kwargs):
        super().__init__(net)
        self.operator = operator
        self.lambda_penalty = lambda_penalty
        self.args = kwargs
        self.params = {n: p for n, p in self.net.named_parameters() if p.requires_grad}
        self.params_old: Dict[str, torch.Tensor] = {}
        self.precision_matrices: Dict[str, torch.Tensor] = {}
        self.recall_dataloaders: Dict[str, DataLoader] = {}
        self.calculate()

    def calculate(self) -> Dict[str, torch.Tensor]:
        self.precision_matrices = self._calculate_importance()
        for n, p in self.params.items():
            self.params_old[n] = p.clone().detach()
        return self.precision_matrices

    def _recall_loaders(self):
        loaders = list(self.recall_dataloaders.values())
        if self.skip_current:
            if len(loaders) <= 1:
                return []
            return loaders[:-1]
        return loaders

    def _calculate_importance(self) -> Dict[str, torch.Tensor]:
        precision = {n: torch.zeros_like(p) for n, p in self.params.items()}
        loaders = self._recall_loaders()
        if not loaders:
            return precision

        device = next(self.net.parameters()).device
        n_batches = sum(len(loader) for loader in loaders)
        for loader in loaders:
            for data, person_id, _classes_id in loader:
                self.net.zero_grad()
                data, target = data.to(device), person_id.to(device)
                with autocast(device):
                    loss = self.operator._invoke_train(self, data, target)["loss"]
                loss.backward()
                scale = len(data) / n_batches
                grads = {n: p.grad.detach()
                         for n, p in self.params.items() if p.grad is not None}
                ops.importance_update(precision, grads, mode=self.mode,
                                      scale=scale)
        self.net.zero_grad(set_to_none=True)
        return precision

    def penalty(self) -> torch.Tensor:
        return self.lambda_penalty * ops.quadratic_penalty(
            self.params, self.params_old, self.precision_matrices)

    def remember_task(self, task_name: str, dataloader: DataLoader) -> None:
        self.recall_dataloaders[task_name] = dataloader
        self.calculate()

    def forward(self, data: torch.Tensor) -> Any:
        return self.net(data)

    # keep auxiliary tensors on the model's device (ref:methods/ewc.py:94-116)
    def _move_aux(self, fn) -> None:
        self.precision_matrices = {n: fn(p) for n, p in self.precision_matrices.items()}
        self.params_old = {n: fn(p) for n, p in self.params_old.items()}

    def to(self, *args, **kwargs):
        out = super().to(*args, **kwargs)
        self._move_aux(lambda p: p.to(*args, **kwargs))
        self.params = {n: p for n, p in self.net.named_parameters() if p.requires_grad}
        return out

    def cpu(self):
        return self.to("cpu")

    def cuda(self, device=None):
        return self.to(device if device is not None else "cuda")

    def model_state(self) -> Dict:
        return {
            "net_params": {n: p.clone().detach() for n, p in self.net.state_dict().items()},
            "params_old": {n: p.clone().detach() for n, p in self.params_old.items()},
            "precision_matrices": {n: p.clone().detach()
                                   for n, p in self.precision_matrices.items()},
        }

    def update_model(self, params_state: Dict) -> None:
        """NOTE: unlike the reference (ref:methods/ewc.py:146-152, which copies
        params_old/precision onto themselves — loading a ckpt never restored
        Fisher), this restores the stored anchors/importance properly."""
        if "net_params" in params_state:
            net_dict = self.net.state_dict()
            for n, p in params_state["net_params"].items():
                key = n[len("net."):] if n.startswith("net.") else n
                if key in net_dict:
                    net_dict[key] = p.clone().detach()
            self.net.load_state_dict(net_dict)
        self.params = {n: p for n, p in self.net.named_parameters() if p.requires_grad}
        if "params_old" in params_state:
            self.params_old = {n: p.clone().detach()
                               for n, p in params_state["params_old"].items()}
        if "precision_matrices" in params_state:
            self.precision_matrices = {n: p.clone().detach()
                                       for n, p in params_state["precision_matrices"].items()}
