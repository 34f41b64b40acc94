"""FedProx — FedAvg + proximal term (ref:methods/fedprox.py).

Adds λ_l2·Σ(p − p_global)² against the last dispatched global parameters,
snapshotted on every server update (ref:methods/fedprox.py:42-57,344-366).
Server aggregation is identical to FedAvg (ref:methods/fedprox.py:488-501)."""

from __future__ import annotations

from typing import Any, Dict, Union

import torch
from torch import nn

from flreid_amd import ops
from flreid_amd.methods.common import BaseReIDClient, BaseReIDOperator
from flreid_amd.methods.fedavg import Server as FedAvgServer
from flreid_amd.modules.model import ModelModule


class Model(ModelModule):
    def __init__(self, net: Union[nn.Sequential, nn.Module],
                 lambda_l2: float = 1e-2, **kwargs):
        super().__init__(net)
        self.lambda_l2 = lambda_l2
        self.args = kwargs
        self.params = {n: p for n, p in self.net.named_parameters() if p.requires_grad}
        self.params_old: Dict[str, torch.Tensor] = {}

    def remember_params(self) -> None:
        self.params_old = {n: p.clone().detach()
                           for n, p in self.net.named_parameters() if p.requires_grad}

    def penalty(self) -> torch.Tensor:
        if not self.params_old:
            dev = next(self.net.parameters()).device
            return torch.zeros((), device=dev)
        return self.lambda_l2 * ops.quadratic_penalty(self.params, self.params_old)

    def to(self, *args, **kwargs):
        out = super().to(*args, **kwargs)
        self.params_old = {n: p.to(*args, **kwargs) for n, p in self.params_old.items()}
        self.params = {n: p for n, p in self.net.named_parameters() if p.requires_grad}
        return out

    def model_state(self) -> Dict:
        return {
            "net_params": {n: p.clone().detach() for n, p in self.net.state_dict().items()},
            "params_old": {n: p.clone().detach() for n, p in self.params_old.items()},
        }

    def update_model(self, params_state: Dict) -> None:
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


class Operator(BaseReIDOperator):
    def penalty(self, model) -> torch.Tensor:
        return model.penalty()


class Client(BaseReIDClient):
    default_ckpt_name = "fedprox_model"

    def update_model(self, params_state: Dict) -> None:
        self.model.update_model(params_state)

    def load_model(self, model_name: str) -> None:
        model_dict = self.model.model_state()
        model_dict = self.load_state(model_name, model_dict)
        self.model.update_model(model_dict)

    def save_model(self, model_name: str) -> None:
        if self._ckpt_disabled():
            return
        self.save_state(model_name, self.model.model_state(), True)

    def get_incremental_state(self, **kwargs) -> Dict:
        increment_params = {
            n: p.clone().detach()
            for n, p in self.model.net.named_parameters() if p.requires_grad
        }
        return {"train_cnt": self.train_cnt,
                "incremental_model_params": increment_params}

    def get_integrated_state(self, **kwargs) -> Dict:
        return {"train_cnt": self.train_cnt,
                "integrated_model_params": {
                    n: p.clone().detach()
                    for n, p in self.model.net.state_dict().items()}}

    def update_by_incremental_state(self, state: Dict, **kwargs) -> Any:
        self.train_cnt = self.test_cnt = 0
        self.load_model(self.model_ckpt_name)
        self.update_model({"net_params": state["incremental_model_params"]})
        self.model.remember_params()
        self.save_model(self.model_ckpt_name)

    def update_by_integrated_state(self, state: Dict, **kwargs) -> Any:
        self.train_cnt = self.test_cnt = 0
        self.load_model(self.model_ckpt_name)
        self.update_model({"net_params": state["integrated_model_params"]})
        self.model.remember_params()
        self.save_model(self.model_ckpt_name)


class Server(FedAvgServer):
    """Identical weighted averaging (ref:methods/fedprox.py:488-501), but
    dispatch/collect uses the net's parameter names (no 'net.' prefix)."""

    def _apply_merged(self, merged) -> None:
        self.model.update_model({"net_params": merged})

    def calculate(self) -> Any:
        states = {c: s for c, s in self.clients.items() if s}
        if not states:
            return
        total = sum(s["train_cnt"] for s in states.values())
        if total == 0:
            return
        merged: Dict[str, torch.Tensor] = {}
        for _c, s in states.items():
            k = s["train_cnt"]
            for n, p in s["incremental_model_params"].items():
                merged[n] = merged.get(n, 0) + p.detach().to(torch.float32) * (k / total)
        self.model.update_model({"net_params": merged})

    def get_dispatch_incremental_state(self, client_name: str) -> Dict:
        return {"incremental_model_params": {
            n: p.clone().detach()
            for n, p in self.model.net.named_parameters() if p.requires_grad
        }}

    def get_dispatch_integrated_state(self, client_name: str) -> Dict:
        return {"integrated_model_params": {
            n: p.clone().detach() for n, p in self.model.net.state_dict().items()
        }}
