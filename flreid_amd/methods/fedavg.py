"""FedAvg (ref:methods/fedavg.py).

Client uploads its trainable params + train sample count; the server computes
the data-count-weighted average Σ p_c·k_c/Σk over ALL registered clients'
latest uploads (stale uploads included — ref:methods/fedavg.py:386-397) and
dispatches the averaged trainable set (incremental) or the full state dict
(integrated, first contact).

MI355X execution: the aggregation is C1 in SURVEY.md §2.9 — each rank
pre-scales its local clients' uploads and the runtime's sync layer reduces
over RCCL; `Server.calculate` below then sees the replicated state and is a
pure local computation identical on every rank.
"""

from __future__ import annotations

from typing import Any, Dict

import torch

from flreid_amd.methods.common import (
    BaseReIDClient,
    BaseReIDOperator,
    weighted_average_states,
)
from flreid_amd.modules.server import ServerModule


class Operator(BaseReIDOperator):
    pass


class Client(BaseReIDClient):
    default_ckpt_name = "fedavg_model"

    def get_incremental_state(self, **kwargs) -> Dict:
        increment_params = {
            n: p.clone().detach()
            for n, p in self.model.named_parameters() if p.requires_grad
        }
        return {"train_cnt": self.train_cnt,
                "incremental_model_params": increment_params}

    def get_integrated_state(self, **kwargs) -> Dict:
        integrated_params = {n: p.clone().detach()
                             for n, p in self.model.state_dict().items()}
        return {"train_cnt": self.train_cnt,
                "integrated_model_params": integrated_params}

    def update_by_incremental_state(self, state: Dict, **kwargs) -> Any:
        self.train_cnt = self.test_cnt = 0
        self.load_model(self.model_ckpt_name)
        self.update_model(state["incremental_model_params"])
        self.save_model(self.model_ckpt_name)

    def update_by_integrated_state(self, state: Dict, **kwargs) -> Any:
        self.train_cnt = self.test_cnt = 0
        self.load_model(self.model_ckpt_name)
        self.update_model(state["integrated_model_params"])
        self.save_model(self.model_ckpt_name)


class Server(ServerModule):
    #: key holding the uploaded trainable-parameter dict
    _params_key = "incremental_model_params"

    def collective_aggregate(self, ctx, local_uploads) -> bool:
        """C1 fast path: each rank pre-scales its OWN
        clients' cached uploads by k_c and the weighted sum rides ONE
        bucketed RCCL all-reduce — no state replication.  Stale-upload
        semantics hold because each rank keeps its own clients' latest
        uploads in `self.clients`.  Returns True when it handled the round
        (the driver then skips the gather + local calculate)."""
        for cname, state in local_uploads.items():
            if cname in self.clients:
                self.clients[cname] = state
        states = {c: s for c, s in self.clients.items() if s}
        local_k = float(sum(s["train_cnt"] for s in states.values()))
        total_k = ctx.all_reduce_scalar(local_k)
        if total_k == 0:
            return True
        local_sum = {}
        for _c, s in states.items():
            w = s["train_cnt"] / total_k
            for n, p in s[self._params_key].items():
                contrib = p.detach().to(torch.float32) * w
                local_sum[n] = local_sum.get(n, 0) + contrib
        if not local_sum:
            # this rank owns no uploads yet: contribute zeros of the right
            # schema (the model's trainable set)
            local_sum = {n: torch.zeros_like(p, dtype=torch.float32)
                         for n, p in self.model.named_parameters()
                         if p.requires_grad}
        merged = ctx.weighted_allreduce(local_sum, 1.0)
        self._apply_merged(merged)
        return True

    def _apply_merged(self, merged) -> None:
        self.update_model(merged)

    def calculate(self) -> Any:
        states = {c: s for c, s in self.clients.items() if s}
        if not states:
            return
        counts = {c: s["train_cnt"] for c, s in states.items()}
        merged = weighted_average_states(states, "incremental_model_params", counts)
        self.update_model(merged)

    def set_client_incremental_state(self, client_name: str, client_state: Dict) -> None:
        if client_name not in self.clients:
            self.logger.warn(f"unregistered client {client_name} upload ignored")
            return
        self.clients[client_name] = client_state

    def set_client_integrated_state(self, client_name: str, client_state: Dict) -> None:
        if client_name not in self.clients:
            self.logger.warn(f"unregistered client {client_name} upload ignored")
            return
        self.clients[client_name] = client_state

    def get_dispatch_incremental_state(self, client_name: str) -> Dict:
        return {"incremental_model_params": {
            n: p.clone().detach()
            for n, p in self.model.named_parameters() if p.requires_grad
        }}

    def get_dispatch_integrated_state(self, client_name: str) -> Dict:
        return {"integrated_model_params": {
            n: p.clone().detach() for n, p in self.model.state_dict().items()
        }}
