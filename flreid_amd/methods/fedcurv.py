"""FedCurv — FedAvg + cross-client curvature (ref:methods/fedcurv.py).

Each client uploads its trainable params AND its Fisher matrix; the server
re-broadcasts every client's (params, Fisher) so the local penalty becomes
  λ·( Σ F_own·(p−p_old)² + Σ_other F_o·(p−p_o)² )
(ref:methods/fedcurv.py:79-86,621-645).  Aggregation of the params is plain
FedAvg weighted averaging (ref:methods/fedcurv.py:592-606).

MI355X comm note: the full-mesh (params, Fisher) broadcast is C3 in
SURVEY.md §2.9 — it rides the round driver's flat tensor-codec all-gather
(parallel/codec.py), 2 tensor-sets per client per round.
"""

from __future__ import annotations

from typing import Any, Dict, List, Tuple

import torch

from flreid_amd import ops
from flreid_amd.methods._importance import ImportanceModel
from flreid_amd.methods.common import BaseReIDOperator
from flreid_amd.methods.ewc import _LocalLifelongClient
from flreid_amd.modules.server import ServerModule


class Model(ImportanceModel):
    mode = "sq"
    skip_current = False   # every remembered task incl. current (ref :62-66)

    def __init__(self, net, operator=None, lambda_penalty: float = 100.0, **kwargs):
        self.other_precision_matrices: List[Tuple[Dict, Dict]] = []
        super().__init__(net, operator, lambda_penalty, **kwargs)

    def penalty(self) -> torch.Tensor:
        own = ops.quadratic_penalty(self.params, self.params_old,
                                    self.precision_matrices)
        for importance, params in self.other_precision_matrices:
            own = own + ops.quadratic_penalty(self.params, params, importance)
        return self.lambda_penalty * own

    def _move_aux(self, fn) -> None:
        super()._move_aux(fn)
        self.other_precision_matrices = [
            ({n: fn(p) for n, p in imp.items()},
             {n: fn(p) for n, p in par.items()})
            for imp, par in self.other_precision_matrices
        ]

    def model_state(self) -> Dict:
        state = super().model_state()
        state["other_precision_matrices"] = [
            ({n: p.clone().detach() for n, p in imp.items()},
             {n: p.clone().detach() for n, p in par.items()})
            for imp, par in self.other_precision_matrices
        ]
        return state

    def update_model(self, params_state: Dict) -> None:
        super().update_model(params_state)
        if "other_precision_matrices" in params_state:
            self.other_precision_matrices = [
                ({n: p.clone().detach() for n, p in imp.items()},
                 {n: p.clone().detach() for n, p in par.items()})
                for imp, par in params_state["other_precision_matrices"]
            ]


class Operator(BaseReIDOperator):
    def penalty(self, model) -> torch.Tensor:
        return model.penalty()


class Client(_LocalLifelongClient):
    default_ckpt_name = "fedcurv_model"
    remember_with = "train"

    def get_incremental_state(self, **kwargs) -> Dict:
        return {
            "train_cnt": self.train_cnt,
            "incremental_model_params": {
                n: p.clone().detach()
                for n, p in self.model.net.named_parameters() if p.requires_grad},
            "incremental_precision_matrices": {
                n: p.clone().detach()
                for n, p in self.model.precision_matrices.items()},
        }

    def get_integrated_state(self, **kwargs) -> Dict:
        return {
            "train_cnt": self.train_cnt,
            "integrated_model_params": {
                n: p.clone().detach()
                for n, p in self.model.net.state_dict().items()},
            "integrated_precision_matrices": {
                n: p.clone().detach()
                for n, p in self.model.precision_matrices.items()},
        }

    def _apply_dispatch(self, net_params: Dict, others: List[Tuple[Dict, Dict]]):
        self.train_cnt = self.test_cnt = 0
        self.load_model(self.model_ckpt_name)
        self.update_model({"net_params": net_params,
                           "other_precision_matrices": others})
        self.save_model(self.model_ckpt_name)

    def update_by_incremental_state(self, state: Dict, **kwargs) -> Any:
        others = list(zip(state["other_clients_precision_matrices"],
                          state["other_clients_incremental_params"]))
        self._apply_dispatch(state["incremental_model_params"], others)

    def update_by_integrated_state(self, state: Dict, **kwargs) -> Any:
        others = list(zip(state["other_clients_precision_matrices"],
                          state["other_clients_integrated_params"]))
        self._apply_dispatch(state["integrated_model_params"], others)


class Server(ServerModule):
    def update_model(self, params_state: Dict) -> None:
        self.model.update_model(params_state)

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
        self.update_model({"net_params": merged})

    def set_client_incremental_state(self, client_name: str, client_state: Dict) -> None:
        if client_name not in self.clients:
            self.logger.warn(f"unregistered client {client_name} upload ignored")
            return
        self.clients[client_name] = client_state

    set_client_integrated_state = set_client_incremental_state

    def _others(self, key_params: str) -> Tuple[List[Dict], List[Dict]]:
        params, fishers = [], []
        for _c, state in self.clients.items():
            if state is None:
                continue
            params.append({n: p.clone().detach()
                           for n, p in state[key_params].items()})
            fkey = ("incremental_precision_matrices"
                    if "incremental_precision_matrices" in state
                    else "integrated_precision_matrices")
            fishers.append({n: p.clone().detach()
                            for n, p in state[fkey].items()})
        return params, fishers

    def get_dispatch_incremental_state(self, client_name: str) -> Dict:
        params, fishers = self._others("incremental_model_params")
        return {
            "incremental_model_params": {
                n: p.clone().detach()
                for n, p in self.model.net.named_parameters() if p.requires_grad},
            "other_clients_incremental_params": params,
            "other_clients_precision_matrices": fishers,
        }

    def get_dispatch_integrated_state(self, client_name: str) -> Dict:
        return {
            "integrated_model_params": {
                n: p.clone().detach()
                for n, p in self.model.net.state_dict().items()},
            "other_clients_integrated_params": [],
            "other_clients_precision_matrices": [],
        }
