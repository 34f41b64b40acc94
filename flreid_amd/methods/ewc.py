"""EWC — Elastic Weight Consolidation, local lifelong method
(ref:methods/ewc.py).

Fisher ≈ E[g²] replayed over *previous* tasks' train loaders after each task;
train loss adds λ·ΣF·(p−p_old)².  No upload: the server only dispatches the
full model state on first contact (ref:methods/ewc.py:496-502)."""

from __future__ import annotations

from typing import Any, Dict

import torch

from flreid_amd.methods._importance import ImportanceModel
from flreid_amd.methods.common import BaseReIDClient, BaseReIDOperator
from flreid_amd.modules.server import ServerModule


class Model(ImportanceModel):
    mode = "sq"
    skip_current = True   # ref:methods/ewc.py:63-64


class Operator(BaseReIDOperator):
    def penalty(self, model) -> torch.Tensor:
        return model.penalty()

    def invoke_train(self, model, dataloader, **kwargs) -> Any:
        # the reference reports the raw loss (penalty excluded) in metrics
        # (ref:methods/ewc.py:173-176); BaseReIDOperator reports loss+penalty,
        # which is what early-stop should see — keep base behaviour.
        return super().invoke_train(model, dataloader, **kwargs)


class _LocalLifelongClient(BaseReIDClient):
    """Shared by EWC/MAS: model-state ckpts + remember_task after training."""

    remember_with = "train"   # 'train' (EWC) or 'val' (MAS)

    def __init__(self, client_name, model, operator, ckpt_root,
                 model_ckpt_name=None, **kwargs):
        super().__init__(client_name, model, operator, ckpt_root,
                         model_ckpt_name, **kwargs)
        self.model.operator = operator

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

    def update_by_incremental_state(self, state: Dict, **kwargs) -> Any:
        if state is None:
            return
        self.load_model(self.model_ckpt_name)
        self.update_model({"net_params": state["model_params"]})
        self.save_model(self.model_ckpt_name)

    update_by_integrated_state = update_by_incremental_state

    def train(self, epochs, task_name, tr_loader, val_loader,
              early_stop_threshold: int = 3, device: str = "cpu", **kwargs) -> Any:
        self._remember_loaders = (tr_loader, val_loader)
        self._last_task_name = task_name
        return super().train(epochs, task_name, tr_loader, val_loader,
                             early_stop_threshold, device, **kwargs)

    def after_task_train(self, output, tr_loader, device) -> None:
        # consolidate importance on the remembered loader set while the model
        # is still on-device (ref:methods/ewc.py:89 / ref:methods/mas.py:416)
        loader = (self._remember_loaders[0] if self.remember_with == "train"
                  else self._remember_loaders[1])
        self.model.remember_task(self._last_task_name, loader)


class Client(_LocalLifelongClient):
    default_ckpt_name = "ewc_model"
    remember_with = "train"


class Server(ServerModule):
    def get_dispatch_integrated_state(self, client_name: str) -> Dict:
        return {"model_params": {
            n: p.clone().detach() for n, p in self.model.state_dict().items()
        }}
