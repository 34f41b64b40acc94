"""Baseline — pure local lifelong learning, no federation
(ref:methods/baseline.py).

No upload (get_incremental_state stays None, so the round driver skips the
server — ref:experiment.py:239-240); first contact receives the server's full
state dict once (ref:methods/baseline.py:341-345).  With `model_ckpt_name`
unset, checkpoints are kept per task ("multi model" mm config); with it set,
one shared checkpoint ("single model" sm config)
(ref:configs/basis_exp/experiment_sm.yaml, experiment_mm.yaml).
"""

from __future__ import annotations

from typing import Any, Dict

from flreid_amd.methods.common import BaseReIDClient, BaseReIDOperator
from flreid_amd.modules.server import ServerModule


class Operator(BaseReIDOperator):
    pass


class Client(BaseReIDClient):
    default_ckpt_name = None  # per-task ckpts unless yaml names one

    def __init__(self, client_name, model, operator, ckpt_root,
                 model_ckpt_name=None, **kwargs):
        super().__init__(client_name, model, operator, ckpt_root,
                         model_ckpt_name, **kwargs)
        self.current_task = None

    def _ckpt(self, task_name: str) -> str:
        return self.model_ckpt_name if self.model_ckpt_name else task_name

    def update_by_integrated_state(self, state: Dict, **kwargs) -> Any:
        self.train_cnt = self.test_cnt = 0
        self.update_model(state["integrated_model_params"])

    def train(self, epochs, task_name, tr_loader, val_loader,
              early_stop_threshold: int = 3, device: str = "cpu", **kwargs) -> Any:
        self.current_task = task_name
        ckpt = self._ckpt(task_name)
        saved_name = self.model_ckpt_name
        try:
            self.model_ckpt_name = ckpt
            return super().train(epochs, task_name, tr_loader, val_loader,
                                 early_stop_threshold, device, **kwargs)
        finally:
            self.model_ckpt_name = saved_name

    def validate(self, task_name, query_loader, gallery_loader,
                 device: str = "cpu", **kwargs) -> Any:
        saved_name = self.model_ckpt_name
        try:
            self.model_ckpt_name = self._ckpt(task_name)
            return super().validate(task_name, query_loader, gallery_loader,
                                    device, **kwargs)
        finally:
            self.model_ckpt_name = saved_name

    def inference(self, task_name, query_loader, gallery_loader,
                  device: str = "cpu", **kwargs) -> Any:
        saved_name = self.model_ckpt_name
        try:
            self.model_ckpt_name = self._ckpt(task_name)
            return super().inference(task_name, query_loader, gallery_loader,
                                     device, **kwargs)
        finally:
            self.model_ckpt_name = saved_name


class Server(ServerModule):
    def get_dispatch_integrated_state(self, client_name: str) -> Dict:
        return {"integrated_model_params": {
            n: p.clone().detach() for n, p in self.model.state_dict().items()
        }}
