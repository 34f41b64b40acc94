"""Client contract (ref:modules/client.py:12-129).

A Client is one simulated edge node.  In the MI355X framework a client lives
inside exactly one rank (one process per GPU); the reference's thread-pool
device juggling (ref:experiment.py:58-99) does not exist here.

Checkpoint I/O keeps the reference layout byte-for-byte:
    {ckpt_root}/{client_name}/{state_name}.ckpt          (ref:modules/client.py:28,39)
with `load_state` mapping to CPU (ref:modules/client.py:43) and
`save_state(cover=False)` raising on collision (ref:modules/client.py:59-60).
"""

from __future__ import annotations

import os
from typing import Any, Dict, List, Union

import torch
from torch.utils.data import DataLoader

from flreid_amd.modules.model import ModelModule
from flreid_amd.modules.operator import OperatorModule
from flreid_amd.tools.logger import Logger


class ClientModule:
    def __init__(self, client_name: str, model: ModelModule,
                 operator: OperatorModule, ckpt_root: str,
                 model_ckpt_name: str = None, **kwargs):
        self.client_name = client_name
        self.model = model
        self.operator = operator
        # arbitrary yaml kwargs become attributes (ref:modules/client.py:25-26)
        for n, p in kwargs.items():
            setattr(self, n, p)

        self.ckpt_path = os.path.join(ckpt_root, self.client_name)
        self.model_ckpt_name = model_ckpt_name
        self.logger = Logger(client_name)
        self.operator.logger = self.logger

    # ------------------------------------------------------------------ ckpt
    # FLREID_DISABLE_CKPT=1 turns the per-round ckpt audit trail off for
    # benchmarking (the reference's disk round trip per dispatch/upload is an
    # auditing feature, not training semantics — ref:experiment.py:199-202)
    @staticmethod
    def _ckpt_disabled() -> bool:
        return os.environ.get("FLREID_DISABLE_CKPT", "0") == "1"

    def load_state(self, state_name: str, default_value: Any = None) -> Any:
        if self._ckpt_disabled():
            if default_value is not None:
                return default_value
            raise ValueError("ckpt disabled and no default value")
        from flreid_amd.runtime.io import before_ckpt_read
        before_ckpt_read()
        state_path = os.path.join(self.ckpt_path, f"{state_name}.ckpt")
        os.makedirs(self.ckpt_path, exist_ok=True)
        if os.path.exists(state_path):
            return torch.load(state_path, map_location="cpu", weights_only=False)
        if default_value is not None:
            return default_value
        raise ValueError(f"State checkpoint does not exist in '{state_path}'.")

    def state_exists(self, state_name: str) -> bool:
        if self._ckpt_disabled():
            return False
        from flreid_amd.runtime.io import before_ckpt_read
        before_ckpt_read()
        return os.path.exists(os.path.join(self.ckpt_path, f"{state_name}.ckpt"))

    def save_state(self, state_name: str, state: Any, cover: bool = False) -> None:
        if state_name is None or self._ckpt_disabled():
            return
        state_path = os.path.join(self.ckpt_path, f"{state_name}.ckpt")
        os.makedirs(self.ckpt_path, exist_ok=True)
        if not cover and os.path.exists(state_path):
            raise ValueError(f"State checkpoint already exists in '{state_path}'.")
        from flreid_amd.runtime.io import save_ckpt
        save_ckpt(state_path, state)

    # ----------------------------------------------------------------- model
    def load_model(self, model_name: str) -> None:
        self.model.load_state_dict(
            self.load_state(model_name, default_value=self.model.state_dict())
        )

    def save_model(self, model_name: str) -> None:
        self.save_state(model_name, self.model.state_dict(), True)

    def update_model(self, params_state: Dict[str, torch.Tensor]) -> None:
        model_dict = self.model.state_dict()
        for n, p in params_state.items():
            model_dict[n] = p.clone().detach()
        self.model.load_state_dict(model_dict)

    # ------------------------------------------------- federation state hooks
    def get_incremental_state(self, **kwargs) -> Dict:
        return None

    def get_integrated_state(self, **kwargs) -> Dict:
        return None

    def update_by_incremental_state(self, state: Dict, **kwargs) -> Any:
        return None

    def update_by_integrated_state(self, state: Dict, **kwargs) -> Any:
        return None

    # ------------------------------------------------------------- execution
    def train(self, epochs: int, task_name: str,
              tr_loader: Union[List[DataLoader], DataLoader],
              val_loader: Union[List[DataLoader], DataLoader],
              device: str, **kwargs) -> Any:
        raise NotImplementedError

    def train_one_epoch(self, task_name: str, tr_loader, val_loader, **kwargs) -> Any:
        raise NotImplementedError

    def inference(self, task_name: str, query_loader, gallery_loader,
                  device: str, **kwargs) -> Any:
        raise NotImplementedError

    def validate(self, task_name: str, query_loader, gallery_loader,
                 device: str, **kwargs) -> Any:
        raise NotImplementedError
