"""Server contract (ref:modules/server.py:11-108).

The server is a *replicated* object in the MI355X framework: every rank holds
an identical copy, and the aggregation hooks (`calculate`,
`set_client_*_state`, `get_dispatch_*`) operate on state that the comm layer
(flreid_amd/parallel) keeps synchronised via RCCL collectives.  In
single-process mode it behaves exactly like the reference's in-process server.

Checkpoint layout: {ckpt_root}/{server_name}/{state_name}.ckpt
(ref:modules/server.py:25,36); in multi-rank runs only rank 0 writes.
"""

from __future__ import annotations

import os
from typing import Any, Dict

import torch

from flreid_amd.modules.model import ModelModule
from flreid_amd.modules.operator import OperatorModule
from flreid_amd.tools.logger import Logger


def _is_writer_rank() -> bool:
    return os.environ.get("RANK", "0") in ("0", "")


class ServerModule:
    def __init__(self, server_name: str, model: ModelModule,
                 operator: OperatorModule, ckpt_root: str, **kwargs):
        self.server_name = server_name
        self.model = model
        self.operator = operator
        for n, p in kwargs.items():
            setattr(self, n, p)
        self.ckpt_path = os.path.join(ckpt_root, self.server_name)
        self.clients: Dict[str, Any] = {}
        self.logger = Logger(server_name)
        self.operator.logger = self.logger

    # ------------------------------------------------------------------ ckpt
    def load_state(self, state_name: str, default_value: Any = None) -> Any:
        if os.environ.get("FLREID_DISABLE_CKPT", "0") == "1":
            if default_value is not None:
                return default_value
            raise ValueError("ckpt disabled and no default value")
        from flreid_amd.runtime.io import before_ckpt_read
        before_ckpt_read()
        state_path = os.path.join(self.ckpt_path, f"{state_name}.ckpt")
        os.makedirs(self.ckpt_path, exist_ok=True)
        if os.path.exists(state_path):
            return torch.load(state_path, weights_only=False)
        if default_value is not None:
            return default_value
        raise ValueError(f"State checkpoint does not exist in '{state_path}'.")

    def save_state(self, state_name: str, state: Any, cover: bool = False) -> None:
        if not _is_writer_rank() or os.environ.get("FLREID_DISABLE_CKPT", "0") == "1":
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

    # ------------------------------------------------------------- clients
    def register_client(self, client_name: str) -> bool:
        if client_name in self.clients:
            self.logger.warn(f"'{client_name}' already registered in server.")
            return False
        self.clients[client_name] = self.init_client_state()
        self.logger.info(f"'{client_name}' registered in server.")
        return True

    def unregister_client(self, client_name: str) -> bool:
        if client_name in self.clients:
            self.clients.pop(client_name)
            return True
        self.logger.warn(f"'{client_name}' is not registered in server.")
        return False

    # ------------------------------------------------- federation state hooks
    def calculate(self) -> Any:
        return None

    def init_client_state(self) -> Any:
        return None

    def set_client_incremental_state(self, client_name: str, client_state: Dict) -> None:
        return None

    def set_client_integrated_state(self, client_name: str, client_state: Dict) -> None:
        return None

    def get_dispatch_incremental_state(self, client_name: str) -> Dict:
        return None

    def get_dispatch_integrated_state(self, client_name: str) -> Dict:
        return None
