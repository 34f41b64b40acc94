"""MAS — Memory-Aware Synapses, local lifelong method (ref:methods/mas.py).

Importance = E[|g|] over ALL remembered tasks including the current one
(ref:methods/mas.py:61-74); the remembered loader is the VALIDATION loader
(ref:methods/mas.py:416)."""

from __future__ import annotations

from typing import Dict

import torch

from flreid_amd.methods._importance import ImportanceModel
from flreid_amd.methods.common import BaseReIDOperator
from flreid_amd.methods.ewc import _LocalLifelongClient
from flreid_amd.modules.server import ServerModule


class Model(ImportanceModel):
    mode = "abs"
    skip_current = False


class Operator(BaseReIDOperator):
    def penalty(self, model) -> torch.Tensor:
        return model.penalty()


class Client(_LocalLifelongClient):
    default_ckpt_name = "mas_model"
    remember_with = "val"


class Server(ServerModule):
    def get_dispatch_integrated_state(self, client_name: str) -> Dict:
        return {"model_params": {
            n: p.clone().detach() for n, p in self.model.state_dict().items()
        }}
