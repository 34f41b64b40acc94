"""FedSTIL-atten — FedSTIL with stacked global weights + learnable per-slot
attention (ref:methods/fedstil_atten.py).

Differences from FedSTIL (see the diff against ref:methods/fedstil.py):
  - θ = Σ_k atten_k·W_glob[...,k] + W_adapt with TRAINABLE atten
    (ref:methods/fedstil_atten.py:89-90); adaptive_weight persists across
    dispatches while atten re-inits to atten_default sized to the stack;
  - the server CONCATENATES client composed uploads along a new last dim
    instead of weighted-averaging (ref:methods/fedstil_atten.py:1100-1121);
  - dispatch is the server's current stacked global weights — no KL-token
    personalization (ref:methods/fedstil_atten.py:1146-1148).
"""

from __future__ import annotations

from typing import Any, Dict

import torch

from flreid_amd.methods import fedstil as _fedstil
from flreid_amd.models.adaptive import convert_to_stacked


class Model(_fedstil.Model):
    atten_trainable = True

    def _convert_net(self) -> None:
        convert_to_stacked(self.net, self.atten_default)

    def composed_upload(self) -> Dict[str, torch.Tensor]:
        """Composed θ with a trailing stack dim of 1
        (ref:methods/fedstil_atten.py:870-873)."""
        return {f"{n}.global_weight": l.composed_weight().detach().unsqueeze(-1)
                for n, l in self.adaptive_module_leaves()}


class Operator(_fedstil.Operator):
    pass


class Client(_fedstil.Client):
    default_ckpt_name = "fedstil_atten_model"


class Server(_fedstil.Server):
    def calculate(self) -> Any:
        """Stack client uploads along the last dim (client order = upload
        application order — deterministic across ranks); persist tokens."""
        states = {c: s for c, s in self.clients.items() if s}
        if not states:
            return
        stacked: Dict[str, torch.Tensor] = {}
        for _c, s in states.items():
            for n, p in s["incremental_sw"].items():
                p = p.detach()
                stacked[n] = p if n not in stacked else torch.cat(
                    [stacked[n], p], dim=-1)
        self.model.update_model({"global_weight": stacked})
        self.save_state(f"{self.server_name}_tokens", self.token_memory, True)

    def get_dispatch_incremental_state(self, client_name: str) -> Dict:
        return {"incremental_shared_params":
                self.model.model_state()["global_weight"]}
