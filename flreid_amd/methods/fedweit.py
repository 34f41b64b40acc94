"""FedWeIT — Federated Weighted Inter-client Transfer (ref:methods/fedweit.py).

Decomposition θ = mask⊙sw + aw + Σ_k atten_k·aw_kb_k (models/decomposed.py);
loss adds sparsity λ1(|aw|+|mask|) and inter-task drift
λ2‖(sw−sw_old)·mask + (aw−aw_old)‖² against remembered per-task snapshots
(ref:methods/fedweit.py:592-625; NOTE the reference computed the drift term
against the model's own live tensors, i.e. always zero — this implements the
intended per-task anchors).

Exchange: client uploads {aw, composed gw, bn} + train_cnt
(ref:methods/fedweit.py:785-802); server weighted-averages composed weights
into sw and stacks kb_cnt sampled client aw's into the knowledge base along
a new last dim (ref:methods/fedweit.py:983-1007); dispatch = {sw, aw_kb}
(ref:methods/fedweit.py:1031-1045); client re-init on dispatch:
aw=(1−mask)·sw, atten=0 (ref:methods/fedweit.py:824-854).

MI355X comm: the per-layer aw_kb build is C2 in SURVEY.md §2.9 — client aw's
ride the flat tensor-codec all-gather; the kb stack itself is a local
device-side cat.
"""

from __future__ import annotations

import random
from typing import Any, Dict, List, Optional

import torch

from flreid_amd.methods.common import BaseReIDClient, BaseReIDOperator
from flreid_amd.models.decomposed import (
    convert_to_decomposed,
    decomposed_leaves,
    non_decomposed_leaves,
)
from flreid_amd.modules.model import ModelModule
from flreid_amd.modules.server import ServerModule


class Model(ModelModule):
    def __init__(self, net, lambda_l1: float = 1e-3, lambda_l2: float = 1e2,
                 lambda_mask: float = 0.0, kb_cnt: int = 5, **kwargs):
        super().__init__(net)
        self.lambda_l1 = lambda_l1
        self.lambda_l2 = lambda_l2
        self.lambda_mask = lambda_mask
        self.kb_cnt = kb_cnt
        self.args = kwargs
        # per-task anchors for the drift term: {task: {layer: (sw, aw)}}
        self.task_anchors: Dict[str, Dict[str, tuple]] = {}
        convert_to_decomposed(self.net, lambda_l1, lambda_mask, kb_cnt)

    def decomposed_module_leaves(self):
        return decomposed_leaves(self.net)

    def pre_trained_module_leaves(self):
        return non_decomposed_leaves(self.net)

    def remember_params(self, task_name: str) -> None:
        """Anchor (sw, aw) per layer for the finished task
        (ref:methods/fedweit.py:388-393 deep-copied whole nets; snapshots of
        the two tensors the drift term reads are equivalent and cheap)."""
        self.task_anchors[task_name] = {
            n: (m.sw.detach().clone(), m.aw.detach().clone())
            for n, m in self.decomposed_module_leaves()
        }

    def sparsity_loss(self) -> torch.Tensor:
        total = None
        for _n, m in self.decomposed_module_leaves():
            s = m.aw.abs().sum() + m.mask.abs().sum()
            total = s if total is None else total + s
        return total

    def drift_loss(self) -> Optional[torch.Tensor]:
        if not self.task_anchors:
            return None
        total = None
        for _task, anchors in self.task_anchors.items():
            for n, m in self.decomposed_module_leaves():
                sw_old, aw_old = anchors[n]
                bshape = m._mask_bcast_shape()
                d = ((m.sw - sw_old) * m.mask.view(bshape) + (m.aw - aw_old)) ** 2
                s = d.sum()
                total = s if total is None else total + s
        return total

    def forward(self, data: torch.Tensor) -> Any:
        return self.net(data)

    def model_state(self) -> Dict:
        layers = self.decomposed_module_leaves()
        return {
            "sw": {f"{n}.sw": m.sw.detach().clone() for n, m in layers},
            "aw": {f"{n}.aw": m.aw.detach().clone() for n, m in layers},
            "mask": {f"{n}.mask": m.mask.detach().clone() for n, m in layers},
            "bias": {f"{n}.bias": m.bias.detach().clone()
                     for n, m in layers if m.bias is not None},
            "atten": {f"{n}.atten": m.atten.detach().clone() for n, m in layers},
            "aw_kb": {f"{n}.aw_kb": m.aw_kb.detach().clone() for n, m in layers},
            "bn_params": {},
            "pre_trained_params": {
                f"{ln}.{pn}": p.detach().clone()
                for ln, layer in self.pre_trained_module_leaves()
                for pn, p in layer.state_dict().items()},
        }

    def update_model(self, params_state: Dict) -> None:
        live = dict(self.net.state_dict(keep_vars=True))
        with torch.no_grad():
            for section in ("sw", "aw", "mask", "bias", "atten", "aw_kb",
                            "bn_params", "pre_trained_params"):
                for n, p in params_state.get(section, {}).items():
                    dst = live.get(n)
                    if dst is None:
                        continue
                    if dst.shape == p.shape:
                        dst.copy_(p.detach())
                    else:  # aw_kb grows its last dim with the kb fill level
                        mod = self.net.get_submodule(n.rsplit(".", 1)[0])
                        getattr(mod, n.rsplit(".", 1)[1]).data = \
                            p.detach().clone().to(dst.device, dst.dtype)

    def composed_upload(self) -> Dict[str, torch.Tensor]:
        """{name.sw: mask⊙sw + aw + kb·atten} in EVAL composition (no
        pruning), ref:methods/fedweit.py:791-797."""
        out = {}
        for n, m in self.decomposed_module_leaves():
            kb = (m.atten * m.aw_kb).sum(dim=-1)
            out[f"{n}.sw"] = (m.mask.view(m._mask_bcast_shape()) * m.sw
                              + m.aw + kb).detach().clone()
        return out


class Operator(BaseReIDOperator):
    def penalty(self, model: Model) -> torch.Tensor:
        pen = model.sparsity_loss() * model.lambda_l1
        drift = model.drift_loss()
        if drift is not None:
            pen = pen + drift * model.lambda_l2
        return pen


class Client(BaseReIDClient):
    default_ckpt_name = "fedweit_model"

    def __init__(self, client_name, model, operator, ckpt_root,
                 model_ckpt_name=None, **kwargs):
        super().__init__(client_name, model, operator, ckpt_root,
                         model_ckpt_name, **kwargs)
        self.current_task: Optional[str] = None

    def update_model(self, params_state: Dict) -> None:
        self.model.update_model(params_state)

    def load_model(self, model_name: str) -> None:
        if self.state_exists(model_name):
            self.model.update_model(self.load_state(model_name, None))

    def save_model(self, model_name: str) -> None:
        if self._ckpt_disabled():
            return
        self.save_state(model_name, self.model.model_state(), True)

    def get_incremental_state(self, **kwargs) -> Dict:
        layers = self.model.decomposed_module_leaves()
        return {
            "train_cnt": self.train_cnt,
            "incremental_aw": {f"{n}.aw": m.aw.detach().clone() for n, m in layers},
            "incremental_gw": self.model.composed_upload(),
            "incremental_bn": {},
        }

    def get_integrated_state(self, **kwargs) -> Dict:
        state = self.model.model_state()
        layers = self.model.decomposed_module_leaves()
        return {
            "train_cnt": self.train_cnt,
            "integrated_aw": {f"{n}.aw": m.aw.detach().clone() for n, m in layers},
            "integrated_gw": self.model.composed_upload(),
            "integrated_bn": state["bn_params"],
            "pre_trained_params": state["pre_trained_params"],
        }

    def _apply_dispatch(self, params: Dict) -> None:
        if self.current_task:
            self.load_model(self.model_ckpt_name or self.current_task)
        self.update_model(params)
        # re-init: aw = (1-mask)·sw, atten = 0 (ref:methods/fedweit.py:824-838)
        with torch.no_grad():
            for _n, m in self.model.decomposed_module_leaves():
                m.aw.data = ((1.0 - m.mask.data.view(m._mask_bcast_shape()))
                             * m.sw.data).clone()
                m.atten.data.zero_()

    def update_by_incremental_state(self, state: Dict, **kwargs) -> Any:
        self._apply_dispatch({"sw": state["incremental_sw"],
                              "aw_kb": state["incremental_aw_kb"]})

    def update_by_integrated_state(self, state: Dict, **kwargs) -> Any:
        self._apply_dispatch({
            "sw": state["integrated_sw"],
            "aw_kb": state["integrated_aw_kb"],
            "bn_params": state["integrated_bn"],
            "pre_trained_params": state["pre_trained_params"],
        })

    def before_task_train(self, task_name: str, tr_loader) -> None:
        if self.current_task is not None and self.current_task != task_name:
            self.model.remember_params(task_name)
        self.current_task = task_name


class Server(ServerModule):
    def __init__(self, server_name, model: Model, operator, ckpt_root, **kwargs):
        super().__init__(server_name, model, operator, ckpt_root, **kwargs)
        self.client_aw: List[Dict] = []

    def update_model(self, params_state: Dict) -> None:
        self.model.update_model(params_state)

    def calculate(self) -> Any:
        """Weighted-average composed weights into sw; rebuild the knowledge
        base from kb_cnt sampled client aw's (ref:methods/fedweit.py:983-1015)."""
        states = {c: s for c, s in self.clients.items() if s}
        if not states:
            return
        total = sum(s["train_cnt"] for s in states.values())
        if total == 0:
            return
        merged_sw: Dict[str, torch.Tensor] = {}
        merged_bn: Dict[str, torch.Tensor] = {}
        for _c, s in states.items():
            k = s["train_cnt"]
            for n, p in s["incremental_gw"].items():
                merged_sw[n] = merged_sw.get(n, 0) + p.detach().to(torch.float32) * (k / total)
            for n, p in s.get("incremental_bn", {}).items():
                merged_bn[n] = merged_bn.get(n, 0) + p.detach().to(torch.float32) * (k / total)

        update = {"sw": merged_sw, "bn_params": merged_bn}

        self.client_aw = [s["incremental_aw"] for _c, s in states.items()]
        if len(self.client_aw) >= self.model.kb_cnt:
            # REPLICA-DETERMINISTIC sampling: every rank hosts a server
            # replica and each must build the SAME knowledge base, but the
            # ranks' global RNG streams differ (they trained different
            # clients) — a private generator seeded from the aggregation
            # counter keeps replicas bit-identical and placement-invariant
            self._kb_rounds = getattr(self, "_kb_rounds", 0) + 1
            rng = random.Random(0xFED0 + self._kb_rounds * 1009
                                + len(self.client_aw))
            sampled = rng.sample(self.client_aw, self.model.kb_cnt)
            kb = {}
            for name in sampled[0]:
                kb[name[:-len(".aw")] + ".aw_kb"] = torch.cat(
                    [aw[name].unsqueeze(-1) for aw in sampled], dim=-1)
            update["aw_kb"] = kb
        self.update_model(update)

    def set_client_incremental_state(self, client_name: str, client_state: Dict) -> None:
        if client_name not in self.clients:
            self.logger.warn(f"unregistered client {client_name} upload ignored")
            return
        self.clients[client_name] = client_state

    set_client_integrated_state = set_client_incremental_state

    def get_dispatch_incremental_state(self, client_name: str) -> Dict:
        state = self.model.model_state()
        return {"incremental_sw": state["sw"],
                "incremental_aw_kb": state["aw_kb"]}

    def get_dispatch_integrated_state(self, client_name: str) -> Dict:
        state = self.model.model_state()
        return {"integrated_sw": state["sw"],
                "integrated_aw_kb": state["aw_kb"],
                "integrated_bn": state["bn_params"],
                "pre_trained_params": state["pre_trained_params"]}
