"""Dependency-injection builders (ref:builder.py:16-104).

Same parser surface: model / criterion / optimizer / scheduler / server /
clients, driven by the merged experiment yaml.  `parser_clients` additionally
accepts the rank context so each process only instantiates the clients it
owns (one process per GPU; ref's thread-pool sharing is gone).
"""

from __future__ import annotations

import os
from typing import Any, Callable, Dict, List, Optional

import torch.nn as nn
from torch.optim import Optimizer

from flreid_amd.criterions import criterions
from flreid_amd.data.pipeline import ReIDTaskPipeline
from flreid_amd.methods import methods
from flreid_amd.models import nets, optimizers, schedulers
from flreid_amd.modules.client import ClientModule
from flreid_amd.modules.model import ModelModule
from flreid_amd.modules.server import ServerModule


def parser_model(method_name: str, model_config: Dict) -> nn.Module:
    factory_kwargs = {n: p for n, p in model_config.items()
                      if n not in ("name", "fine_tuning")}
    net = nets[model_config["name"]](**factory_kwargs)
    if model_config.get("fine_tuning"):
        for p in net.parameters():
            p.requires_grad = False
        for layer_name in model_config["fine_tuning"]:
            for p in net.get_submodule(layer_name).parameters():
                p.requires_grad = True

    method = methods[method_name]
    if hasattr(method, "Model"):
        return method.Model(net=net, **factory_kwargs)
    return ModelModule(net)


def parser_criterion(criterion_configs: Any) -> List[Callable]:
    if isinstance(criterion_configs, dict):
        criterion_configs = [criterion_configs]
    out = []
    for cfg in criterion_configs:
        kwargs = {n: p for n, p in cfg.items() if n != "name"}
        out.append(criterions[cfg["name"]](**kwargs))
    return out


def parser_optimizer(model: nn.Module, optim_config: Dict) -> Optimizer:
    kwargs = {n: p for n, p in optim_config.items() if n != "name"}
    params = [p for p in model.net.parameters() if p.requires_grad]
    return optimizers[optim_config["name"]](params=params, **kwargs)


def parser_scheduler(optim: Optimizer, scheduler_config: Dict):
    kwargs = {n: p for n, p in scheduler_config.items() if n != "name"}
    return schedulers[scheduler_config["name"]](optimizer=optim, **kwargs)


def _build_operator(exp_config: Dict) -> Any:
    model = parser_model(exp_config["exp_method"], exp_config["model_opts"])
    criterion = parser_criterion(exp_config["criterion_opts"])
    optimizer = parser_optimizer(model, exp_config["optimizer_opts"])
    scheduler = parser_scheduler(optimizer, exp_config["scheduler_opts"])
    operator = methods[exp_config["exp_method"]].Operator(
        method_name=exp_config["exp_method"],
        criterion=criterion, optimizer=optimizer, scheduler=scheduler,
    )
    return model, operator


def _actor_seed(exp_config: Dict, actor: str) -> int:
    import hashlib
    h = hashlib.sha256(f"{exp_config.get('random_seed', 0)}|{actor}".encode())
    return int.from_bytes(h.digest()[:4], "little") % (2 ** 31)


def parser_server(exp_config: Dict, common_config: Dict) -> ServerModule:
    # deterministic per-actor init: a client's weights do not depend on which
    # rank builds it or on sibling build order (placement-invariant sharding)
    from flreid_amd.tools.utils import same_seeds
    same_seeds(_actor_seed(exp_config, "server"))
    model, operator = _build_operator(exp_config)
    kwargs = {n: p for n, p in exp_config["server"].items() if n != "server_name"}
    return methods[exp_config["exp_method"]].Server(
        server_name=exp_config["server"]["server_name"],
        model=model, operator=operator,
        ckpt_root=os.path.join(common_config["checkpoints_dir"], exp_config["exp_name"]),
        **kwargs,
    )


def parser_clients(exp_config: Dict, common_config: Dict,
                   owned_indices: Optional[List[int]] = None) -> List[ClientModule]:
    """Build client objects; with `owned_indices`, only those (rank sharding)."""
    from flreid_amd.tools.utils import same_seeds
    clients = []
    for idx, client_config in enumerate(exp_config["clients"]):
        if owned_indices is not None and idx not in owned_indices:
            continue
        same_seeds(_actor_seed(exp_config, client_config["client_name"]))
        model, operator = _build_operator(exp_config)
        task_pipeline = ReIDTaskPipeline(
            task_list=client_config["tasks"],
            task_opts=exp_config["task_opts"],
            datasets_dir=common_config["datasets_dir"],
        )
        kwargs = {n: p for n, p in client_config.items() if n != "client_name"}
        clients.append(methods[exp_config["exp_method"]].Client(
            client_name=client_config["client_name"],
            model=model, operator=operator,
            ckpt_root=os.path.join(common_config["checkpoints_dir"], exp_config["exp_name"]),
            task_pipeline=task_pipeline,
            **kwargs,
        ))
    return clients
