#!/usr/bin/env python3
"""Per-round communication cost by method (ResNet-50 ReID config).

Measures the bytes each method ships per client per round (upload =
`get_incremental_state`, dispatch = `get_dispatch_*`) with the live
`params_state_size` accountant — the counter the reference shipped as dead
code (ref:tools/utils.py:39-48).  This is the quantitative backing for the
comm-cost story (the paper claims FedSTIL cuts communication
62% vs full-parameter exchange).

Runs on CPU (no training needed): python benchmarks/comm_bench.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("FLREID_DISABLE_CKPT", "1")

import torch

from flreid_amd.methods import methods
from flreid_amd.runtime.builder import parser_clients, parser_server
from flreid_amd.tools.utils import params_state_size

MODEL = {"name": "resnet50", "num_classes": 8000, "last_stride": 1,
         "neck": "bnneck", "fine_tuning": ["base.layer4", "classifier"]}

EXTRA = {
    "fedavg": {},
    "fedprox": {"lambda_l2": 1e-5},
    "fedcurv": {"lambda_penalty": 50.0},
    "fedweit": {"lambda_l1": 5e-6, "lambda_l2": 1e-3, "lambda_mask": 0.0,
                "kb_cnt": 5},
    "fedstil": {"atten_default": 0.9, "lambda_l1": 1e-3, "lambda_k": 12000},
    "fedstil-atten": {"atten_default": 0.0, "lambda_l1": 1e-5,
                      "lambda_k": 12000},
}
SERVER = {
    "fedstil": {"distance_calculate_step": 10, "distance_calculate_decay": 0.8},
    "fedstil-atten": {"distance_calculate_step": 10,
                      "distance_calculate_decay": 0.8},
}


def build(method):
    exp = {
        "exp_name": f"comm-{method}", "exp_method": method, "random_seed": 1,
        "model_opts": {**MODEL, **EXTRA[method]},
        "criterion_opts": {"name": "cross_entropy", "num_classes": 8000,
                           "epsilon": 0.1},
        "optimizer_opts": {"name": "adam", "lr": 1e-3},
        "scheduler_opts": {"name": "step_lr", "step_size": 5},
        "server": {"server_name": "server", **SERVER.get(method, {})},
        "clients": [{"client_name": "client-0", "tasks": ["task-0-0"]}],
        "task_opts": {"sustain_rounds": 1, "train_epochs": 1,
                      "augment_opts": {"level": "none", "img_size": [128, 64],
                                       "norm_mean": [0.5] * 3,
                                       "norm_std": [0.5] * 3},
                      "loader_opts": {"batch_size": 64, "num_workers": 0,
                                      "pin_memory": False,
                                      "persistent_workers": False,
                                      "multiprocessing_context": None}},
    }
    common = {"datasets_dir": "synthetic://", "checkpoints_dir": "/tmp/commck",
              "logs_dir": "/tmp/commlg", "parallel": 1, "device": ["cpu"],
              "defaults": {}}
    server = parser_server(exp, common)
    client = parser_clients(exp, common)[0]
    return server, client


def mb(x):
    return x / 1e6


def main():
    print(f"{'method':14s} {'upload/client':>14s} {'dispatch':>14s} "
          f"{'full state (ref)':>17s}")
    for method in EXTRA:
        server, client = build(method)
        client.train_cnt = 512
        client.task_token = torch.randn(2048)
        up = client.get_incremental_state()
        up_bytes = params_state_size(up)
        server.register_client("client-0")
        if up is not None:
            server.set_client_incremental_state("client-0", up)
            try:
                server.calculate()
            except Exception:
                pass
        disp = server.get_dispatch_incremental_state("client-0")
        disp_bytes = params_state_size(disp)
        full_bytes = params_state_size(
            {n: p for n, p in client.model.state_dict().items()})
        print(f"{method:14s} {mb(up_bytes):11.1f} MB {mb(disp_bytes):11.1f} MB "
              f"{mb(full_bytes):14.1f} MB")


if __name__ == "__main__":
    main()
