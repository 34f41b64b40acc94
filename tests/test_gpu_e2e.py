"""GPU end-to-end: FedSTIL/FedAvg rounds on a real MI355X through the native
ops path (no eager fallback)."""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


def _common(tmp_path):
    return {
        "datasets_dir": "synthetic://ids=8,train=4,query=2,gallery=3,hw=64x32,idspace=128",
        "checkpoints_dir": str(tmp_path / "ckpts"),
        "logs_dir": str(tmp_path / "logs"),
        "parallel": 1, "device": ["cuda:0"], "defaults": {},
    }


def _exp(method):
    cfg = {
        "exp_name": f"gpu-{method}", "exp_method": method, "random_seed": 3,
        "exp_opts": {"comm_rounds": 2, "val_interval": 2, "online_clients": 2},
        "model_opts": {"name": "resnet50", "num_classes": 256,
                       "last_stride": 1, "neck": "bnneck",
                       "fine_tuning": ["base.layer4", "classifier"]},
        "criterion_opts": {"name": "cross_entropy", "num_classes": 256,
                           "epsilon": 0.1},
        "optimizer_opts": {"name": "adam", "lr": 1e-3, "weight_decay": 1e-5},
        "scheduler_opts": {"name": "step_lr", "step_size": 5},
        "task_opts": {"sustain_rounds": 1, "train_epochs": 1,
                      "augment_opts": {"level": "default", "img_size": [64, 32],
                                       "norm_mean": [0.485, 0.456, 0.406],
                                       "norm_std": [0.229, 0.224, 0.225]},
                      "loader_opts": {"batch_size": 16, "num_workers": 0,
                                      "pin_memory": False,
                                      "persistent_workers": False,
                                      "multiprocessing_context": None}},
        "server": {"server_name": "server"},
        "clients": [
            {"client_name": "client-0", "tasks": ["task-0-0", "task-0-1"]},
            {"client_name": "client-1", "tasks": ["task-1-0", "task-1-1"]},
        ],
    }
    if method in ("fedstil", "fedstil-atten"):
        cfg["model_opts"].update({"atten_default": 0.9 if method == "fedstil" else 0.0,
                                  "lambda_l1": 1e-4, "lambda_k": 64})
        cfg["server"].update({"distance_calculate_step": 10,
                              "distance_calculate_decay": 0.8})
    if method in ("ewc", "fedcurv"):
        cfg["model_opts"]["lambda_penalty"] = 1.0
    if method == "fedweit":
        cfg["model_opts"].update({"lambda_l1": 5e-6, "lambda_l2": 1e-3,
                                  "lambda_mask": 0.0, "kb_cnt": 2})
    if method == "icarl":
        cfg["model_opts"].update({"num_classes": 10, "k": 32, "n_classes": 10})
    return cfg


@pytest.mark.parametrize("method", ["fedavg", "fedstil", "fedprox", "fedcurv", "fedweit", "ewc", "icarl", "fedstil-atten"])
def test_gpu_round(method, tmp_path, monkeypatch):
    from flreid_amd import ops
    assert ops.extension_available()
    monkeypatch.chdir(tmp_path)
    from flreid_amd.parallel.comm import FedContext
    from flreid_amd.runtime.experiment import ExperimentStage

    common = _common(tmp_path)
    cfg = _exp(method)
    ctx = FedContext(device="cuda:0")
    stage = ExperimentStage(common, [cfg], ctx=ctx)
    log = stage.run_experiment(cfg)
    data = log.records["data"]
    assert "client-0" in data and "client-1" in data
    r2 = data["client-0"].get("2", {})
    assert any("val_map" in v for v in r2.values())
    assert "comm" in log.records  # per-round comm-bytes accounting


def test_smoke_entry():
    import __graft_entry__
    __graft_entry__.smoke()


def test_gpu_round_epoch_graph(tmp_path, monkeypatch):
    """FedSTIL with whole-epoch hipGraph capture (FLREID_EPOCH_GRAPH=1,
    runtime/hipgraph.py::EpochGraph): rounds must train and validate like
    the per-step-graph path."""
    from flreid_amd import ops
    assert ops.extension_available()
    monkeypatch.chdir(tmp_path)
    monkeypatch.setenv("FLREID_EPOCH_GRAPH", "1")
    from flreid_amd.parallel.comm import FedContext
    from flreid_amd.runtime.experiment import ExperimentStage

    common = _common(tmp_path)
    cfg = _exp("fedstil")
    cfg["exp_name"] = "gpu-fedstil-epochgraph"
    cfg["exp_opts"]["comm_rounds"] = 4   # reach capture + replay rounds
    cfg["exp_opts"]["val_interval"] = 4
    ctx = FedContext(device="cuda:0")
    stage = ExperimentStage(common, [cfg], ctx=ctx)
    log = stage.run_experiment(cfg)
    data = log.records["data"]
    r4 = data["client-0"].get("4", {})
    assert any("val_map" in v for v in r4.values())
    for r in ("1", "2", "3", "4"):
        rec = data["client-0"].get(r, {})
        tr = [v for v in rec.values() if "tr_loss" in v]
        assert tr and all(v["tr_loss"] == v["tr_loss"] for v in tr)  # finite


def test_gpu_round_fedstil_swin(tmp_path, monkeypatch):
    """FedSTIL over the Swin backbone on GPU: 3-D token-grid taps through
    the device-resident prototype path (regression: the 4-D NHWC permute
    must not touch Swin's [B, L, C] taps), fused window attention + K4
    PatchMerging in the loop."""
    monkeypatch.setenv("FLREID_DISABLE_CKPT", "1")
    from flreid_amd.runtime.builder import parser_clients, parser_server
    from flreid_amd.runtime.experiment import ExperimentStage
    from flreid_amd.runtime.log import ExperimentLog
    from flreid_amd.tools.utils import same_seeds

    common = _common(tmp_path)
    exp = _exp("fedstil")
    exp["model_opts"].update({"name": "swin_transformer_tiny",
                              "fine_tuning": ["base.layers.3", "classifier"]})
    same_seeds(3)
    stage = ExperimentStage(common, [exp])
    log = ExperimentLog(str(tmp_path / "logs" / "log.json"))
    server = parser_server(exp, common)
    names = [c["client_name"] for c in exp["clients"]]
    clients = {c.client_name: c for c in parser_clients(exp, common)}
    for r in (1, 2):
        stage.process_one_round(r, server, clients, names, exp, log)
    rec = log.records["data"]
    for cname in names:
        assert any("tr_acc" in t for rnd in rec[cname].values()
                   for t in rnd.values())
