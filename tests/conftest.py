import os
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a ROCm GPU (runs on MI355X boxes only)")


def pytest_collection_modifyitems(config, items):
    import torch
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture()
def tiny_exp_config():
    """BASELINE config 1 scaled down: FedAvg, resnet18, 2 clients, synthetic."""
    return {
        "exp_name": "tiny", "exp_method": "fedavg", "random_seed": 7,
        "exp_opts": {"comm_rounds": 2, "val_interval": 2, "online_clients": 2},
        "model_opts": {"name": "resnet18", "num_classes": 128, "last_stride": 1,
                       "neck": "bnneck",
                       "fine_tuning": ["base.layer4", "classifier"]},
        "criterion_opts": {"name": "cross_entropy", "num_classes": 128,
                           "epsilon": 0.1},
        "optimizer_opts": {"name": "adam", "lr": 1e-3, "weight_decay": 1e-5},
        "scheduler_opts": {"name": "step_lr", "step_size": 5},
        "task_opts": {"sustain_rounds": 1, "train_epochs": 1,
                      "augment_opts": {"level": "default", "img_size": [32, 16],
                                       "norm_mean": [0.485, 0.456, 0.406],
                                       "norm_std": [0.229, 0.224, 0.225]},
                      "loader_opts": {"batch_size": 8, "num_workers": 0,
                                      "pin_memory": False,
                                      "persistent_workers": False,
                                      "multiprocessing_context": None}},
        "server": {"server_name": "server"},
        "clients": [
            {"client_name": "client-0", "tasks": ["task-0-0", "task-0-1"]},
            {"client_name": "client-1", "tasks": ["task-1-0", "task-1-1"]},
        ],
    }


@pytest.fixture()
def tiny_common(tmp_path):
    return {
        "datasets_dir": "synthetic://ids=4,train=3,query=2,gallery=3,hw=32x16,idspace=48",
        "checkpoints_dir": str(tmp_path / "ckpts"),
        "logs_dir": str(tmp_path / "logs"),
        "parallel": 1, "device": ["cpu"], "defaults": {},
    }
