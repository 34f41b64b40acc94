"""Hermetic integration test — BASELINE config 1 shape:
FedAvg, resnet18, 2 clients x tasks on CPU, single process."""

import json
import os

from flreid_amd.parallel.comm import FedContext
from flreid_amd.runtime.experiment import ExperimentStage


def test_fedavg_e2e(tiny_common, tiny_exp_config, tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    stage = ExperimentStage(tiny_common, [tiny_exp_config], ctx=FedContext())
    stage.check_environment()
    log = stage.run_experiment(tiny_exp_config)

    data = log.records["data"]
    assert set(data.keys()) == {"client-0", "client-1"}
    # round 0 = initial validation; round "2" has train + val records
    r2 = data["client-0"]["2"]
    any_task = next(iter(r2.values()))
    assert "val_map" in any_task or "tr_acc" in any_task

    # ckpt audit layout (ref:experiment.py:199-202,233-241 + ckpts/README.md)
    server_dir = os.path.join(tiny_common["checkpoints_dir"], "tiny", "server")
    assert os.path.exists(os.path.join(server_dir, "1-server-client-0.ckpt"))
    client_dir = os.path.join(tiny_common["checkpoints_dir"], "tiny", "client-0")
    assert os.path.exists(os.path.join(client_dir, "1-client-0-server.ckpt"))
    assert os.path.exists(os.path.join(client_dir, "fedavg_model.ckpt"))

    # json log written
    logs = os.listdir(tiny_common["logs_dir"])
    assert len(logs) == 1
    payload = json.loads(open(os.path.join(tiny_common["logs_dir"], logs[0])).read())
    assert payload["config"]["exp_name"] == "tiny"


def test_baseline_method_e2e(tiny_common, tiny_exp_config, tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    cfg = dict(tiny_exp_config)
    cfg["exp_name"] = "tiny-baseline"
    cfg["exp_method"] = "baseline"
    stage = ExperimentStage(tiny_common, [cfg], ctx=FedContext())
    log = stage.run_experiment(cfg)
    assert "client-0" in log.records["data"]
