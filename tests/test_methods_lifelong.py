"""EWC / MAS / FedProx semantics + e2e (ref:methods/{ewc,mas,fedprox}.py)."""

import torch
import torch.nn as nn

from flreid_amd.parallel.comm import FedContext
from flreid_amd.runtime.experiment import ExperimentStage


def _mk(method, tiny_exp_config, extra_model=None):
    cfg = dict(tiny_exp_config)
    cfg["exp_name"] = f"tiny-{method}"
    cfg["exp_method"] = method
    if extra_model:
        cfg["model_opts"] = {**cfg["model_opts"], **extra_model}
    return cfg


def test_importance_model_modes():
    from flreid_amd.methods import methods
    torch.manual_seed(0)

    class Op:
        @staticmethod
        def _invoke_train(model, data, target):
            score = model.net(data)
            return {"loss": ((score - 1.0) ** 2).sum()}

    net = nn.Linear(4, 2)
    ewc_model = methods["ewc"].Model(net=net, operator=Op())
    data = torch.randn(16, 4)
    loader = [(data[:8], torch.zeros(8, dtype=torch.long), torch.zeros(8)),
              (data[8:], torch.zeros(8, dtype=torch.long), torch.zeros(8))]
    # EWC skips the current (only) task -> importance stays zero
    ewc_model.remember_task("t0", loader)
    assert all((v == 0).all() for v in ewc_model.precision_matrices.values())
    # second task: first task replayed, importance becomes positive
    ewc_model.remember_task("t1", loader)
    assert any((v > 0).any() for v in ewc_model.precision_matrices.values())

    mas_model = methods["mas"].Model(net=nn.Linear(4, 2), operator=Op())
    mas_model.remember_task("t0", loader)   # MAS includes current task
    assert any((v > 0).any() for v in mas_model.precision_matrices.values())


def test_fedprox_penalty_anchors_on_remembered():
    from flreid_amd.methods import methods
    model = methods["fedprox"].Model(net=nn.Linear(2, 2), lambda_l2=0.5)
    assert float(model.penalty()) == 0.0
    model.remember_params()
    with torch.no_grad():
        model.net.weight += 1.0
    expected = 0.5 * 4.0   # 4 weight entries moved by 1, bias unchanged
    assert abs(float(model.penalty()) - expected) < 1e-6


def test_ewc_e2e(tiny_common, tiny_exp_config, tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    cfg = _mk("ewc", tiny_exp_config, {"lambda_penalty": 10.0})
    stage = ExperimentStage(tiny_common, [cfg], ctx=FedContext())
    log = stage.run_experiment(cfg)
    assert "client-0" in log.records["data"]


def test_mas_e2e(tiny_common, tiny_exp_config, tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    cfg = _mk("mas", tiny_exp_config, {"lambda_penalty": 10.0})
    stage = ExperimentStage(tiny_common, [cfg], ctx=FedContext())
    log = stage.run_experiment(cfg)
    assert "client-1" in log.records["data"]


def test_fedprox_e2e(tiny_common, tiny_exp_config, tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    cfg = _mk("fedprox", tiny_exp_config, {"lambda_l2": 1e-2})
    stage = ExperimentStage(tiny_common, [cfg], ctx=FedContext())
    log = stage.run_experiment(cfg)
    r1 = log.records["data"]["client-0"].get("1", {})
    assert any("tr_acc" in v for v in r1.values())
