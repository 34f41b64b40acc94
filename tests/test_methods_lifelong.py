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


def test_importance_values_match_manual():
    """Golden math: Fisher = Σ_batches g²·(|batch|/n_batches) for MAS/EWC
    accumulation (methods/_importance.py; ref:methods/ewc.py:56-78 scale)."""
    from flreid_amd.methods import methods
    torch.manual_seed(1)

    class Op:
        @staticmethod
        def _invoke_train(model, data, target):
            return {"loss": (model.net(data) ** 2).sum()}

    net = nn.Linear(3, 2, bias=False)
    w0 = net.weight.detach().clone()
    data = torch.randn(4, 3)
    loader = [(data, torch.zeros(4, dtype=torch.long), torch.zeros(4))]

    mas = methods["mas"].Model(net=net, operator=Op())
    mas.remember_task("t0", loader)

    # manual: single batch -> scale = len(batch)/n_batches = 4/1
    w = w0.clone().requires_grad_(True)
    loss = ((data @ w.t()) ** 2).sum()
    loss.backward()
    expected = w.grad.abs() * 4.0           # MAS: |g|·scale
    got = mas.precision_matrices["weight"]
    assert torch.allclose(got, expected, atol=1e-5), (got, expected)

    # penalty = λ·Σ F·(p−p_old)² — zero right after calculate, positive after drift
    assert float(mas.penalty()) == 0.0
    with torch.no_grad():
        net.weight += 0.1
    assert float(mas.penalty()) > 0.0


def test_fedcurv_cross_client_penalty_manual():
    """FedCurv penalty = λ·[own-EWC + Σ_other F_o·(p−p_o)²]
    (ref:methods/fedcurv.py:79-86)."""
    from flreid_amd.methods import methods
    torch.manual_seed(2)

    class Op:
        @staticmethod
        def _invoke_train(model, data, target):
            return {"loss": (model.net(data) ** 2).sum()}

    net = nn.Linear(3, 2, bias=False)
    m = methods["fedcurv"].Model(net=net, operator=Op(), lambda_penalty=2.0)
    # no remembered tasks -> own term zero; inject one other client
    f_o = {"weight": torch.rand(2, 3)}
    p_o = {"weight": net.weight.detach() + 0.5}
    m.other_precision_matrices = [(f_o, p_o)]
    pen = m.penalty()
    expected = 2.0 * (f_o["weight"] * 0.25).sum()
    assert torch.allclose(pen, expected, atol=1e-6)
