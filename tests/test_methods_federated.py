"""iCaRL / FedCurv / FedWeIT / FedSTIL-atten semantics + e2e."""

import torch
import torch.nn as nn

from flreid_amd.parallel.comm import FedContext
from flreid_amd.runtime.experiment import ExperimentStage


def _mk(method, tiny_exp_config, extra_model=None, server=None):
    cfg = dict(tiny_exp_config)
    cfg["exp_name"] = f"tiny-{method}"
    cfg["exp_method"] = method
    if extra_model:
        cfg["model_opts"] = {**cfg["model_opts"], **extra_model}
    if server:
        cfg["server"] = {**cfg["server"], **server}
    return cfg


# --------------------------------------------------------------------- units

def test_decomposed_layer_compose_and_pruning():
    from flreid_amd.models.decomposed import DecomposedLinear, l1_hard_threshold
    lin = nn.Linear(4, 3)
    layer = DecomposedLinear(shared_weight=lin.weight, bias=lin.bias,
                             lambda_l1=1e-3, lambda_mask=0.0, kb_cnt=5)
    assert layer.mask.shape == (3,)          # per-output-channel
    assert layer.aw_kb.shape == (3, 4, 5)
    assert layer.atten.shape == (5,)
    # eval composition: mask=0.5 everywhere, aw=(1-mask)sw, kb=0
    layer.eval()
    theta = layer.composed_weight()
    assert torch.allclose(theta, lin.weight, atol=1e-6)
    # pruning zeroes sub-threshold entries
    w = torch.tensor([0.5, -0.0005, 0.002])
    assert torch.allclose(l1_hard_threshold(w, 1e-3),
                          torch.tensor([0.5, 0.0, 0.002]))


def test_fedweit_kb_stacking(tmp_path):
    from flreid_amd.methods import methods
    fw = methods["fedweit"]
    from flreid_amd.runtime.builder import parser_model

    def mk_model():
        return parser_model("fedweit", {
            "name": "resnet18", "num_classes": 32, "last_stride": 1,
            "neck": "bnneck", "lambda_l1": 5e-6, "lambda_l2": 1e-3,
            "lambda_mask": 0.0, "kb_cnt": 2,
            "fine_tuning": ["classifier"]})

    model = mk_model()
    op = fw.Operator(criterion=[], optimizer=torch.optim.SGD(
        [p for p in model.net.parameters() if p.requires_grad], lr=0.1))
    server = fw.Server("server", model, op, str(tmp_path))
    for c in ("c0", "c1"):
        server.register_client(c)
        cm = mk_model()
        client = fw.Client(c, cm, fw.Operator(criterion=[],
                           optimizer=torch.optim.SGD([p for p in cm.net.parameters()
                                                      if p.requires_grad], lr=0.1)),
                           str(tmp_path))
        client.train_cnt = 4
        server.set_client_incremental_state(c, client.get_incremental_state())
    server.calculate()
    # kb built from 2 clients -> aw_kb last dim == kb_cnt == 2
    _, layer = server.model.decomposed_module_leaves()[0]
    assert layer.aw_kb.shape[-1] == 2
    d = server.get_dispatch_incremental_state("c0")
    assert any(k.endswith(".sw") for k in d["incremental_sw"])


def test_fedstil_atten_server_stacks(tmp_path):
    from flreid_amd.methods import methods
    fa = methods["fedstil-atten"]
    from flreid_amd.runtime.builder import parser_model

    opts = {"name": "resnet18", "num_classes": 32, "last_stride": 1,
            "neck": "bnneck", "atten_default": 0.0, "lambda_l1": 1e-5,
            "lambda_k": 32, "fine_tuning": ["classifier"]}

    server_model = parser_model("fedstil-atten", opts)
    op = fa.Operator(criterion=[], optimizer=torch.optim.SGD(
        [p for p in server_model.net.parameters() if p.requires_grad], lr=0.1))
    server = fa.Server("server", server_model, op, str(tmp_path),
                       distance_calculate_step=10, distance_calculate_decay=0.8)
    for c in ("c0", "c1"):
        server.register_client(c)
        cm = parser_model("fedstil-atten", opts)
        client = fa.Client(c, cm, fa.Operator(criterion=[],
                           optimizer=torch.optim.SGD([p for p in cm.net.parameters()
                                                      if p.requires_grad], lr=0.1)),
                           str(tmp_path))
        client.train_cnt = 4
        client.task_token = torch.randn(8)
        server.set_client_incremental_state(c, client.get_incremental_state())
    server.calculate()
    _n, layer = server.model.adaptive_module_leaves()[0]
    assert layer.global_weight.shape[-1] == 2    # stacked client dim

    # dispatch re-init resizes the learnable attention to the stack depth
    cm = parser_model("fedstil-atten", opts)
    client = fa.Client("c0", cm, fa.Operator(criterion=[],
                       optimizer=torch.optim.SGD([p for p in cm.net.parameters()
                                                  if p.requires_grad], lr=0.1)),
                       str(tmp_path))
    client.update_by_incremental_state(server.get_dispatch_incremental_state("c0"))
    _n, clayer = client.model.adaptive_module_leaves()[0]
    assert clayer.global_weight_atten.shape == (2,)
    assert clayer.global_weight_atten.requires_grad


# ----------------------------------------------------------------------- e2e

def test_icarl_e2e(tiny_common, tiny_exp_config, tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    cfg = _mk("icarl", tiny_exp_config,
              {"num_classes": 10, "k": 24, "n_classes": 10})
    cfg["criterion_opts"] = {"name": "cross_entropy", "num_classes": 128,
                             "epsilon": 0.1}
    stage = ExperimentStage(tiny_common, [cfg], ctx=FedContext())
    log = stage.run_experiment(cfg)
    assert "client-0" in log.records["data"]


def test_fedcurv_e2e(tiny_common, tiny_exp_config, tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    cfg = _mk("fedcurv", tiny_exp_config, {"lambda_penalty": 1.0})
    stage = ExperimentStage(tiny_common, [cfg], ctx=FedContext())
    log = stage.run_experiment(cfg)
    r1 = log.records["data"]["client-0"].get("1", {})
    assert any("tr_acc" in v for v in r1.values())


def test_fedweit_e2e(tiny_common, tiny_exp_config, tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    cfg = _mk("fedweit", tiny_exp_config,
              {"lambda_l1": 5e-6, "lambda_l2": 1e-3, "lambda_mask": 0.0,
               "kb_cnt": 2})
    stage = ExperimentStage(tiny_common, [cfg], ctx=FedContext())
    log = stage.run_experiment(cfg)
    assert "client-1" in log.records["data"]


def test_fedstil_atten_e2e(tiny_common, tiny_exp_config, tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    cfg = _mk("fedstil-atten", tiny_exp_config,
              {"atten_default": 0.0, "lambda_l1": 1e-5, "lambda_k": 32},
              server={"distance_calculate_step": 10,
                      "distance_calculate_decay": 0.8})
    stage = ExperimentStage(tiny_common, [cfg], ctx=FedContext())
    log = stage.run_experiment(cfg)
    assert "client-0" in log.records["data"]
