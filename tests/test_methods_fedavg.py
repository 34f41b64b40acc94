"""FedAvg state exchange + aggregation math (ref:methods/fedavg.py)."""

import numpy as np
import torch
import torch.nn as nn

from flreid_amd.methods import methods
from flreid_amd.methods.common import weighted_average_states
from flreid_amd.modules.model import ModelModule


class TinyModel(ModelModule):
    def __init__(self):
        super().__init__(nn.Linear(3, 2))


def _op():
    fedavg = methods["fedavg"]
    m = TinyModel()
    opt = torch.optim.SGD(m.parameters(), lr=0.1)
    return fedavg.Operator(criterion=[], optimizer=opt)


def test_weighted_average_matches_numpy():
    states = {
        "c0": {"train_cnt": 3, "p": {"w": torch.tensor([1.0, 2.0])}},
        "c1": {"train_cnt": 1, "p": {"w": torch.tensor([5.0, 6.0])}},
    }
    counts = {c: s["train_cnt"] for c, s in states.items()}
    merged = weighted_average_states(states, "p", counts)
    expected = (np.array([1.0, 2.0]) * 3 + np.array([5.0, 6.0]) * 1) / 4
    assert np.allclose(merged["w"].numpy(), expected)


def test_server_calculate_stale_upload_semantics(tmp_path):
    """Offline clients contribute their cached uploads
    (ref:methods/fedavg.py:388-390)."""
    fedavg = methods["fedavg"]
    server = fedavg.Server("server", TinyModel(), _op(), str(tmp_path))
    server.register_client("c0")
    server.register_client("c1")
    w0 = {"train_cnt": 2, "incremental_model_params": {
        "net.weight": torch.ones(2, 3), "net.bias": torch.zeros(2)}}
    w1 = {"train_cnt": 2, "incremental_model_params": {
        "net.weight": torch.full((2, 3), 3.0), "net.bias": torch.ones(2)}}
    server.set_client_incremental_state("c0", w0)
    server.set_client_incremental_state("c1", w1)
    server.calculate()
    assert torch.allclose(server.model.net.weight, torch.full((2, 3), 2.0))
    # next round: only c0 uploads fresh state; c1's stays cached
    w0b = {"train_cnt": 6, "incremental_model_params": {
        "net.weight": torch.zeros(2, 3), "net.bias": torch.zeros(2)}}
    server.set_client_incremental_state("c0", w0b)
    server.calculate()
    expected = (0.0 * 6 + 3.0 * 2) / 8
    assert torch.allclose(server.model.net.weight, torch.full((2, 3), expected))


def test_client_upload_contains_only_trainable(tmp_path):
    fedavg = methods["fedavg"]
    model = TinyModel()
    model.net.bias.requires_grad = False
    client = fedavg.Client("c", model, _op(), str(tmp_path))
    state = client.get_incremental_state()
    assert "net.weight" in state["incremental_model_params"]
    assert "net.bias" not in state["incremental_model_params"]
    assert state["train_cnt"] == 0


def test_dispatch_update_roundtrip(tmp_path):
    fedavg = methods["fedavg"]
    server = fedavg.Server("server", TinyModel(), _op(), str(tmp_path))
    client = fedavg.Client("c", TinyModel(), _op(), str(tmp_path))
    state = server.get_dispatch_integrated_state("c")
    client.update_by_integrated_state(state)
    for (n, ps), (_, pc) in zip(server.model.state_dict().items(),
                                client.model.state_dict().items()):
        assert torch.allclose(ps, pc), n
