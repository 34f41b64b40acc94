"""Unit tests for tools/utils (ref:tools/utils.py parity helpers)."""

import torch

from flreid_amd.tools.utils import (get_one_hot, model_on_device,
                                    params_state_size, same_seeds,
                                    tensor_reverse_permute, trainable_params)


def test_same_seeds_reproducible():
    same_seeds(123)
    a = torch.randn(4)
    same_seeds(123)
    b = torch.randn(4)
    assert torch.equal(a, b)


def test_get_one_hot():
    t = torch.tensor([0, 2, 1])
    oh = get_one_hot(t, 3)
    assert oh.shape == (3, 3)
    assert torch.equal(oh.argmax(dim=1), t)
    assert oh.sum() == 3


def test_tensor_reverse_permute_roundtrip():
    t = torch.randn(2, 3, 4)
    r = tensor_reverse_permute(t)
    assert r.shape == (4, 3, 2)
    assert torch.equal(tensor_reverse_permute(r), t)
    # 1-D and scalar stay unchanged
    v = torch.randn(5)
    assert torch.equal(tensor_reverse_permute(v), v)


def test_params_state_size_live():
    state = {"a": torch.zeros(3, 4), "n": [torch.zeros(2), {"b": torch.zeros(5)}],
             "s": 7}
    # bytes of every tensor (fp32) + 8 per python scalar
    assert params_state_size(state) == (12 + 2 + 5) * 4 + 8


def test_model_on_device_restores_cpu(monkeypatch):
    monkeypatch.setenv("FLREID_RESIDENT", "0")
    from flreid_amd.tools import utils as U
    monkeypatch.setattr(U, "_RESIDENT", False)
    m = torch.nn.Linear(4, 2)
    with model_on_device(m, "cpu"):
        assert next(m.parameters()).device.type == "cpu"
    assert next(m.parameters()).device.type == "cpu"


def test_trainable_params_filters():
    m = torch.nn.Sequential(torch.nn.Linear(4, 4), torch.nn.Linear(4, 2))
    for p in m[0].parameters():
        p.requires_grad_(False)
    names = set(trainable_params(m).keys())
    assert names == {"1.weight", "1.bias"}


def test_epoch_graph_auto_threshold(monkeypatch):
    """Long runs auto-enable whole-epoch capture; explicit env always wins
    (runtime/hipgraph.py)."""
    from flreid_amd.runtime import hipgraph as hg

    monkeypatch.delenv("FLREID_EPOCH_GRAPH", raising=False)
    monkeypatch.setattr(hg.torch.cuda, "is_available", lambda: True)

    hg.suggest_epoch_graph(60)
    assert not hg.epoch_graph_enabled()
    hg.suggest_epoch_graph(500)
    assert hg.epoch_graph_enabled()

    monkeypatch.setenv("FLREID_EPOCH_GRAPH", "0")
    assert not hg.epoch_graph_enabled()          # explicit off wins
    monkeypatch.setenv("FLREID_EPOCH_GRAPH", "1")
    hg.suggest_epoch_graph(5)
    assert hg.epoch_graph_enabled()              # explicit on wins
    hg.suggest_epoch_graph(0)                    # reset module state
