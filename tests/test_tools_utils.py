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


def test_fedcontext_single_process_collectives():
    """World-size-1 FedContext: every collective is the identity
    (parallel/comm.py — the reference-simulator-compatible mode)."""
    from flreid_amd.parallel.comm import FedContext

    ctx = FedContext()
    assert not ctx.is_distributed and ctx.is_rank0()
    assert [ctx.owner_of(i) for i in range(4)] == [0, 0, 0, 0]
    assert ctx.all_gather_object({"a": 1}) == [{"a": 1}]
    assert ctx.broadcast_object(5) == 5
    assert ctx.all_reduce_scalar(2.5) == 2.5
    flat = ctx.all_gather_flat(torch.arange(3.0))
    assert flat.shape == (1, 3) and torch.equal(flat[0], torch.arange(3.0))
    out = ctx.weighted_allreduce({"w": torch.ones(2)}, 0.5)
    assert torch.allclose(out["w"], torch.full((2,), 0.5))


def test_synthetic_uri_parsing():
    from flreid_amd.data.synthetic import parse_synthetic_dir

    opts = parse_synthetic_dir(
        "synthetic://ids=8,train=4,query=2,gallery=3,hw=64x32,idspace=128")
    assert opts["ids"] == 8 and opts["train"] == 4
    assert opts["shape"] == (3, 64, 32) and opts["idspace"] == 128
    # defaults fill unspecified keys
    d = parse_synthetic_dir("synthetic://ids=2")
    assert d["ids"] == 2 and "train" in d and d["shape"] == (3, 128, 64)
    # non-synthetic paths decline
    assert parse_synthetic_dir("/data/reid") is None


def test_async_ckpt_writer_ordering(tmp_path):
    """runtime/io.py: same-path writes apply in submission order; flush
    drains; the snapshot decouples from later mutation."""
    from flreid_amd.runtime.io import AsyncCkptWriter

    w = AsyncCkptWriter()
    path = str(tmp_path / "s.ckpt")
    live = {"x": torch.zeros(4)}
    w.submit(path, live)
    live["x"].fill_(1.0)          # mutation AFTER submit must not leak
    w.flush()
    assert torch.equal(torch.load(path, weights_only=False)["x"],
                       torch.zeros(4))

    for i in range(5):
        w.submit(path, {"i": torch.full((2,), float(i))})
    w.flush()
    assert torch.equal(torch.load(path, weights_only=False)["i"],
                       torch.full((2,), 4.0))


def test_save_ckpt_async_env(tmp_path, monkeypatch):
    """save_ckpt routes through the async writer when FLREID_ASYNC_CKPT=1
    and before_ckpt_read drains it."""
    from flreid_amd.runtime.io import before_ckpt_read, save_ckpt

    monkeypatch.setenv("FLREID_ASYNC_CKPT", "1")
    p = str(tmp_path / "a.ckpt")
    save_ckpt(p, {"v": torch.ones(3)})
    before_ckpt_read()
    assert torch.equal(torch.load(p, weights_only=False)["v"], torch.ones(3))
