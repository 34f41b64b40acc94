"""ExperimentLog semantics (ref:experiment.py:16-55)."""

import json

from flreid_amd.runtime.log import ExperimentLog


def test_dotted_nested_keys(tmp_path):
    log = ExperimentLog(str(tmp_path / "log.json"))
    log.record("data.client-0.3.task-0-1", {"tr_acc": 0.5})
    log.record("data.client-0.3.task-0-1", {"tr_loss": 1.0})  # dict merge
    assert log.records["data"]["client-0"]["3"]["task-0-1"] == {
        "tr_acc": 0.5, "tr_loss": 1.0}


def test_list_append_and_scalar_replace(tmp_path):
    log = ExperimentLog(str(tmp_path / "log.json"))
    log.record("a", [1])
    log.record("a", 2)   # appended to existing list
    assert log.records["a"] == [1, 2]
    log.record("b", 1)
    log.record("b", 3)   # scalar replaced
    assert log.records["b"] == 3


def test_flush_writes_json(tmp_path):
    path = tmp_path / "x" / "log.json"
    log = ExperimentLog(str(path))
    log.record("config", {"exp_name": "t"})
    log.flush()
    assert json.loads(path.read_text())["config"]["exp_name"] == "t"


def test_async_ckpt_write_then_read(tmp_path, monkeypatch):
    """Async writer ordering: a read after a submitted write sees the data."""
    import torch
    monkeypatch.setenv("FLREID_ASYNC_CKPT", "1")
    from flreid_amd.runtime import io as rio
    p = str(tmp_path / "x.ckpt")
    rio.save_ckpt(p, {"w": torch.ones(4)})
    rio.before_ckpt_read()
    assert torch.load(p, weights_only=False)["w"].sum() == 4


def test_maybe_sync_batches_distributed_gathers():
    """maybe_sync gathers only every FLREID_LOG_SYNC_EVERY rounds (and on
    the final round); single-process mode flushes every call."""
    import os

    from flreid_amd.runtime.log import ExperimentLog

    class _FakeCtx:
        is_distributed = True

        def __init__(self):
            self.gathers = 0
            self.rank = 0

        def is_rank0(self):
            return True

        def all_gather_object(self, obj):
            self.gathers += 1
            return [obj]

    os.environ["FLREID_LOG_SYNC_EVERY"] = "5"
    try:
        log = ExperimentLog("/tmp/flreid_test_log.json")
        ctx = _FakeCtx()
        for r in range(1, 13):
            log.record(f"data.c.{r}.t", {"x": r})
            log.maybe_sync(ctx, r, 12)
        # rounds 5, 10 and the final round 12
        assert ctx.gathers == 3
        assert log.records["data"]["c"]["12"]["t"]["x"] == 12
    finally:
        del os.environ["FLREID_LOG_SYNC_EVERY"]
