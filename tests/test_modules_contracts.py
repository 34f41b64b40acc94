"""Module-contract unit tests (ref API surface — SURVEY.md §2.2):
ckpt I/O layout, cover=False collision raise, CPU-mapped loads,
register/unregister, kwargs-to-attributes."""

import os

import pytest
import torch

from flreid_amd.modules.client import ClientModule
from flreid_amd.modules.model import ModelModule
from flreid_amd.modules.server import ServerModule


class _Model(ModelModule):
    def model_state(self):
        return {n: p.detach().clone() for n, p in self.net.state_dict().items()}

    def update_model(self, state):
        self.net.load_state_dict(state, strict=False)


def _tiny_model():
    return _Model(torch.nn.Linear(4, 2))


class _Op:
    """Minimal operator stand-in (clients bind their logger onto it)."""
    logger = None


class _Client(ClientModule):
    def train(self, *a, **k): ...
    def train_one_epoch(self, *a, **k): ...
    def inference(self, *a, **k): ...
    def validate(self, *a, **k): ...


def test_client_ckpt_layout_and_collision(tmp_path):
    c = _Client("client-0", _tiny_model(), operator=_Op(),
                ckpt_root=str(tmp_path), custom_attr=42)
    # arbitrary yaml kwargs become attributes (ref:modules/client.py:25-26)
    assert c.custom_attr == 42

    c.save_state("s1", {"x": torch.ones(2)}, cover=False)
    path = os.path.join(str(tmp_path), "client-0", "s1.ckpt")
    assert os.path.exists(path)                 # {root}/{actor}/{name}.ckpt
    with pytest.raises(ValueError):
        c.save_state("s1", {"x": torch.zeros(2)}, cover=False)
    c.save_state("s1", {"x": torch.zeros(2)}, cover=True)   # cover overwrites
    loaded = c.load_state("s1")
    assert torch.equal(loaded["x"], torch.zeros(2))
    assert loaded["x"].device.type == "cpu"     # CPU-mapped client loads
    assert c.state_exists("s1") and not c.state_exists("nope")

    with pytest.raises(ValueError):
        c.load_state("missing-and-no-default")
    assert c.load_state("missing", default_value={"d": 1}) == {"d": 1}


def test_server_register_unregister(tmp_path):
    s = ServerModule("server", _tiny_model(), operator=_Op(),
                     ckpt_root=str(tmp_path))
    s.register_client("c0")
    s.register_client("c1")
    assert set(s.clients) == {"c0", "c1"}
    s.set_client_incremental_state("c0", {"k": 1}) if hasattr(
        s, "set_client_incremental_state") else None
    s.unregister_client("c0")
    assert set(s.clients) == {"c1"}


def test_ckpt_disabled_short_circuits(tmp_path, monkeypatch):
    monkeypatch.setenv("FLREID_DISABLE_CKPT", "1")
    c = _Client("client-0", _tiny_model(), operator=_Op(),
                ckpt_root=str(tmp_path))
    c.save_state("s1", {"x": torch.ones(2)}, cover=False)
    assert not os.path.exists(os.path.join(str(tmp_path), "client-0",
                                           "s1.ckpt"))
    assert not c.state_exists("s1")
    assert c.load_state("s1", default_value={"d": 2}) == {"d": 2}
