"""Config surface tests (ref:main.py:12-22 merge semantics)."""

import os

from flreid_amd.config import load_common, load_experiments, merge_experiment


def test_shallow_merge_exp_wins():
    common = {"defaults": {"a": 1, "model_opts": {"name": "resnet18", "x": 2}}}
    exp = {"model_opts": {"name": "resnet50"}, "b": 3}
    merged = merge_experiment(common, exp)
    assert merged["a"] == 1
    assert merged["b"] == 3
    # shallow: the whole model_opts block is replaced, not deep-merged
    assert merged["model_opts"] == {"name": "resnet50"}


def test_device_normalised_to_list(tmp_path):
    p = tmp_path / "common.yaml"
    p.write_text("device: cpu\ndefaults:\n  random_seed: 5\n")
    cfg = load_common(str(p))
    assert cfg["device"] == ["cpu"]
    assert cfg["defaults"]["random_seed"] == 5


def test_load_experiments(tmp_path):
    c = tmp_path / "common.yaml"
    c.write_text("defaults:\n  random_seed: 9\n  exp_opts:\n    comm_rounds: 3\n")
    e = tmp_path / "exp.yaml"
    e.write_text("exp_name: t\nexp_method: fedavg\n")
    common = load_common(str(c))
    exps = load_experiments(common, [str(e)])
    assert exps[0]["exp_name"] == "t"
    assert exps[0]["random_seed"] == 9
    assert exps[0]["exp_opts"]["comm_rounds"] == 3


def test_reference_configs_parse():
    """Every shipped yaml under configs/ must parse + merge."""
    root = os.path.join(os.path.dirname(os.path.dirname(__file__)), "configs")
    if not os.path.isdir(root):
        return
    common = load_common(os.path.join(root, "common.yaml"))
    count = 0
    for dirpath, _dirs, files in os.walk(root):
        for f in files:
            if f.endswith(".yaml") and not f.startswith("common"):
                merged = load_experiments(common, os.path.join(dirpath, f))[0]
                assert "exp_name" in merged and "exp_method" in merged
                count += 1
    assert count >= 1
