"""Offline analysis layer (ref:analyse/)."""

import torch

from flreid_amd.analyse.accuracy import accuracy_on_round, accuracy_per_task
from flreid_amd.analyse.forgetting import forgetting_per_client, mean_forgetting

RECORDS = {"data": {
    "client-0": {
        "10": {"task-0-0": {"val_rank_1": 0.8, "val_map": 0.6}},
        "20": {"task-0-0": {"val_rank_1": 0.6, "val_map": 0.5},
               "task-0-1": {"val_rank_1": 1.0, "val_map": 0.9}},
    },
    "client-1": {
        "10": {"task-1-0": {"val_rank_1": 0.4, "val_map": 0.3}},
        "20": {"task-1-0": {"val_rank_1": 0.2, "val_map": 0.1}},
    },
}}


def test_accuracy_on_round():
    curve = accuracy_on_round(RECORDS, "val_rank_1")
    assert curve[10] == (0.8 + 0.4) / 2
    assert abs(curve[20] - (0.6 + 1.0 + 0.2) / 3) < 1e-9


def test_accuracy_per_task():
    per = accuracy_per_task(RECORDS, "val_map")
    assert per["task-0-0"][20] == 0.5


def test_forgetting():
    fpc = forgetting_per_client(RECORDS, "val_rank_1")
    # client-0 task-0-0: peak .8 final .6 -> .2 (task-0-1 single point skipped)
    assert abs(fpc["client-0"] - 0.2) < 1e-9
    assert abs(fpc["client-1"] - 0.2) < 1e-9
    assert abs(mean_forgetting(RECORDS, "val_rank_1") - 0.2) < 1e-9


def test_grad_cam_runs():
    from flreid_amd.analyse.visualize import grad_cam
    from flreid_amd.models import nets
    net = nets["resnet18"](num_classes=8, last_stride=1, neck="bnneck")
    img = torch.randn(1, 3, 64, 32)
    cam = grad_cam(net, net.base.layer4[-1], img)
    assert cam.shape == (64, 32)
    assert float(cam.min()) >= 0.0 and float(cam.max()) <= 1.0


def test_analyse_consumes_real_run_log(tiny_common, tiny_exp_config, tmp_path,
                                       monkeypatch):
    """Schema integration: the analysis layer must read the JSON an actual
    experiment run writes (guards against log-schema drift)."""
    import json
    import os

    from flreid_amd.analyse.accuracy import load_log
    from flreid_amd.parallel.comm import FedContext
    from flreid_amd.runtime.experiment import ExperimentStage

    monkeypatch.chdir(tmp_path)
    stage = ExperimentStage(tiny_common, [tiny_exp_config], ctx=FedContext())
    stage.run_experiment(tiny_exp_config)
    logs = os.listdir(tiny_common["logs_dir"])
    assert len(logs) == 1
    records = load_log(os.path.join(tiny_common["logs_dir"], logs[0]))

    curve = accuracy_on_round(records, "val_rank_1")
    assert curve and all(0.0 <= v <= 1.0 for v in curve.values())
    forg = mean_forgetting(records, "val_rank_1")
    assert isinstance(forg, float)


def test_plots_render(tmp_path):
    """Round-curve and forgetting plots render to PNG (matplotlib is in the
    image; the functions still return None gracefully without it)."""
    import json

    from flreid_amd.analyse.accuracy import plot_accuracy_curves
    from flreid_amd.analyse.forgetting import plot_forgetting

    log_path = tmp_path / "run.json"
    log_path.write_text(json.dumps(RECORDS))

    p1 = plot_accuracy_curves([str(log_path)], "val_rank_1",
                              str(tmp_path / "acc.png"))
    p2 = plot_forgetting([str(log_path)], "val_rank_1",
                         str(tmp_path / "forg.png"))
    import importlib.util
    if importlib.util.find_spec("matplotlib") is None:
        assert p1 is None and p2 is None
    else:
        import os
        assert os.path.getsize(p1) > 0 and os.path.getsize(p2) > 0


def test_per_task_grid_and_merged_plots(tmp_path):
    import json
    import os

    from flreid_amd.analyse.accuracy import plot_merged, plot_per_task_grid

    log_path = tmp_path / "run.json"
    log_path.write_text(json.dumps(RECORDS))
    p1 = plot_per_task_grid(str(log_path), "val_map",
                            str(tmp_path / "grid.png"))
    p2 = plot_merged([str(log_path)], str(tmp_path / "merged.png"))
    import importlib.util
    if importlib.util.find_spec("matplotlib") is None:
        assert p1 is None and p2 is None
    else:
        assert os.path.getsize(p1) > 0 and os.path.getsize(p2) > 0
