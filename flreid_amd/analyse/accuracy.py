"""Offline accuracy aggregation over experiment JSON logs
(ref:analyse/accuracy.py).

The compute layer (per-round averages across clients/tasks) is dependency-
free; plotting requires matplotlib and degrades gracefully without it.
"""

from __future__ import annotations

import json
from typing import Dict, List, Optional, Sequence


def load_log(path: str) -> Dict:
    with open(path) as f:
        return json.load(f)


def accuracy_on_round(records: Dict, metric: str = "val_rank_1",
                      clients: Optional[Sequence[str]] = None) -> Dict[int, float]:
    """round -> mean(metric) over every (client, task) entry that logged it
    (ref:analyse/accuracy.py:10-29)."""
    data = records.get("data", records)
    out: Dict[int, List[float]] = {}
    for cname, rounds in data.items():
        if clients is not None and cname not in clients:
            continue
        for rnd, tasks in rounds.items():
            for _task, metrics in tasks.items():
                if isinstance(metrics, dict) and metric in metrics:
                    out.setdefault(int(rnd), []).append(float(metrics[metric]))
    return {r: sum(v) / len(v) for r, v in sorted(out.items())}


def accuracy_per_task(records: Dict, metric: str = "val_map") -> Dict[str, Dict[int, float]]:
    """task -> {round -> mean metric over clients} (per-task-stage curves,
    ref:analyse/accuracy.py:138-215)."""
    data = records.get("data", records)
    out: Dict[str, Dict[int, List[float]]] = {}
    for _cname, rounds in data.items():
        for rnd, tasks in rounds.items():
            for task, metrics in tasks.items():
                if isinstance(metrics, dict) and metric in metrics:
                    out.setdefault(task, {}).setdefault(int(rnd), []).append(
                        float(metrics[metric]))
    return {t: {r: sum(v) / len(v) for r, v in sorted(rs.items())}
            for t, rs in out.items()}


def plot_accuracy_curves(log_paths: Sequence[str], metric: str = "val_rank_1",
                         out_path: str = "accuracy.png") -> Optional[str]:
    """Round curves for one or many jobs (ref:analyse/accuracy.py:32-135);
    returns the written path or None when matplotlib is unavailable."""
    try:
        import matplotlib
        matplotlib.use("Agg")
        import matplotlib.pyplot as plt
    except ImportError:
        return None
    fig, ax = plt.subplots(figsize=(6, 4))
    for path in log_paths:
        curve = accuracy_on_round(load_log(path), metric)
        ax.plot(list(curve.keys()), list(curve.values()), label=path)
    ax.set_xlabel("communication round")
    ax.set_ylabel(metric)
    ax.legend(fontsize=6)
    fig.tight_layout()
    fig.savefig(out_path, dpi=120)
    return out_path


def plot_per_task_grid(log_path: str, metric: str = "val_map",
                       out_path: str = "per_task.png",
                       cols: int = 3) -> Optional[str]:
    """Per-task-stage subplot grid — one panel per task, metric over rounds
    (ref:analyse/accuracy.py:138-215).  None without matplotlib."""
    try:
        import math

        import matplotlib
        matplotlib.use("Agg")
        import matplotlib.pyplot as plt
    except ImportError:
        return None
    per = accuracy_per_task(load_log(log_path), metric)
    if not per:
        return None
    n = len(per)
    rows = math.ceil(n / cols)
    fig, axes = plt.subplots(rows, cols, figsize=(3.2 * cols, 2.4 * rows),
                             squeeze=False)
    for i, (task, curve) in enumerate(sorted(per.items())):
        ax = axes[i // cols][i % cols]
        ax.plot(list(curve.keys()), list(curve.values()), marker=".")
        ax.set_title(task, fontsize=8)
        ax.tick_params(labelsize=6)
    for j in range(n, rows * cols):
        axes[j // cols][j % cols].axis("off")
    fig.supxlabel("communication round", fontsize=8)
    fig.supylabel(metric, fontsize=8)
    fig.tight_layout()
    fig.savefig(out_path, dpi=120)
    return out_path


def plot_merged(log_paths: Sequence[str], out_path: str = "merged.png",
                rank_metric: str = "val_rank_1",
                map_metric: str = "val_map") -> Optional[str]:
    """Merged rank-1 + mAP round curves on twin axes
    (ref:analyse/accuracy.py:218-295).  None without matplotlib."""
    try:
        import matplotlib
        matplotlib.use("Agg")
        import matplotlib.pyplot as plt
    except ImportError:
        return None
    fig, ax = plt.subplots(figsize=(6.5, 4))
    ax2 = ax.twinx()
    for path in log_paths:
        records = load_log(path)
        r1 = accuracy_on_round(records, rank_metric)
        mp = accuracy_on_round(records, map_metric)
        ax.plot(list(r1.keys()), list(r1.values()), marker=".",
                label=f"{path} rank-1")
        ax2.plot(list(mp.keys()), list(mp.values()), marker="x", ls="--",
                 label=f"{path} mAP")
    ax.set_xlabel("communication round")
    ax.set_ylabel(rank_metric)
    ax2.set_ylabel(map_metric)
    h1, l1 = ax.get_legend_handles_labels()
    h2, l2 = ax2.get_legend_handles_labels()
    ax.legend(h1 + h2, l1 + l2, fontsize=6)
    fig.tight_layout()
    fig.savefig(out_path, dpi=120)
    return out_path
