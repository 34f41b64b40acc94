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
