"""Forgetting analysis (ref:analyse/forgetting.py:8-41).

forgetting(task) = peak metric over rounds − value at later rounds, averaged;
aggregated per client and across clients.
"""

from __future__ import annotations

from typing import Dict, Optional, Sequence


def forgetting_per_client(records: Dict, metric: str = "val_rank_1",
                          ) -> Dict[str, float]:
    """client -> mean over tasks of (peak - final) metric."""
    data = records.get("data", records)
    out: Dict[str, float] = {}
    for cname, rounds in data.items():
        task_curves: Dict[str, Dict[int, float]] = {}
        for rnd, tasks in rounds.items():
            for task, metrics in tasks.items():
                if isinstance(metrics, dict) and metric in metrics:
                    task_curves.setdefault(task, {})[int(rnd)] = float(metrics[metric])
        drops = []
        for _task, curve in task_curves.items():
            if len(curve) < 2:
                continue
            values = [curve[r] for r in sorted(curve)]
            peak = max(values)
            drops.append(peak - values[-1])
        if drops:
            out[cname] = sum(drops) / len(drops)
    return out


def mean_forgetting(records: Dict, metric: str = "val_rank_1") -> Optional[float]:
    per_client = forgetting_per_client(records, metric)
    if not per_client:
        return None
    return sum(per_client.values()) / len(per_client)


def plot_forgetting(log_paths: Sequence[str], metric: str = "val_rank_1",
                    out_path: str = "forgetting.png") -> Optional[str]:
    """Bar plots per client (ref:analyse/forgetting.py:44-157); None without
    matplotlib."""
    try:
        import matplotlib
        matplotlib.use("Agg")
        import matplotlib.pyplot as plt
    except ImportError:
        return None
    from flreid_amd.analyse.accuracy import load_log
    fig, ax = plt.subplots(figsize=(6, 4))
    for path in log_paths:
        fpc = forgetting_per_client(load_log(path), metric)
        ax.bar(list(fpc.keys()), list(fpc.values()), alpha=0.6, label=path)
    ax.set_ylabel(f"forgetting ({metric})")
    ax.legend(fontsize=6)
    fig.tight_layout()
    fig.savefig(out_path, dpi=120)
    return out_path
