"""YAML config surface (ref:main.py:12-22, ref:configs/common.yaml).

Exact merge semantics of the reference CLI:
  - `common.yaml` provides run-level keys (`datasets_dir`, `checkpoints_dir`,
    `logs_dir`, `parallel`, `device` list) and a `defaults` block;
  - each experiment yaml is `dict(defaults)` then `.update(exp_yaml)` —
    a SHALLOW top-level merge where the experiment yaml wins
    (ref:main.py:20-21);
  - a scalar `device` is normalised to a one-element list (ref:main.py:13-15).
"""

from __future__ import annotations

import copy
import os
from typing import Any, Dict, List, Sequence, Union

import yaml

DEFAULT_COMMON = {
    "datasets_dir": "./datasets/preprocessed_shuffle/",
    "checkpoints_dir": "./ckpts/",
    "logs_dir": "./logs/",
    "parallel": 1,
    "device": ["cuda:0"],
    "defaults": {},
}


def load_yaml(path: str) -> Dict[str, Any]:
    with open(path, "r") as f:
        return yaml.safe_load(f) or {}


def load_common(path: str = "./configs/common.yaml") -> Dict[str, Any]:
    cfg = dict(DEFAULT_COMMON)
    if os.path.exists(path):
        cfg.update(load_yaml(path))
    if not isinstance(cfg.get("device"), list):
        cfg["device"] = [cfg["device"]]
    cfg.setdefault("defaults", {})
    return cfg


def merge_experiment(common: Dict[str, Any], exp: Dict[str, Any]) -> Dict[str, Any]:
    """Shallow top-level merge: defaults under the experiment, exp wins."""
    merged = copy.deepcopy(dict(common.get("defaults", {})))
    merged.update(copy.deepcopy(exp))
    return merged


def load_experiments(common: Dict[str, Any],
                     paths: Union[str, Sequence[str]]) -> List[Dict[str, Any]]:
    if isinstance(paths, str):
        paths = [paths]
    return [merge_experiment(common, load_yaml(p)) for p in paths]
