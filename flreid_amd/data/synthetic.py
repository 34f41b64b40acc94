"""Synthetic ReID task generation.

The benchmark contract (BASELINE.json) is synthetic data + random-init
weights, so the framework ships a first-class synthetic task source:

  - `SyntheticReIDDataset`: in-memory dataset producing deterministic
    pseudo-images per (task, person, index) without touching disk.  Each
    person id has a fixed random "identity pattern"; samples are the pattern
    plus noise, so ReID metrics (CMC/mAP) are non-trivial.
  - `materialize_task_dir`: writes the reference's on-disk layout
    `task-{client}-{task}/{train,query,gallery}/{person_id}/{i}.pt`
    (ref:datasets/preprocessed_shuffle/README.md) with `.pt` tensors, for
    tests of the directory pipeline.

A `datasets_dir` of the form `synthetic://ids=32,train=4,query=2,gallery=4,
hw=128x64,base=0` makes ReIDTaskPipeline generate tasks on the fly.
"""

from __future__ import annotations

import hashlib
import os
from typing import Callable, Dict, Optional, Tuple

import torch
from torch.utils.data import Dataset


def _seed_from(*parts) -> int:
    h = hashlib.sha256("|".join(str(p) for p in parts).encode()).digest()
    return int.from_bytes(h[:8], "little") % (2 ** 31)


# process-wide cache of materialised synthetic splits (see _materialized)
_SPLIT_CACHE: Dict = {}


def parse_synthetic_dir(datasets_dir: str) -> Optional[Dict]:
    if not datasets_dir.startswith("synthetic:"):
        return None
    spec = datasets_dir.split("//", 1)[-1]
    opts = {"ids": 32, "train": 4, "query": 2, "gallery": 4, "hw": "128x64",
            "base": 0, "idspace": 4096}
    if spec:
        for kv in spec.split(","):
            if not kv:
                continue
            k, v = kv.split("=")
            opts[k] = v if k == "hw" else int(v)
    h, w = (int(x) for x in str(opts["hw"]).split("x"))
    opts["shape"] = (3, h, w)
    return opts


class SyntheticReIDDataset(Dataset):
    """Deterministic synthetic split of one task.

    person ids are `base + [0, n_ids)` offset by a per-task stride so distinct
    tasks hold distinct identities (like the reference's disjoint task
    shards).  Returns (img, person_id, class_index).
    """

    def __init__(self, task_name: str, split: str, n_ids: int,
                 imgs_per_id: int, shape: Tuple[int, int, int] = (3, 128, 64),
                 id_base: int = 0, transform: Callable = None,
                 idspace: int = 4096):
        super().__init__()
        self.task_name = task_name
        self.split = split
        self.n_ids = n_ids
        self.imgs_per_id = imgs_per_id
        self.shape = shape
        self.transform = transform
        # person ids stay inside [id_base, id_base + idspace) so they always
        # fit the classifier head (num_classes >= id_base + idspace)
        slots = max(1, idspace // max(1, n_ids))
        task_stride = (_seed_from(task_name, "ids") % slots) * n_ids
        task_stride = min(task_stride, max(0, idspace - n_ids))
        self.classes = [id_base + task_stride + i for i in range(n_ids)]

    @property
    def person_ids(self):
        return self.classes

    def __len__(self) -> int:
        return self.n_ids * self.imgs_per_id

    def _image(self, person_id: int, index: int) -> torch.Tensor:
        gid = torch.Generator().manual_seed(_seed_from("id-pattern", person_id))
        pattern = torch.rand(self.shape, generator=gid)
        gs = torch.Generator().manual_seed(
            _seed_from(self.task_name, self.split, person_id, index))
        noise = torch.randn(self.shape, generator=gs) * 0.25
        return (pattern + noise).clamp_(-1.0, 2.0)

    def _materialized(self) -> torch.Tensor:
        """Whole-split tensor, generated once per (task, split, geometry) and
        cached process-wide — the task repeats for `sustain_rounds` rounds, so
        regenerating per round is pure host overhead."""
        key = (self.task_name, self.split, self.n_ids, self.imgs_per_id,
               self.shape, tuple(self.classes))
        cached = _SPLIT_CACHE.get(key)
        if cached is None:
            imgs = [self._image(self.classes[i // self.imgs_per_id],
                                i % self.imgs_per_id)
                    for i in range(len(self))]
            cached = torch.stack(imgs)
            if len(_SPLIT_CACHE) > 64:     # bound memory across many tasks
                _SPLIT_CACHE.clear()
            _SPLIT_CACHE[key] = cached
        return cached

    def __getitem__(self, index: int):
        class_index = index // self.imgs_per_id
        person_id = self.classes[class_index]
        img = self._materialized()[index]
        if self.transform is not None:
            img = self.transform(img)
        return img, person_id, class_index


def materialize_task_dir(root: str, client: int, task: int, n_ids: int = 8,
                         train: int = 4, query: int = 2, gallery: int = 3,
                         shape: Tuple[int, int, int] = (3, 64, 32),
                         id_base: int = 0) -> str:
    """Write the on-disk layout the reference documents
    (task-{client}-{task}/{train,query,gallery}/{person_id}/*.pt)."""
    task_name = f"task-{client}-{task}"
    task_dir = os.path.join(root, task_name)
    for split, count in (("train", train), ("query", query), ("gallery", gallery)):
        ds = SyntheticReIDDataset(task_name, split, n_ids, count, shape, id_base)
        for idx in range(len(ds)):
            img, person_id, _ = ds[idx]
            pdir = os.path.join(task_dir, split, str(person_id))
            os.makedirs(pdir, exist_ok=True)
            torch.save(img, os.path.join(pdir, f"{idx % count}.pt"))
    return task_dir
