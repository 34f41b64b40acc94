"""Per-client ordered task stream (ref:datasets/datasets_pipeline.py:10-93).

Semantics preserved exactly:
  - each task persists `sustain_rounds` calls of `next_task()` before the
    pipeline advances (ref:datasets/datasets_pipeline.py:86-93);
  - `get_task` builds train (augmented, shuffled) / query / gallery loaders
    with `drop_last` only when len % batch == 1 (BatchNorm safety,
    ref:datasets/datasets_pipeline.py:41).

New: a `synthetic://` datasets_dir generates tasks in memory (data/synthetic.py),
which is the benchmark path — no disk, no decode, deterministic.
"""

from __future__ import annotations

import os
from typing import Dict, List

import os

import torch
from torch.utils.data import DataLoader

from flreid_amd.data.augment import DeviceAugmentLoader, augmentations
from flreid_amd.data.loader import ReIDImageDataset
from flreid_amd.data.synthetic import SyntheticReIDDataset, parse_synthetic_dir


class ReIDTaskPipeline:
    def __init__(self, task_list: List[str], task_opts: Dict, datasets_dir: str):
        self.task_list = task_list
        self.task_opts = task_opts
        self.datasets_dir = datasets_dir
        self.synthetic_opts = parse_synthetic_dir(datasets_dir)
        self.current_task_idx = -1
        self.task_round_rest = [task_opts["sustain_rounds"] for _ in task_list]

    def reach_final_task(self) -> bool:
        return self.current_task_idx + 1 == len(self.task_list)

    def _make_loader(self, dataset, shuffle: bool) -> DataLoader:
        lo = self.task_opts.get("loader_opts", {})
        batch_size = lo.get("batch_size", 32)
        kwargs = dict(
            dataset=dataset,
            shuffle=shuffle,
            drop_last=len(dataset) % batch_size == 1,
            batch_size=batch_size,
            num_workers=lo.get("num_workers", 0),
            pin_memory=lo.get("pin_memory", False),
        )
        if lo.get("num_workers", 0):
            kwargs["persistent_workers"] = lo.get("persistent_workers", False)
            if lo.get("multiprocessing_context"):
                kwargs["multiprocessing_context"] = lo["multiprocessing_context"]
        return DataLoader(**kwargs)

    @staticmethod
    def _gpu_augment() -> bool:
        return (os.environ.get("FLREID_GPU_AUGMENT", "0") == "1"
                and torch.cuda.is_available())

    def _dataset(self, task: str, split: str, transform):
        if self.synthetic_opts is not None:
            so = self.synthetic_opts
            per_id = {"train": so["train"], "query": so["query"], "gallery": so["gallery"]}[split]
            return SyntheticReIDDataset(task, split, so["ids"], per_id,
                                        so["shape"], so["base"], transform,
                                        idspace=so["idspace"])
        return ReIDImageDataset(os.path.join(self.datasets_dir, task, split), transform)

    def get_task(self, idx: int = -1) -> Dict:
        task = self.task_list[idx]
        ao = self.task_opts["augment_opts"]
        tr_aug = augmentations[ao["level"]](size=ao["img_size"], mean=ao["norm_mean"],
                                            std=ao["norm_std"])
        no_aug = augmentations["none"](size=ao["img_size"], mean=ao["norm_mean"],
                                       std=ao["norm_std"])
        if self._gpu_augment():
            dev = "cuda"
            return {
                "task_name": task,
                "tr_epochs": self.task_opts["train_epochs"],
                "tr_loader": DeviceAugmentLoader(
                    self._make_loader(self._dataset(task, "train", None), True),
                    tr_aug, dev),
                "query_loader": DeviceAugmentLoader(
                    self._make_loader(self._dataset(task, "query", None), False),
                    no_aug, dev),
                "gallery_loaders": DeviceAugmentLoader(
                    self._make_loader(self._dataset(task, "gallery", None), False),
                    no_aug, dev),
            }
        return {
            "task_name": task,
            "tr_epochs": self.task_opts["train_epochs"],
            "tr_loader": self._make_loader(self._dataset(task, "train", tr_aug), True),
            "query_loader": self._make_loader(self._dataset(task, "query", no_aug), False),
            "gallery_loaders": self._make_loader(self._dataset(task, "gallery", no_aug), False),
        }

    def current_task(self) -> Dict:
        if self.current_task_idx == -1:
            self.current_task_idx = 0
        return self.get_task(self.current_task_idx)

    def next_task(self) -> Dict:
        if not self.reach_final_task():
            if self.current_task_idx != -1 and self.task_round_rest[self.current_task_idx]:
                self.task_round_rest[self.current_task_idx] -= 1
            else:
                self.current_task_idx += 1
                self.task_round_rest[self.current_task_idx] -= 1
        return self.current_task()
