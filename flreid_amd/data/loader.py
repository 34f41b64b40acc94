"""ReID datasets (capability parity with ref:datasets/datasets_loader.py).

`ReIDImageDataset` accepts:
  - a directory path: subdirectories named by person id, each holding image
    tensors saved as `.pt` files (the tensor-native on-disk format this
    framework uses instead of JPEG folders — see data/synthetic.py for the
    generator that materialises the reference's
    `task-{client}-{task}/{train,query,gallery}/{person_id}/` layout);
  - a dict `{person_id: [(img, class_id), ...]}` for in-memory exemplars /
    prototypes (ref:datasets/datasets_loader.py:20-26).

`__getitem__` returns `(img, person_id, class_index)`
(ref:datasets/datasets_loader.py:34-40).
"""

from __future__ import annotations

import os
from typing import Any, Callable, Dict, Union

import torch
from torch.utils.data import Dataset

from flreid_amd.data.augment import augmentation_none


class ReIDImageDataset(Dataset):
    def __init__(self, source: Union[str, Dict], transform: Callable = None):
        super().__init__()
        self.reload_source(source, transform or augmentation_none())

    def reload_source(self, source, transform: Callable = None) -> None:
        self.transform = transform
        if isinstance(source, str):
            self.items = []       # list of (path_or_tensor, class_index)
            class_dirs = sorted(
                (d for d in os.listdir(source)
                 if os.path.isdir(os.path.join(source, d))),
                key=lambda d: d,
            )
            # class_index -> person_id, in sorted-name order like ImageFolder
            self.classes = [int(d) for d in class_dirs]
            for class_index, d in enumerate(class_dirs):
                cdir = os.path.join(source, d)
                for fname in sorted(os.listdir(cdir)):
                    if fname.endswith(".pt"):
                        self.items.append((os.path.join(cdir, fname), class_index))
        elif isinstance(source, dict):
            self.items = []
            self.classes = {}
            for person_id, protos in source.items():
                for img, class_id in protos:
                    self.items.append((img, class_id))
                    self.classes[class_id] = person_id
        else:
            raise ValueError("source must be a directory path or an in-memory dict")

    @property
    def person_ids(self):
        return self.classes

    def __getitem__(self, index: int) -> Any:
        data, class_index = self.items[index]
        if isinstance(data, str):
            data = torch.load(data, map_location="cpu", weights_only=False)
            if self.transform is not None:
                data = self.transform(data)
        elif not isinstance(data, torch.Tensor):
            data = torch.as_tensor(data, dtype=torch.float32)
        class_index = int(class_index)
        person_id = int(self.classes[class_index])
        return data, person_id, class_index

    def __len__(self) -> int:
        return len(self.items)
