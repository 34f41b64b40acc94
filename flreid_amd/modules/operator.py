"""Operator contract (ref:modules/operator.py:12-52).

An Operator owns the optimisation machinery for one actor: the criterion
list, optimizer and LR scheduler, plus the four invoke_* loops
(train / predict / valid / inference) that methods implement.
"""

from __future__ import annotations

from typing import Any, List

from torch.utils.data import DataLoader

from flreid_amd.tools.logger import Logger


class OperatorModule:
    def __init__(self, optimizer, criterion, scheduler=None, logger=None, **kwargs):
        self.logger = logger if logger is not None else Logger("operator")
        self.criterion: List = criterion if isinstance(criterion, (list, tuple)) else [criterion]
        self.optimizer = optimizer
        self.scheduler = scheduler
        self.args = kwargs

    @staticmethod
    def iter_dataloader(*dataloaders: DataLoader):
        if len(dataloaders) == 1 and isinstance(dataloaders[0], list):
            dataloaders = dataloaders[0]
        for dataloader in dataloaders:
            yield from dataloader

    # method-specific loops
    def invoke_train(self, model, dataloader: DataLoader, **kwargs) -> Any:
        raise NotImplementedError

    def _invoke_train(self, model, data: Any, target: Any, **kwargs) -> Any:
        raise NotImplementedError

    def invoke_predict(self, model, dataloader: DataLoader, **kwargs) -> Any:
        raise NotImplementedError

    def _invoke_predict(self, model, data: Any, target: Any, **kwargs) -> Any:
        raise NotImplementedError

    def invoke_valid(self, model, dataloader: DataLoader, **kwargs) -> Any:
        raise NotImplementedError

    def _invoke_valid(self, model, data: Any, target: Any, **kwargs) -> Any:
        raise NotImplementedError

    def invoke_inference(self, model, dataloader: DataLoader, **kwargs) -> Any:
        raise NotImplementedError

    def _invoke_inference(self, model, data: Any, **kwargs) -> Any:
        raise NotImplementedError
