"""Model / optimizer / scheduler registries (ref:models/__init__.py:6-25)."""

import torch

from flreid_amd.models.resnet import (
    resnet18, resnet34, resnet50, resnet101, resnet152,
)


def _swin_factory(name):
    def make(**kwargs):
        from flreid_amd.models.swin import swin_reid
        return swin_reid(name, **kwargs)
    return make


nets = {
    "resnet18": resnet18,
    "resnet34": resnet34,
    "resnet50": resnet50,
    "resnet101": resnet101,
    "resnet152": resnet152,
    "swin_transformer_tiny": _swin_factory("tiny"),
    "swin_transformer_small": _swin_factory("small"),
    "swin_transformer_base": _swin_factory("base"),
    "swin_transformer_large": _swin_factory("large"),
}

optimizers = {
    "adam": torch.optim.Adam,
    "sgd": torch.optim.SGD,
}

schedulers = {
    "step_lr": torch.optim.lr_scheduler.StepLR,
}
