"""Weight initialisation for ReID heads (ref:tools/winit.py:8-28).

Same scheme as the reid-strong-baseline heads: kaiming for linear/conv, unit
BN, and a small-std normal classifier init.
"""

from __future__ import annotations

from torch import nn


def weights_init_kaiming(m: nn.Module) -> None:
    name = m.__class__.__name__
    if "Linear" in name:
        nn.init.kaiming_normal_(m.weight, a=0, mode="fan_out")
        if m.bias is not None:
            nn.init.constant_(m.bias, 0.0)
    elif "Conv" in name:
        nn.init.kaiming_normal_(m.weight, a=0, mode="fan_in")
        if m.bias is not None:
            nn.init.constant_(m.bias, 0.0)
    elif "BatchNorm" in name:
        if m.affine:
            nn.init.constant_(m.weight, 1.0)
            nn.init.constant_(m.bias, 0.0)


def weights_init_classifier(m: nn.Module) -> None:
    name = m.__class__.__name__
    if "Linear" in name:
        nn.init.normal_(m.weight, std=0.001)
        if m.bias is not None:
            nn.init.constant_(m.bias, 0.0)
