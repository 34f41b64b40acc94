"""Compute-precision policy.

MI355X path: bf16 autocast (MFMA-friendly) with fp32 master weights and fp32
loss/penalty accumulation — no GradScaler needed for bf16.  CPU path: fp32.
The benchmark contract (BASELINE.json) quotes bf16 compute.
"""

from __future__ import annotations

import contextlib
import os

import torch

_BF16 = os.environ.get("FLREID_BF16", "1") != "0"


def set_bf16(enabled: bool) -> None:
    global _BF16
    _BF16 = enabled


def bf16_enabled() -> bool:
    return _BF16 and torch.cuda.is_available()


def autocast(device: torch.device | str = None):
    dev = str(device) if device is not None else ("cuda" if torch.cuda.is_available() else "cpu")
    if dev.startswith("cuda") and _BF16:
        return torch.autocast(device_type="cuda", dtype=torch.bfloat16)
    return contextlib.nullcontext()
