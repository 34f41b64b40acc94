"""FedSTIL additive-decomposition layers (ref:methods/fedstil.py:24-225).

θ = global_weight_atten ⊙ global_weight + adaptive_weight, with the attention
vector broadcast over the LAST weight dimension
(ref:methods/fedstil.py:66-67,84-92).  Only `adaptive_weight` (and bias)
train; `global_weight` is the federated state and `global_weight_atten` is a
fixed mixing vector (it becomes a learnable stacked attention in the
fedstil-atten variant — ref:methods/fedstil_atten.py).

`init_training_weights()` (no args) is the dispatch-time re-initialisation:
atten resets to `atten_default` and adaptive_weight to (1−atten)·W_glob, so
the composed weight right after dispatch equals the dispatched global weight
(ref:methods/fedstil.py:53-82, applied at ref:methods/fedstil.py:879-911).
`initial_*` snapshots anchor the round's L1 drift regulariser
(ref:methods/fedstil.py:639-644).

On GPU the composition runs through the fused HIP compose kernel
(flreid_amd.ops.adaptive_compose) so θ is produced in one pass; the backward
for adaptive_weight is identity and atten (when learnable) reduces over all
dims but the last.
"""

from __future__ import annotations

import os
from typing import Optional, Tuple

import torch
import torch.nn.functional as F
from torch import nn
from torch.nn import Parameter

from flreid_amd import ops


class AdaptiveBase(nn.Module):
    """Holds the decomposed parameter set; subclasses implement forward."""

    def __init__(self, global_weight: torch.Tensor,
                 global_weight_atten: Optional[torch.Tensor] = None,
                 adaptive_weight: Optional[torch.Tensor] = None,
                 adaptive_bias: Optional[torch.Tensor] = None,
                 atten_default: float = 0.80, atten_trainable: bool = False,
                 **kwargs):
        super().__init__()
        self.atten_default = atten_default
        self.atten_trainable = atten_trainable
        self.global_weight = Parameter(torch.empty(0), requires_grad=False)
        self.global_weight_atten = Parameter(torch.empty(0), requires_grad=False)
        self.adaptive_weight = Parameter(torch.empty(0))
        self.adaptive_bias = Parameter(torch.empty(0)) if adaptive_bias is not None else None
        self.initial_global_weight_atten = Parameter(torch.empty(0), requires_grad=False)
        self.initial_adaptive_weight = Parameter(torch.empty(0), requires_grad=False)
        self.init_training_weights(global_weight, global_weight_atten,
                                   adaptive_weight, adaptive_bias)

    @staticmethod
    def _assign(param: Parameter, value: torch.Tensor) -> None:
        """In-place when shapes match (keeps optimizer state and captured
        hipGraph pointers valid across per-round re-inits); rebind otherwise
        (clearing any stale gradient of the old shape)."""
        if param.data.shape == value.shape:
            param.data.copy_(value)
        else:
            param.data = value.detach().clone().to(param.device)
            param.grad = None

    @torch.no_grad()
    def init_training_weights(self, global_weight=None, global_weight_atten=None,
                              adaptive_weight=None, adaptive_bias=None) -> None:
        if global_weight is None:
            global_weight = self.global_weight.data
        self._assign(self.global_weight, global_weight.detach())
        self.global_weight.requires_grad = False

        if global_weight_atten is None:
            global_weight_atten = torch.ones(
                self.global_weight.data.shape[-1],
                device=self.global_weight.device) * self.atten_default
        self._assign(self.global_weight_atten, global_weight_atten.detach())
        self._assign(self.initial_global_weight_atten, global_weight_atten.detach())
        self.global_weight_atten.requires_grad = self.atten_trainable

        if adaptive_weight is None:
            adaptive_weight = (1.0 - self.global_weight_atten.data) * self.global_weight.data
        self._assign(self.adaptive_weight, adaptive_weight.detach())
        self._assign(self.initial_adaptive_weight, adaptive_weight.detach())
        self.adaptive_weight.requires_grad = True

        if self.adaptive_bias is not None and adaptive_bias is not None:
            self._assign(self.adaptive_bias, adaptive_bias.detach())
            self.adaptive_bias.requires_grad = True

    def composed_weight(self) -> torch.Tensor:
        return ops.adaptive_compose(self.global_weight, self.global_weight_atten,
                                    self.adaptive_weight)

    def drift_pairs(self):
        """(current, round-start) pairs for the L1 drift regulariser."""
        return [(self.global_weight_atten, self.initial_global_weight_atten),
                (self.adaptive_weight, self.initial_adaptive_weight)]


class AdaptiveLinear(AdaptiveBase):
    def forward(self, data: torch.Tensor) -> torch.Tensor:
        if data.dim() == 2:
            # fused MFMA GEMM with compose-in-prologue (K2); falls back to
            # compose + F.linear off-GPU or when atten trains
            return ops.adaptive_linear(data, self.global_weight,
                                       self.global_weight_atten,
                                       self.adaptive_weight,
                                       self.adaptive_bias)
        return F.linear(data, self.composed_weight(), self.adaptive_bias)


def _is_one(v) -> bool:
    return v == 1 or v == (1, 1) or v == [1, 1]


def _is_zero(v) -> bool:
    return v == 0 or v == (0, 0) or v == [0, 0]


class AdaptiveConv2d(AdaptiveBase):
    def __init__(self, global_weight, stride=1, padding=0, **kwargs):
        super().__init__(global_weight, **kwargs)
        self.stride = stride
        self.padding = padding

    def forward(self, data: torch.Tensor) -> torch.Tensor:
        gw = self.global_weight
        if (data.is_cuda and gw.dim() == 4 and gw.shape[2] == 3
                and gw.shape[3] == 3 and _is_one(self.stride)
                and _is_one(self.padding) and self.adaptive_bias is None
                and not gw.requires_grad
                and not self.global_weight_atten.requires_grad):
            # hand-written K1 conv (conv3x3_img.hip): θ composed to bf16 in
            # the weight fetch, halo-staged MFMA fwd + dgrad/wgrad kernels
            y = ops.conv3x3_try(data, gw, self.global_weight_atten,
                                self.adaptive_weight)
            if y is not None:
                return y
        if (data.is_cuda and gw.dim() == 4 and gw.shape[2] == 1
                and gw.shape[3] == 1 and _is_one(self.stride)
                and _is_zero(self.padding)):
            # pointwise conv == GEMM over N·H·W; on channels-last input the
            # [NHW, C] view is free.  DEFAULT route (K2): θ composed
            # straight to bf16 (compose2 — one pass, no fp32 θ, no autocast
            # cast) + bf16 hipBLASLt GEMMs fwd/bwd, aw's grad cast once to
            # fp32.  Disable with FLREID_NO_FUSED_1X1=1.
            b, c, h, w = data.shape
            n_out = gw.shape[0]
            xv = data.permute(0, 2, 3, 1).reshape(-1, c)
            y = None
            if not gw.requires_grad and not self.global_weight_atten.requires_grad:
                aw2d = self.adaptive_weight.view(n_out, c) \
                    if self.adaptive_weight.numel() else None
                y = ops.adaptive_linear_1x1(xv, gw.view(n_out, c),
                                            self.global_weight_atten, aw2d,
                                            self.adaptive_bias)
            if y is None:
                theta = self.composed_weight().view(n_out, c)
                y = F.linear(xv, theta, self.adaptive_bias)
            return y.view(b, h, w, -1).permute(0, 3, 1, 2)
        return F.conv2d(data, self.composed_weight(), self.adaptive_bias,
                        stride=self.stride, padding=self.padding)


class AdaptiveBatchNorm2d(AdaptiveBase):
    """Completeness parity with ref:methods/fedstil.py:131-198 (the reference
    ships this but keeps BatchNorm out of its transform LUT)."""

    def __init__(self, global_weight, running_mean=None, running_var=None,
                 num_batches_tracked=None, track_running_stats=False,
                 momentum=0.1, eps=1e-5, **kwargs):
        super().__init__(global_weight, **kwargs)
        self.register_buffer("running_mean", running_mean)
        self.register_buffer("running_var", running_var)
        self.register_buffer("num_batches_tracked", num_batches_tracked)
        self.track_running_stats = track_running_stats
        self.momentum = momentum
        self.eps = eps

    def forward(self, data: torch.Tensor) -> torch.Tensor:
        training = self.training or (self.running_mean is None and self.running_var is None)
        use_stats = not self.training or self.track_running_stats
        return F.batch_norm(
            data,
            self.running_mean if use_stats else None,
            self.running_var if use_stats else None,
            self.composed_weight(), self.adaptive_bias,
            training, self.momentum or 0.0, self.eps)


class AdaptiveLayerNorm(AdaptiveBase):
    def __init__(self, global_weight, normalized_shape: Tuple[int, ...] = None,
                 eps: float = 1e-5, **kwargs):
        super().__init__(global_weight, **kwargs)
        self.normalized_shape = normalized_shape or tuple(global_weight.shape)
        self.eps = eps

    def forward(self, data: torch.Tensor) -> torch.Tensor:
        return F.layer_norm(data, self.normalized_shape, self.composed_weight(),
                            self.adaptive_bias, self.eps)


ADAPTIVE_TYPES = (AdaptiveLinear, AdaptiveConv2d, AdaptiveBatchNorm2d, AdaptiveLayerNorm)


def convert_to_adaptive(net: nn.Module, atten_default: float = 0.80,
                        atten_trainable: bool = False) -> int:
    """Rewrite every fully-trainable Linear/Conv2d leaf into its adaptive
    counterpart (ref:methods/fedstil.py:290-347; BN/LN stay untouched like the
    reference's LUT).  Returns the number of converted leaves."""
    count = 0
    for name, module in list(net.named_modules()):
        if isinstance(module, (nn.Linear, nn.Conv2d)):
            if not all(p.requires_grad for p in module.parameters()):
                continue
            if isinstance(module, nn.Linear):
                new = AdaptiveLinear(
                    global_weight=module.weight, adaptive_bias=module.bias,
                    atten_default=atten_default, atten_trainable=atten_trainable)
            else:
                new = AdaptiveConv2d(
                    global_weight=module.weight, adaptive_bias=module.bias,
                    atten_default=atten_default, atten_trainable=atten_trainable,
                    stride=module.stride, padding=module.padding)
            parent = net
            parts = name.split(".")
            for p in parts[:-1]:
                parent = getattr(parent, p)
            setattr(parent, parts[-1], new)
            count += 1
    return count


def adaptive_leaves(net: nn.Module):
    """[(qualname, module)] of adaptive leaves, in module order."""
    return [(n, m) for n, m in net.named_modules() if isinstance(m, ADAPTIVE_TYPES)]


def non_adaptive_leaves(net: nn.Module):
    """Leaves that are NOT adaptive (the 'pre-trained' frozen set,
    ref:methods/fedstil.py:421-433)."""
    out = []
    for n, m in net.named_modules():
        if len(list(m.children())) == 0 and not isinstance(m, ADAPTIVE_TYPES) and n:
            out.append((n, m))
    return out


# ---------------------------------------------------------------------------
# FedSTIL-atten stacked variant (ref:methods/fedstil_atten.py)
# ---------------------------------------------------------------------------

class StackedAttenBase(nn.Module):
    """θ = Σ_k atten_k · W_glob[..., k] + W_adapt.

    The server CONCATENATES client uploads along a new last dim instead of
    averaging, and the per-slot attention is LEARNABLE; `adaptive_weight`
    persists across dispatch re-inits while `atten` is re-initialised to
    `atten_default` sized to the current stack depth
    (ref:methods/fedstil_atten.py:46-90)."""

    def __init__(self, global_weight: torch.Tensor,
                 adaptive_bias: Optional[torch.Tensor] = None,
                 atten_default: float = 0.0, **kwargs):
        super().__init__()
        self.atten_default = atten_default
        self.atten_trainable = True
        self.global_weight = Parameter(torch.empty(0), requires_grad=False)
        self.global_weight_atten = Parameter(torch.empty(0))
        self.adaptive_weight = Parameter(torch.empty(0))
        self.adaptive_bias = Parameter(torch.empty(0)) if adaptive_bias is not None else None
        self.initial_global_weight_atten = Parameter(torch.empty(0), requires_grad=False)
        self.initial_adaptive_weight = Parameter(torch.empty(0), requires_grad=False)
        # the constructor always receives a RAW module weight: add the stack
        # dim unconditionally (re-inits via init_training_weights pass
        # already-stacked tensors)
        self.init_training_weights(global_weight.unsqueeze(-1),
                                   adaptive_bias=adaptive_bias)

    _assign = staticmethod(AdaptiveBase._assign)

    @torch.no_grad()
    def init_training_weights(self, global_weight=None, global_weight_atten=None,
                              adaptive_weight=None, adaptive_bias=None) -> None:
        if global_weight is None:
            global_weight = self.global_weight.data
        self._assign(self.global_weight, global_weight.detach())
        self.global_weight.requires_grad = False

        k = self.global_weight.shape[-1]
        if global_weight_atten is None:
            global_weight_atten = torch.ones(k, device=self.global_weight.device) \
                                  * self.atten_default
        self._assign(self.global_weight_atten, global_weight_atten.detach())
        self._assign(self.initial_global_weight_atten, global_weight_atten.detach())
        self.global_weight_atten.requires_grad = True

        if self.adaptive_weight.numel() == 0:   # created once, then persists
            if adaptive_weight is None:
                adaptive_weight = ((1.0 - self.global_weight_atten.data)
                                   * self.global_weight.data).sum(dim=-1)
            self._assign(self.adaptive_weight, adaptive_weight.detach())
        elif adaptive_weight is not None:
            self._assign(self.adaptive_weight, adaptive_weight.detach())
        self.adaptive_weight.requires_grad = True
        self._assign(self.initial_adaptive_weight, self.adaptive_weight.detach())

        if self.adaptive_bias is not None and adaptive_bias is not None:
            self._assign(self.adaptive_bias, adaptive_bias.detach())
            self.adaptive_bias.requires_grad = True

    def composed_weight(self) -> torch.Tensor:
        return (self.global_weight_atten * self.global_weight).sum(dim=-1) \
               + self.adaptive_weight

    def drift_pairs(self):
        return [(self.global_weight_atten, self.initial_global_weight_atten),
                (self.adaptive_weight, self.initial_adaptive_weight)]


class StackedAttenLinear(StackedAttenBase):
    def forward(self, data: torch.Tensor) -> torch.Tensor:
        return F.linear(data, self.composed_weight(), self.adaptive_bias)


class StackedAttenConv2d(StackedAttenBase):
    def __init__(self, global_weight, stride=1, padding=0, **kwargs):
        super().__init__(global_weight, **kwargs)
        self.stride = stride
        self.padding = padding

    def forward(self, data: torch.Tensor) -> torch.Tensor:
        return F.conv2d(data, self.composed_weight(), self.adaptive_bias,
                        stride=self.stride, padding=self.padding)


ADAPTIVE_TYPES = ADAPTIVE_TYPES + (StackedAttenLinear, StackedAttenConv2d)


def convert_to_stacked(net: nn.Module, atten_default: float = 0.0) -> int:
    count = 0
    for name, module in list(net.named_modules()):
        if isinstance(module, (nn.Linear, nn.Conv2d)):
            if not all(p.requires_grad for p in module.parameters()):
                continue
            if isinstance(module, nn.Linear):
                new = StackedAttenLinear(global_weight=module.weight,
                                         adaptive_bias=module.bias,
                                         atten_default=atten_default)
            else:
                new = StackedAttenConv2d(global_weight=module.weight,
                                         adaptive_bias=module.bias,
                                         atten_default=atten_default,
                                         stride=module.stride,
                                         padding=module.padding)
            parent = net
            parts = name.split(".")
            for p in parts[:-1]:
                parent = getattr(parent, p)
            setattr(parent, parts[-1], new)
            count += 1
    return count
