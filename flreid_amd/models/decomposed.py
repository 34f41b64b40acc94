"""FedWeIT parameter-decomposition layers (ref:methods/fedweit.py:33-268).

θ = mask ⊙ sw + aw + Σ_k atten_k · aw_kb[..., k], with train-time L1
hard-threshold pruning of aw (λ_l1) and mask (λ_mask)
(ref:methods/fedweit.py:122-136).

Design note: the reference stores every tensor fully TRANSPOSED
(`tensor_reverse_permute`, an artifact of porting the original TF code —
ref:methods/fedweit.py:87-96) and transposes back on every forward.  Here
tensors live in their natural torch orientation: `mask` is a per-OUTPUT-
channel vector broadcast over dim 0, `aw_kb` stacks kb_cnt client adaptives
along a NEW LAST dim.  Semantics are identical; the ckpt schema stores the
natural orientation.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn.functional as F
from torch import nn
from torch.nn import Parameter


def l1_hard_threshold(weights: torch.Tensor, threshold: float) -> torch.Tensor:
    """w · 1[|w| > λ] (ref:methods/fedweit.py:122-125)."""
    return weights * torch.greater(weights.abs(), threshold).to(weights.dtype)


class DecomposedBase(nn.Module):
    def __init__(self, shared_weight: torch.Tensor,
                 bias: Optional[torch.Tensor] = None,
                 mask: Optional[torch.Tensor] = None,
                 adaptive: Optional[torch.Tensor] = None,
                 knowledge_base: Optional[torch.Tensor] = None,
                 atten: Optional[torch.Tensor] = None,
                 lambda_l1: float = 1e-3, lambda_mask: float = 0.0,
                 kb_cnt: int = 5, **kwargs):
        super().__init__()
        self.kb_cnt = kb_cnt if kb_cnt is not None else 5
        self.lambda_l1 = lambda_l1 if lambda_l1 is not None else 1e-3
        self.lambda_mask = lambda_mask if lambda_mask is not None else 0.0

        self.sw = Parameter(torch.empty(0), requires_grad=False)
        self.bias = Parameter(torch.empty(0)) if bias is not None else None
        self.mask = Parameter(torch.empty(0))
        self.aw = Parameter(torch.empty(0))
        self.aw_kb = Parameter(torch.empty(0), requires_grad=False)
        self.atten = Parameter(torch.empty(0))
        self.init_training_weights(shared_weight, bias, mask, adaptive,
                                   knowledge_base, atten)

    @staticmethod
    def _assign(param: Parameter, value: torch.Tensor) -> None:
        if param.data.shape == value.shape:
            param.data.copy_(value)
        else:
            param.data = value.detach().clone().to(param.device)

    def _mask_bcast_shape(self):
        """mask is per-output-channel: broadcast over all dims but dim 0."""
        return (-1,) + (1,) * (self.sw.dim() - 1)

    @torch.no_grad()
    def init_training_weights(self, shared_weight=None, bias=None, mask=None,
                              adaptive=None, knowledge_base=None, atten=None):
        if shared_weight is None:
            shared_weight = self.sw.data
        self._assign(self.sw, shared_weight.detach())
        self.sw.requires_grad = False

        if bias is not None and self.bias is not None:
            self._assign(self.bias, bias.detach())
            self.bias.requires_grad = True

        if mask is None:
            mask = torch.sigmoid(torch.zeros(self.sw.shape[0],
                                             device=self.sw.device))
        self._assign(self.mask, mask.detach())
        self.mask.requires_grad = True

        if adaptive is None:
            adaptive = (1.0 - self.mask.data.view(self._mask_bcast_shape())) \
                       * self.sw.data
        self._assign(self.aw, adaptive.detach())
        self.aw.requires_grad = True

        if knowledge_base is None:
            knowledge_base = torch.zeros(*self.sw.shape, self.kb_cnt,
                                         device=self.sw.device)
        self._assign(self.aw_kb, knowledge_base.detach())
        self.aw_kb.requires_grad = False

        if atten is None:
            atten = torch.zeros(self.kb_cnt, device=self.sw.device)
        self._assign(self.atten, atten.detach())
        self.atten.requires_grad = True

    def composed_weight(self) -> torch.Tensor:
        aw = self.aw if not self.training else l1_hard_threshold(self.aw, self.lambda_l1)
        mask = self.mask if not self.training else l1_hard_threshold(self.mask, self.lambda_mask)
        kb = (self.atten * self.aw_kb).sum(dim=-1)
        return mask.view(self._mask_bcast_shape()) * self.sw + aw + kb


class DecomposedLinear(DecomposedBase):
    def forward(self, data: torch.Tensor) -> torch.Tensor:
        return F.linear(data, self.composed_weight(), self.bias)


class DecomposedConv2d(DecomposedBase):
    def __init__(self, shared_weight, stride=1, padding=0, **kwargs):
        super().__init__(shared_weight, **kwargs)
        self.stride = stride
        self.padding = padding

    def forward(self, data: torch.Tensor) -> torch.Tensor:
        return F.conv2d(data, self.composed_weight(), self.bias,
                        stride=self.stride, padding=self.padding)


class DecomposedBatchNorm2d(DecomposedBase):
    """Parity with ref:methods/fedweit.py:176-238 (out of the reference's
    transform LUT, shipped for completeness)."""

    def __init__(self, shared_weight, running_mean=None, running_var=None,
                 num_batches_tracked=None, track_running_stats=False,
                 momentum=0.1, eps=1e-5, **kwargs):
        super().__init__(shared_weight, **kwargs)
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
            self.composed_weight(), self.bias,
            training, self.momentum or 0.0, self.eps)


class DecomposedLayerNorm(DecomposedBase):
    def __init__(self, shared_weight, normalized_shape: Tuple[int, ...] = None,
                 eps: float = 1e-5, **kwargs):
        super().__init__(shared_weight, **kwargs)
        self.normalized_shape = normalized_shape or tuple(shared_weight.shape)
        self.eps = eps

    def forward(self, data: torch.Tensor) -> torch.Tensor:
        return F.layer_norm(data, self.normalized_shape, self.composed_weight(),
                            self.bias, self.eps)


DECOMPOSED_TYPES = (DecomposedLinear, DecomposedConv2d, DecomposedBatchNorm2d,
                    DecomposedLayerNorm)


def convert_to_decomposed(net: nn.Module, lambda_l1: float, lambda_mask: float,
                          kb_cnt: int) -> int:
    """Rewrite fully-trainable Linear/Conv2d leaves
    (ref:methods/fedweit.py:297-363; BN/LN stay plain like the reference LUT)."""
    count = 0
    for name, module in list(net.named_modules()):
        if isinstance(module, (nn.Linear, nn.Conv2d)):
            if not all(p.requires_grad for p in module.parameters()):
                continue
            if isinstance(module, nn.Linear):
                new = DecomposedLinear(shared_weight=module.weight,
                                       bias=module.bias, lambda_l1=lambda_l1,
                                       lambda_mask=lambda_mask, kb_cnt=kb_cnt)
            else:
                new = DecomposedConv2d(shared_weight=module.weight,
                                       bias=module.bias, stride=module.stride,
                                       padding=module.padding,
                                       lambda_l1=lambda_l1,
                                       lambda_mask=lambda_mask, kb_cnt=kb_cnt)
            parent = net
            parts = name.split(".")
            for p in parts[:-1]:
                parent = getattr(parent, p)
            setattr(parent, parts[-1], new)
            count += 1
    return count


def decomposed_leaves(net: nn.Module):
    return [(n, m) for n, m in net.named_modules() if isinstance(m, DECOMPOSED_TYPES)]


def non_decomposed_leaves(net: nn.Module):
    out = []
    for n, m in net.named_modules():
        if len(list(m.children())) == 0 and not isinstance(m, DECOMPOSED_TYPES) and n:
            out.append((n, m))
    return out
