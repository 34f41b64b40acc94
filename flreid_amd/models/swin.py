"""Swin Transformer ReID backbones (capability parity with
ref:models/swin_transformer.py).

From-scratch Swin v1 (tiny/small/base/large) with the ReID head of the
reference wrapper: input resized to 224², 4 stages of shifted-window blocks,
LN + token avgpool, BNNeck head, dual train/eval outputs
(ref:models/swin_transformer.py:604-700).

MI355X notes: the per-window attention (49 tokens) runs through
flreid_amd.ops.window_attention — eager composition on CPU, the fused
CDNA4 MFMA kernel (K3 in SURVEY.md §2.9) on GPU.  Staged execution
(`stage_of`/`run_stages`) mirrors models/resnet.py so FedSTIL's head-only
training path works unchanged (stages: stem, layers.0-3, head).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn.functional as F
from torch import nn

from flreid_amd import ops
from flreid_amd.tools.winit import weights_init_classifier, weights_init_kaiming


def window_partition(x: torch.Tensor, ws: int) -> torch.Tensor:
    """[B, H, W, C] -> [B·nW, ws·ws, C]"""
    b, h, w, c = x.shape
    x = x.view(b, h // ws, ws, w // ws, ws, c)
    return x.permute(0, 1, 3, 2, 4, 5).reshape(-1, ws * ws, c)


def window_reverse(win: torch.Tensor, ws: int, h: int, w: int) -> torch.Tensor:
    """[B·nW, ws·ws, C] -> [B, H, W, C]"""
    b = win.shape[0] // ((h // ws) * (w // ws))
    x = win.view(b, h // ws, w // ws, ws, ws, -1)
    return x.permute(0, 1, 3, 2, 4, 5).reshape(b, h, w, -1)


class DropPath(nn.Module):
    def __init__(self, p: float = 0.0):
        super().__init__()
        self.p = p

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.p == 0.0 or not self.training:
            return x
        keep = 1.0 - self.p
        mask = torch.rand(x.shape[0], *([1] * (x.dim() - 1)),
                          device=x.device, dtype=x.dtype) < keep
        return x * mask / keep


class WindowAttention(nn.Module):
    """W-MSA with relative position bias (ref:models/swin_transformer.py:208-286)."""

    def __init__(self, dim: int, window_size: int, num_heads: int,
                 qkv_bias: bool = True, attn_drop: float = 0.0,
                 proj_drop: float = 0.0):
        super().__init__()
        self.dim = dim
        self.window_size = window_size
        self.num_heads = num_heads
        self.scale = (dim // num_heads) ** -0.5

        n = 2 * window_size - 1
        self.relative_position_bias_table = nn.Parameter(
            torch.zeros(n * n, num_heads))
        coords = torch.stack(torch.meshgrid(
            torch.arange(window_size), torch.arange(window_size),
            indexing="ij")).flatten(1)                       # [2, ws*ws]
        rel = coords[:, :, None] - coords[:, None, :]        # [2, N, N]
        rel = rel.permute(1, 2, 0) + (window_size - 1)
        index = rel[..., 0] * n + rel[..., 1]                # [N, N]
        self.register_buffer("relative_position_index", index)

        self.qkv = nn.Linear(dim, dim * 3, bias=qkv_bias)
        self.attn_drop = nn.Dropout(attn_drop)
        self.proj = nn.Linear(dim, dim)
        self.proj_drop = nn.Dropout(proj_drop)
        nn.init.trunc_normal_(self.relative_position_bias_table, std=0.02)

    def forward(self, x: torch.Tensor, mask: Optional[torch.Tensor] = None):
        bn, n, c = x.shape
        qkv = self.qkv(x).view(bn, n, 3, self.num_heads, c // self.num_heads)
        q, k, v = qkv.permute(2, 0, 3, 1, 4).unbind(0)       # [bn, h, n, d]
        bias = self.relative_position_bias_table[
            self.relative_position_index.view(-1)].view(n, n, -1)
        bias = bias.permute(2, 0, 1).contiguous()            # [h, n, n]
        out = ops.window_attention(q, k, v, bias * 1.0, mask, self.scale,
                                   self.attn_drop if self.training else None)
        out = out.transpose(1, 2).reshape(bn, n, c)
        return self.proj_drop(self.proj(out))


class SwinBlock(nn.Module):
    def __init__(self, dim: int, input_resolution: Tuple[int, int],
                 num_heads: int, window_size: int = 7, shift: int = 0,
                 mlp_ratio: float = 4.0, qkv_bias: bool = True,
                 drop: float = 0.0, attn_drop: float = 0.0,
                 drop_path: float = 0.0):
        super().__init__()
        self.dim = dim
        self.input_resolution = input_resolution
        self.window_size = min(window_size, *input_resolution)
        self.shift = 0 if self.window_size >= min(input_resolution) and shift else shift
        if min(input_resolution) <= self.window_size:
            self.shift = 0

        self.norm1 = nn.LayerNorm(dim)
        self.attn = WindowAttention(dim, self.window_size, num_heads, qkv_bias,
                                    attn_drop, drop)
        self.drop_path = DropPath(drop_path)
        self.norm2 = nn.LayerNorm(dim)
        hidden = int(dim * mlp_ratio)
        self.mlp = nn.Sequential(nn.Linear(dim, hidden), nn.GELU(),
                                 nn.Dropout(drop), nn.Linear(hidden, dim),
                                 nn.Dropout(drop))

        if self.shift > 0:
            self.register_buffer("attn_mask", self._build_mask(), persistent=False)
        else:
            self.attn_mask = None

    def _build_mask(self) -> torch.Tensor:
        """Shifted-window cross-region mask (ref:models/swin_transformer.py:333-356)."""
        h, w = self.input_resolution
        ws, sh = self.window_size, self.shift
        img_mask = torch.zeros(1, h, w, 1)
        cnt = 0
        for hs in (slice(0, -ws), slice(-ws, -sh), slice(-sh, None)):
            for wslice in (slice(0, -ws), slice(-ws, -sh), slice(-sh, None)):
                img_mask[:, hs, wslice, :] = cnt
                cnt += 1
        windows = window_partition(img_mask, ws).squeeze(-1)   # [nW, ws*ws]
        diff = windows.unsqueeze(1) - windows.unsqueeze(2)
        return diff.masked_fill(diff != 0, -100.0).masked_fill(diff == 0, 0.0)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        h, w = self.input_resolution
        b, L, c = x.shape
        shortcut = x
        x = self.norm1(x).view(b, h, w, c)
        if self.shift > 0:
            x = torch.roll(x, shifts=(-self.shift, -self.shift), dims=(1, 2))
        win = window_partition(x, self.window_size)            # [b·nW, N, C]
        win = self.attn(win, self.attn_mask)
        x = window_reverse(win, self.window_size, h, w)
        if self.shift > 0:
            x = torch.roll(x, shifts=(self.shift, self.shift), dims=(1, 2))
        x = shortcut + self.drop_path(x.view(b, L, c))
        return x + self.drop_path(self.mlp(self.norm2(x)))


class PatchMerging(nn.Module):
    """2×2 concat + LN + Linear 4C→2C (ref:models/swin_transformer.py:398-444)."""

    def __init__(self, input_resolution: Tuple[int, int], dim: int):
        super().__init__()
        self.input_resolution = input_resolution
        self.dim = dim
        self.norm = nn.LayerNorm(4 * dim)
        self.reduction = nn.Linear(4 * dim, 2 * dim, bias=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        h, w = self.input_resolution
        b, L, c = x.shape
        # fused K4 path (patch_merge.hip): gather + LN in one kernel, the
        # concat tensor never materialises; eager fallback elsewhere
        y = ops.patch_merge_ln(x, self.norm.weight, self.norm.bias, h, w,
                               self.norm.eps)
        if y is None:
            x = x.view(b, h, w, c)
            x = torch.cat([x[:, 0::2, 0::2], x[:, 1::2, 0::2],
                           x[:, 0::2, 1::2], x[:, 1::2, 1::2]], dim=-1)
            y = self.norm(x.view(b, -1, 4 * c))
        return self.reduction(y)


class SwinStage(nn.Module):
    def __init__(self, dim, input_resolution, depth, num_heads, window_size,
                 mlp_ratio, qkv_bias, drop, attn_drop, drop_path,
                 downsample: bool, use_checkpoint: bool = False):
        super().__init__()
        self.use_checkpoint = use_checkpoint
        self.blocks = nn.ModuleList([
            SwinBlock(dim, input_resolution, num_heads, window_size,
                      shift=0 if i % 2 == 0 else window_size // 2,
                      mlp_ratio=mlp_ratio, qkv_bias=qkv_bias, drop=drop,
                      attn_drop=attn_drop,
                      drop_path=drop_path[i] if isinstance(drop_path, (list, tuple)) else drop_path)
            for i in range(depth)])
        self.downsample = PatchMerging(input_resolution, dim) if downsample else None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        for blk in self.blocks:
            if self.use_checkpoint and self.training:
                x = torch.utils.checkpoint.checkpoint(blk, x, use_reentrant=False)
            else:
                x = blk(x)
        if self.downsample is not None:
            x = self.downsample(x)
        return x


class SwinTrunk(nn.Module):
    def __init__(self, img_size=224, patch_size=4, in_chans=3, embed_dim=96,
                 depths=(2, 2, 6, 2), num_heads=(3, 6, 12, 24), window_size=7,
                 mlp_ratio=4.0, qkv_bias=True, drop_rate=0.0,
                 attn_drop_rate=0.0, drop_path_rate=0.1,
                 use_checkpoint=False):
        super().__init__()
        self.num_layers = len(depths)
        self.embed_dim = embed_dim
        self.num_features = int(embed_dim * 2 ** (self.num_layers - 1))
        self.patches_resolution = (img_size // patch_size, img_size // patch_size)

        self.patch_embed = nn.Conv2d(in_chans, embed_dim, patch_size, patch_size)
        self.patch_norm = nn.LayerNorm(embed_dim)
        self.pos_drop = nn.Dropout(drop_rate)

        dpr = torch.linspace(0, drop_path_rate, sum(depths)).tolist()
        self.layers = nn.ModuleList()
        for i in range(self.num_layers):
            res = (self.patches_resolution[0] // 2 ** i,
                   self.patches_resolution[1] // 2 ** i)
            self.layers.append(SwinStage(
                dim=int(embed_dim * 2 ** i), input_resolution=res,
                depth=depths[i], num_heads=num_heads[i],
                window_size=window_size, mlp_ratio=mlp_ratio,
                qkv_bias=qkv_bias, drop=drop_rate, attn_drop=attn_drop_rate,
                drop_path=dpr[sum(depths[:i]):sum(depths[:i + 1])],
                downsample=i < self.num_layers - 1,
                use_checkpoint=use_checkpoint))
        self.norm = nn.LayerNorm(self.num_features)
        self.apply(self._init)

    @staticmethod
    def _init(m):
        if isinstance(m, nn.Linear):
            nn.init.trunc_normal_(m.weight, std=0.02)
            if m.bias is not None:
                nn.init.constant_(m.bias, 0)
        elif isinstance(m, nn.LayerNorm):
            nn.init.constant_(m.weight, 1.0)
            nn.init.constant_(m.bias, 0)

    def stem(self, x: torch.Tensor) -> torch.Tensor:
        x = self.patch_embed(x).flatten(2).transpose(1, 2)   # [B, L, C]
        return self.pos_drop(self.patch_norm(x))

    def head_pool(self, x: torch.Tensor) -> torch.Tensor:
        return self.norm(x).mean(dim=1)

    def forward_features(self, x: torch.Tensor) -> torch.Tensor:
        x = self.stem(x)
        for layer in self.layers:
            x = layer(x)
        return self.head_pool(x)


_SWIN_ARCH = {
    "swin_tiny": dict(embed_dim=96, depths=(2, 2, 6, 2), num_heads=(3, 6, 12, 24)),
    "swin_small": dict(embed_dim=96, depths=(2, 2, 18, 2), num_heads=(3, 6, 12, 24)),
    "swin_base": dict(embed_dim=128, depths=(2, 2, 18, 2), num_heads=(4, 8, 16, 32)),
    "swin_large": dict(embed_dim=192, depths=(2, 2, 18, 2), num_heads=(6, 12, 24, 48)),
}


class SwinReID(nn.Module):
    STAGES = ("stem", "layer0", "layer1", "layer2", "layer3", "head")

    def __init__(self, model_name: str, num_classes: int = 1000,
                 neck: str = "no", img_size: int = 224, window_size: int = 7,
                 use_checkpoint: bool = False, **kwargs):
        super().__init__()
        for n, p in kwargs.items():
            setattr(self, n, p)
        if model_name not in _SWIN_ARCH:
            raise ValueError(f"No model named {model_name}.")
        self.model_name = model_name
        self.num_classes = num_classes
        self.neck = neck
        self.img_size = img_size
        self.base = SwinTrunk(img_size=img_size, window_size=window_size,
                              use_checkpoint=use_checkpoint,
                              **_SWIN_ARCH[model_name])
        self.in_planes = self.base.num_features

        if neck == "no":
            self.classifier = nn.Linear(self.in_planes, num_classes)
        elif neck == "bnneck":
            from flreid_amd.models.resnet import EvalFusedBatchNorm1d
            self.bottleneck = EvalFusedBatchNorm1d(self.in_planes)
            self.bottleneck.bias.requires_grad_(False)
            self.classifier = nn.Linear(self.in_planes, num_classes, bias=False)
            self.bottleneck.apply(weights_init_kaiming)
            self.classifier.apply(weights_init_classifier)
        else:
            raise ValueError(f"Mismatched neck type {neck}.")

    # ---- staged execution (FedSTIL head-only training; see resnet.py) ------
    def stage_of(self, module_path: str) -> int:
        if module_path.startswith("base.layers."):
            return int(module_path.split(".")[2]) + 1
        if module_path.startswith("base."):
            return 0
        return len(self.STAGES) - 1

    def run_stages(self, x: torch.Tensor, start: int = 0, tap: int = None):
        tap_value = None
        if start == 0:
            if x.shape[-2:] != (self.img_size, self.img_size):
                x = F.interpolate(x, size=(self.img_size, self.img_size),
                                  mode="bilinear", align_corners=False)
            if tap == 0:
                tap_value = x
            x = self.base.stem(x)
        for idx in range(max(start - 1, 0), 4):
            stage = idx + 1
            if stage < start:
                continue
            if tap == stage:
                tap_value = x
            x = self.base.layers[idx](x)
        if tap == 5:
            tap_value = x
        global_feat = self.base.head_pool(x)
        feat = self.bottleneck(global_feat) if self.neck == "bnneck" else global_feat
        if self.training:
            return (self.classifier(feat), global_feat), tap_value
        return global_feat, tap_value

    def forward(self, x: torch.Tensor):
        out, _ = self.run_stages(x, 0)
        return out


def swin_reid(size: str, **kwargs):
    return SwinReID(f"swin_{size}", **kwargs)
