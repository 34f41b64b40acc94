"""ResNet ReID backbones (capability parity with ref:models/resnet.py).

A from-scratch implementation of the reid-strong-baseline ResNet family:
  - configurable `last_stride` (stride-1 stage 4 keeps the 16x8 feature map
    for 128x64 ReID crops — ref:models/resnet.py:182-183)
  - GAP + optional BNNeck head (BN1d with frozen bias + bias-free classifier,
    kaiming/small-normal init — ref:models/resnet.py:296-306)
  - dual-output forward: training returns (cls_score, global_feat), eval
    returns global_feat (ref:models/resnet.py:312-324)

MI355X notes: parameter/compute dtype is managed by the runtime (bf16 autocast
with fp32 master weights); convolutions run through MIOpen until the
hand-written implicit-GEMM HIP kernels (ops/csrc) take over per-layer.
Submodule names (`base.layer1..4`, `bottleneck`, `classifier`) match the
reference so `fine_tuning` yaml lists and ckpt schemas transfer unchanged.
"""

from __future__ import annotations

from typing import List, Type, Union

import torch
import torch.nn as nn
import torch.nn.functional as F

from flreid_amd import ops
from flreid_amd.tools.logger import Logger
from flreid_amd.tools.winit import weights_init_classifier, weights_init_kaiming

_log = Logger("models.resnet")


class EvalFusedBatchNorm2d(nn.BatchNorm2d):
    """BatchNorm2d with fused HIP paths on GPU:
      - EVAL: one-pass normalize kernel (prototype capture / validation —
        the ReID hot eval paths);
      - TRAIN on small channels-last batches (the FedSTIL head epoch on
        cached prototype features): single-kernel fwd/bwd with in-kernel
        statistics and running-stat update, replacing MIOpen's 5-kernel
        chain + fp32 autocast round-trip per layer (see ops/csrc/bn_train.hip).
    When `fuse_relu` is set (bn1/bn2 inside the residual blocks and the stem
    BN — everywhere the model applies BN→ReLU directly) the ReLU runs inside
    the same kernels; the torch fallback applies it explicitly so CPU
    semantics are identical.  CPU and out-of-regime shapes fall through."""

    fuse_relu = False

    def forward(self, x):
        if not self.training:
            out = ops.bn_eval_2d(x, self, relu=self.fuse_relu)
            if out is not None:
                return out
        else:
            out = ops.bn_train_2d(x, self, relu=self.fuse_relu)
            if out is not None:
                return out
        out = super().forward(x)
        return F.relu(out, inplace=True) if self.fuse_relu else out


class EvalFusedBatchNorm1d(nn.BatchNorm1d):
    """BatchNorm1d (the BNNeck bottleneck) with the same fused HIP train and
    eval paths as EvalFusedBatchNorm2d — a contiguous [B, C] tensor is one
    NHWC row block with H·W = 1."""

    def forward(self, x):
        if not self.training:
            out = ops.bn_eval_2d(x, self)
            if out is not None:
                return out
        else:
            out = ops.bn_train_2d(x, self)
            if out is not None:
                return out
        return super().forward(x)


class FusedConv3x3(nn.Conv2d):
    """3×3 s1p1 conv that dispatches to the hand-written CDNA4 halo kernel
    (ops/csrc/conv3x3_img.hip — K1) on the ReID layer-4 shapes, with the
    library conv as fallback.  State-dict identical to nn.Conv2d."""

    def forward(self, x):
        if self.stride == (1, 1) and self.bias is None:
            y = ops.conv3x3_try(x, self.weight)
            if y is not None:
                return y
        return super().forward(x)


def conv3x3(cin: int, cout: int, stride: int = 1) -> nn.Conv2d:
    return FusedConv3x3(cin, cout, 3, stride=stride, padding=1, bias=False)


def conv1x1(cin: int, cout: int, stride: int = 1) -> nn.Conv2d:
    return nn.Conv2d(cin, cout, 1, stride=stride, bias=False)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, cin: int, planes: int, stride: int = 1,
                 downsample: nn.Module = None):
        super().__init__()
        self.conv1 = conv3x3(cin, planes, stride)
        self.bn1 = EvalFusedBatchNorm2d(planes)
        self.bn1.fuse_relu = True
        self.relu = nn.ReLU(inplace=True)
        self.conv2 = conv3x3(planes, planes)
        self.bn2 = EvalFusedBatchNorm2d(planes)
        self.downsample = downsample

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        identity = x if self.downsample is None else self.downsample(x)
        out = self.bn1(self.conv1(x))          # ReLU fused into bn1
        out = self.bn2(self.conv2(out))
        return self.relu(out + identity)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin: int, planes: int, stride: int = 1,
                 downsample: nn.Module = None):
        super().__init__()
        self.conv1 = conv1x1(cin, planes)
        self.bn1 = EvalFusedBatchNorm2d(planes)
        self.bn1.fuse_relu = True
        self.conv2 = conv3x3(planes, planes, stride)
        self.bn2 = EvalFusedBatchNorm2d(planes)
        self.bn2.fuse_relu = True
        self.conv3 = conv1x1(planes, planes * self.expansion)
        self.bn3 = EvalFusedBatchNorm2d(planes * self.expansion)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        identity = x if self.downsample is None else self.downsample(x)
        out = self.bn1(self.conv1(x))          # ReLU fused into bn1/bn2
        out = self.bn2(self.conv2(out))
        out = self.bn3(self.conv3(out))
        return self.relu(out + identity)


class ResNetTrunk(nn.Module):
    """conv7x7/s2 + maxpool + 4 stages + GAP (ref:models/resnet.py:144-244)."""

    def __init__(self, block: Type[Union[BasicBlock, Bottleneck]],
                 layers: List[int], last_stride: int = 2):
        super().__init__()
        self.inplanes = 64
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = EvalFusedBatchNorm2d(64)
        self.bn1.fuse_relu = True
        self.relu = nn.ReLU(inplace=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_stage(block, 64, layers[0])
        self.layer2 = self._make_stage(block, 128, layers[1], stride=2)
        self.layer3 = self._make_stage(block, 256, layers[2], stride=2)
        self.layer4 = self._make_stage(block, 512, layers[3], stride=last_stride)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self._init_weights()

    def _make_stage(self, block, planes: int, depth: int, stride: int = 1) -> nn.Sequential:
        downsample = None
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample = nn.Sequential(
                conv1x1(self.inplanes, planes * block.expansion, stride),
                EvalFusedBatchNorm2d(planes * block.expansion),
            )
        blocks = [block(self.inplanes, planes, stride, downsample)]
        self.inplanes = planes * block.expansion
        blocks += [block(self.inplanes, planes) for _ in range(1, depth)]
        return nn.Sequential(*blocks)

    def _init_weights(self) -> None:
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out", nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.constant_(m.weight, 1.0)
                nn.init.constant_(m.bias, 0.0)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.maxpool(self.bn1(self.conv1(x)))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x)
        return torch.flatten(x, 1)


_ARCH = {
    "resnet18": (BasicBlock, [2, 2, 2, 2], 512),
    "resnet34": (BasicBlock, [3, 4, 6, 3], 512),
    "resnet50": (Bottleneck, [3, 4, 6, 3], 2048),
    "resnet101": (Bottleneck, [3, 4, 23, 3], 2048),
    "resnet152": (Bottleneck, [3, 8, 36, 3], 2048),
}


class ResNetReID(nn.Module):
    def __init__(self, model_name: str, num_classes: int = 1000,
                 last_stride: int = 2, neck: str = "no",
                 pretrained: bool = False, **kwargs):
        super().__init__()
        for n, p in kwargs.items():
            setattr(self, n, p)
        if model_name not in _ARCH:
            raise ValueError(f"No model named {model_name}.")
        block, layers, self.in_planes = _ARCH[model_name]
        self.model_name = model_name
        self.num_classes = num_classes
        self.neck = neck
        self.base = ResNetTrunk(block, layers, last_stride=last_stride)

        if neck == "no":
            self.classifier = nn.Linear(self.in_planes, num_classes)
        elif neck == "bnneck":
            self.bottleneck = EvalFusedBatchNorm1d(self.in_planes)
            self.bottleneck.bias.requires_grad_(False)
            self.classifier = nn.Linear(self.in_planes, num_classes, bias=False)
            self.bottleneck.apply(weights_init_kaiming)
            self.classifier.apply(weights_init_classifier)
        else:
            raise ValueError(f"Mismatched neck type {neck}.")

        if pretrained:
            self._try_load_imagenet()

    def _try_load_imagenet(self) -> None:
        """Optional torch.hub ImageNet weights (ref:models/resnet.py:308-310).
        This environment has no network; the benchmark path is random-init
        synthetic (BASELINE.json), so failure is a warning, not an error."""
        try:
            from torch.hub import load_state_dict_from_url
            urls = {
                "resnet18": "https://download.pytorch.org/models/resnet18-f37072fd.pth",
                "resnet34": "https://download.pytorch.org/models/resnet34-b627a593.pth",
                "resnet50": "https://download.pytorch.org/models/resnet50-0676ba61.pth",
                "resnet101": "https://download.pytorch.org/models/resnet101-63fe2227.pth",
                "resnet152": "https://download.pytorch.org/models/resnet152-394f9c45.pth",
            }
            sd = load_state_dict_from_url(urls[self.model_name], progress=False)
            sd.pop("fc.weight", None), sd.pop("fc.bias", None)
            # torchvision naming -> ours: stages/stem names already line up
            self.base.load_state_dict(sd, strict=False)
        except Exception as e:  # pragma: no cover
            _log.warn(f"pretrained weights unavailable ({e}); using random init")

    # ---- staged execution --------------------------------------------------
    # FedSTIL trains only the "head" (everything from the first adaptive layer
    # on) on cached prototype features.  The reference located that split with
    # a torch.fx graph surgery (ref:methods/fedstil.py:258-288 +
    # ref:tools/utils.py:139-182); this framework owns its model definitions,
    # so the split is an explicit stage list — no tracer, no graph rewrite.
    STAGES = ("stem", "layer1", "layer2", "layer3", "layer4", "head")

    def stage_of(self, module_path: str) -> int:
        """Stage index that contains the (dotted) submodule path."""
        if module_path.startswith("base.layer"):
            return self.STAGES.index(module_path.split(".")[1])
        if module_path.startswith("base."):
            return 0
        return len(self.STAGES) - 1          # bottleneck / classifier

    def run_stages(self, x: torch.Tensor, start: int = 0, tap: int = None):
        """Run stages [start, end); optionally capture the INPUT of stage
        `tap` (the prototype feature FedSTIL caches).  Returns (output,
        tap_value) with output matching forward()'s train/eval convention."""
        tap_value = None
        b = self.base
        stages = [
            lambda t: b.maxpool(b.bn1(b.conv1(t))),
            b.layer1, b.layer2, b.layer3, b.layer4,
        ]
        for idx in range(start, 5):
            if tap == idx:
                tap_value = x
            x = stages[idx](x)
        if tap == 5:
            tap_value = x
        # the head stage consumes the (spatial) layer4 output: pool -> neck ->
        # classifier; a head-only invocation passes the cached spatial tap in
        global_feat = torch.flatten(b.avgpool(x), 1)
        feat = self.bottleneck(global_feat) if self.neck == "bnneck" else global_feat
        if self.training:
            return (self.classifier(feat), global_feat), tap_value
        return global_feat, tap_value

    def forward(self, x: torch.Tensor):
        out, _ = self.run_stages(x, 0)
        return out


def resnet18(**kw): return ResNetReID("resnet18", **kw)
def resnet34(**kw): return ResNetReID("resnet34", **kw)
def resnet50(**kw): return ResNetReID("resnet50", **kw)
def resnet101(**kw): return ResNetReID("resnet101", **kw)
def resnet152(**kw): return ResNetReID("resnet152", **kw)
