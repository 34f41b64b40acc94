"""Grad-CAM attention visualisation (ref:analyse/visualize.py:33-54).

Self-contained Grad-CAM (the reference pulled in the grad-cam package):
forward + backward hooks on a target layer, channel-weighted activation map.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F


@torch.enable_grad()
def grad_cam(model: torch.nn.Module, target_layer: torch.nn.Module,
             image: torch.Tensor, class_index: Optional[int] = None) -> torch.Tensor:
    """Returns a [H, W] attention map in [0, 1] for `image` [1, 3, H, W]."""
    acts, grads = {}, {}

    def fwd_hook(_m, _i, out):
        acts["v"] = out

    def bwd_hook(_m, _gi, gout):
        grads["v"] = gout[0]

    h1 = target_layer.register_forward_hook(fwd_hook)
    h2 = target_layer.register_full_backward_hook(bwd_hook)
    try:
        was_training = model.training
        model.train()        # dual-output forward provides the class scores
        if image.shape[0] == 1:
            # batch of 2 keeps train-mode BatchNorm happy; CAM reads sample 0
            image = torch.cat([image, image], dim=0)
        score, _feat = model(image)
        if class_index is None:
            class_index = int(score[0].argmax())
        model.zero_grad(set_to_none=True)
        score[0, class_index].backward()
        a, g = acts["v"], grads["v"]
        weights = g.mean(dim=(2, 3), keepdim=True)       # GAP over spatial
        cam = F.relu((weights * a).sum(dim=1, keepdim=True))
        cam = F.interpolate(cam, size=image.shape[-2:], mode="bilinear",
                            align_corners=False)[0, 0]
        cam = cam - cam.min()
        if float(cam.max()) > 0:
            cam = cam / cam.max()
        if not was_training:
            model.eval()
        return cam.detach()
    finally:
        h1.remove()
        h2.remove()
