"""ReID CMC / mAP evaluation (ref:tools/evaluate.py).

The reference loops over queries in python, doing a GEMV + np.argsort per
query (ref:tools/evaluate.py:103-142).  Here the whole evaluation is one
batched Q×G similarity GEMM + a vectorised rank reduction (ops.cmc_map),
which runs on-device (MI355X: MFMA GEMM + device sort) or on CPU.
"""

from __future__ import annotations

from typing import Any, Optional, Tuple

import numpy as np
import torch

from flreid_amd import ops


@torch.no_grad()
def calculate_similarity_distance(query_feature: torch.Tensor,
                                  gallery_features: torch.Tensor) -> Any:
    """Single query vs gallery similarity (ref:tools/evaluate.py:87-100)."""
    if isinstance(query_feature, np.ndarray):
        return np.dot(gallery_features, query_feature)
    return (gallery_features @ query_feature.view(-1, 1)).squeeze(1).cpu().numpy()


@torch.no_grad()
def evaluate(query_features: torch.Tensor, query_labels: torch.Tensor,
             gallery_features: torch.Tensor, gallery_labels: torch.Tensor,
             query_camera_labels: Optional[torch.Tensor] = None,
             gallery_camera_labels: Optional[torch.Tensor] = None,
             device: str = "cpu") -> Tuple[np.ndarray, float]:
    """CMC curve (len == gallery size) + mAP, averaged over ALL queries
    (no-match queries contribute 0 — ref:tools/evaluate.py:137-142)."""
    qf = query_features.to(device)
    gf = gallery_features.to(device)
    cmc, mAP = ops.cmc_map(qf, query_labels, gf, gallery_labels,
                           query_camera_labels, gallery_camera_labels)
    return cmc.numpy(), mAP
