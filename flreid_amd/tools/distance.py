"""Distance functions (ref:tools/distance.py) — thin wrappers over flreid ops."""

from __future__ import annotations

import torch

from flreid_amd import ops


def compute_euclidean_distance(features: torch.Tensor, others: torch.Tensor) -> torch.Tensor:
    """Squared euclidean pairwise distance (ref:tools/distance.py:9-16)."""
    return ops.pairwise_sqeuclidean(features, others)


def compute_cosine_distance(features: torch.Tensor, others: torch.Tensor) -> torch.Tensor:
    """Cosine distance 1 − cos (ref:tools/distance.py:19-30)."""
    return ops.pairwise_cosine_distance(features, others)


def compute_kl_distance(feature: torch.Tensor, others: torch.Tensor) -> torch.Tensor:
    """KL task-token affinity (ref:tools/distance.py:33-36)."""
    return ops.kl_distance(feature, others)
