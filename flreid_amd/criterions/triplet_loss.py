"""Triplet loss with batch-hard / softmax-weighted mining
(ref:criterions/triplet_loss.py:12-127).  Math lives in flreid_amd.ops
(fused pairwise-distance + mining HIP kernel on GPU)."""

from __future__ import annotations

from flreid_amd import ops
from flreid_amd.modules.criterion import CriterionModule


class TripletLoss(CriterionModule):
    def __init__(self, margin=None, norm_feat: bool = False,
                 hard_mining: bool = False, **kwargs):
        super().__init__()
        for n, p in kwargs.items():
            setattr(self, n, p)
        self.margin = margin
        self.norm_feat = norm_feat
        self.hard_mining = hard_mining

    def forward(self, feature, target, **kwargs):
        # operators call criteria with keywords (score=, feature=, target=);
        # the unused score lands in **kwargs
        return ops.triplet_loss(feature, target, margin=self.margin,
                                norm_feat=self.norm_feat,
                                hard_mining=self.hard_mining)
