"""Label-smoothed cross entropy (ref:criterions/cross_entropy.py:11-41).

The loss itself lives in flreid_amd.ops (fused log-softmax + smoothed NLL HIP
kernel on GPU; the reference built the one-hot on the CPU and shipped it to
the device every batch — ref:criterions/cross_entropy.py:36-38).
"""

from __future__ import annotations

from flreid_amd import ops
from flreid_amd.modules.criterion import CriterionModule


class CrossEntropyLabelSmooth(CriterionModule):
    def __init__(self, num_classes: int, epsilon: float = 0.1, **kwargs):
        super().__init__()
        for n, p in kwargs.items():
            setattr(self, n, p)
        self.num_classes = num_classes
        self.epsilon = epsilon

    def forward(self, score, target, **kwargs):
        return ops.ce_label_smooth(score, target, self.epsilon)
