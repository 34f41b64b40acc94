"""Loss registry (ref:criterions/__init__.py:4-7).

`distill_kl` is additionally registered here (the reference ships DistillKL
but leaves it out of its registry — ref:criterions/kd_loss.py:10-27).
"""

from flreid_amd.criterions.cross_entropy import CrossEntropyLabelSmooth
from flreid_amd.criterions.kd_loss import DistillKL
from flreid_amd.criterions.triplet_loss import TripletLoss

criterions = {
    "cross_entropy": CrossEntropyLabelSmooth,
    "triplet_loss": TripletLoss,
    "distill_kl": DistillKL,
}
