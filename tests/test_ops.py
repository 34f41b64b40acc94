"""Op reference implementations vs hand-computed math
(these same references are the fp32 ground truth for the HIP kernels)."""

import math

import pytest
import torch
import torch.nn.functional as F

from flreid_amd.ops import reference as ops


def test_pairwise_sqeuclidean():
    a = torch.randn(5, 7)
    b = torch.randn(9, 7)
    d = ops.pairwise_sqeuclidean(a, b)
    naive = torch.stack([((a[i] - b[j]) ** 2).sum() for i in range(5) for j in range(9)]).view(5, 9)
    assert torch.allclose(d, naive, atol=1e-5)


def test_pairwise_cosine():
    a = torch.randn(4, 6)
    b = torch.randn(3, 6)
    d = ops.pairwise_cosine_distance(a, b)
    naive = torch.stack([
        1 - F.cosine_similarity(a[i], b[j], dim=0) for i in range(4) for j in range(3)
    ]).view(4, 3)
    assert torch.allclose(d, naive, atol=1e-6)


def test_ce_label_smooth_value_and_grad():
    torch.manual_seed(0)
    score = torch.randn(6, 10, requires_grad=True)
    target = torch.randint(0, 10, (6,))
    eps = 0.1
    loss = ops.ce_label_smooth(score, target, eps)
    # manual: (-t_smooth * logsoftmax).mean(0).sum()  (ref formula)
    lp = F.log_softmax(score, dim=1)
    onehot = F.one_hot(target, 10).float()
    t = (1 - eps) * onehot + eps / 10
    manual = (-t * lp).mean(0).sum()
    assert torch.allclose(loss, manual, atol=1e-6)
    loss.backward()
    g = score.grad.clone()
    # analytic grad: (softmax - t)/B
    expected = (F.softmax(score.detach(), 1) - t) / 6
    assert torch.allclose(g, expected, atol=1e-6)


def test_triplet_hard_mining_matches_manual():
    torch.manual_seed(1)
    feat = torch.randn(8, 4)
    target = torch.tensor([0, 0, 1, 1, 2, 2, 3, 3])
    loss = ops.triplet_loss(feat, target, margin=0.3, hard_mining=True)
    d = ops.pairwise_sqeuclidean(feat, feat)
    n = 8
    is_pos = target.view(n, 1).eq(target.view(1, n)).float()
    is_neg = 1 - is_pos
    ap = (d * is_pos).max(1)[0]
    an = (d * is_neg + is_pos * 1e9).min(1)[0]
    manual = F.margin_ranking_loss(an, ap, torch.ones(n), margin=0.3)
    assert torch.allclose(loss, manual, atol=1e-6)


def test_triplet_soft_margin():
    torch.manual_seed(2)
    feat = torch.randn(8, 4)
    target = torch.tensor([0, 0, 1, 1, 2, 2, 3, 3])
    loss = ops.triplet_loss(feat, target, margin=None, hard_mining=False)
    assert torch.isfinite(loss)


def test_kd_loss_matches_formula():
    torch.manual_seed(3)
    s = torch.randn(5, 12)
    t = torch.randn(5, 12)
    T = 4.0
    loss = ops.kd_loss(s, t, T)
    manual = F.kl_div(F.log_softmax(s / T, 1), F.softmax(t / T, 1),
                      reduction="sum") * T * T / 5
    assert torch.allclose(loss, manual)


def test_importance_update_and_penalty():
    p = {"w": torch.tensor([1.0, 2.0])}
    anchors = {"w": torch.tensor([0.0, 0.0])}
    imp = {"w": torch.zeros(2)}
    ops.importance_update(imp, {"w": torch.tensor([3.0, -2.0])}, mode="sq")
    assert torch.allclose(imp["w"], torch.tensor([9.0, 4.0]))
    ops.importance_update(imp, {"w": torch.tensor([1.0, -1.0])}, mode="abs")
    assert torch.allclose(imp["w"], torch.tensor([10.0, 5.0]))
    pen = ops.quadratic_penalty(p, anchors, imp)
    assert pen.item() == pytest.approx(10 * 1 + 5 * 4)
    prox = ops.quadratic_penalty(p, anchors)       # FedProx: no importance
    assert prox.item() == pytest.approx(1 + 4)


def test_adaptive_compose_last_dim_broadcast():
    gw = torch.randn(4, 3)
    atten = torch.tensor([0.5, 1.0, 2.0])         # shape == last dim
    aw = torch.randn(4, 3)
    theta = ops.adaptive_compose(gw, atten, aw)
    assert torch.allclose(theta, atten * gw + aw)
    # conv weight [out,in,kh,kw] with atten over kw
    gw4 = torch.randn(2, 3, 3, 3)
    atten4 = torch.rand(3)
    theta4 = ops.adaptive_compose(gw4, atten4, torch.zeros_like(gw4))
    assert torch.allclose(theta4[..., 0], gw4[..., 0] * atten4[0])


def test_l1_drift():
    pairs = [(torch.tensor([1.0, -2.0]), torch.tensor([0.0, 0.0]))]
    assert ops.l1_drift(pairs).item() == pytest.approx(3.0)


def test_kl_distance_matches_reference_formula():
    a = torch.randn(1, 16)
    b = torch.randn(1, 16)
    d = ops.kl_distance(a, b)
    manual = F.kl_div(F.log_softmax(a, -1), F.softmax(b, -1), reduction="sum")
    assert torch.allclose(d, manual)
