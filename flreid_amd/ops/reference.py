"""Pure-PyTorch reference implementations of every flreid op.

These are the numerics ground truth: the HIP/CDNA4 kernels in
flreid_amd/ops/csrc are validated against these in fp32 (tests/test_ops_*.py),
and they double as the CPU execution path.  Semantics follow the reference
repo exactly where a counterpart exists (cited per function).
"""

from __future__ import annotations

from typing import Dict, Iterable, Optional, Tuple

import torch
import torch.nn.functional as F


# ---------------------------------------------------------------------------
# distances (ref:tools/distance.py)
# ---------------------------------------------------------------------------

def pairwise_sqeuclidean(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Squared euclidean pairwise distance matrix [m, n].

    ref:tools/distance.py:9-16 (‖a‖² + ‖b‖² − 2·a·bᵀ via addmm).
    """
    m, n = a.size(0), b.size(0)
    aa = a.pow(2).sum(dim=1, keepdim=True).expand(m, n)
    bb = b.pow(2).sum(dim=1, keepdim=True).expand(n, m).t()
    return torch.addmm(aa + bb, a, b.t(), beta=1, alpha=-2)


def pairwise_cosine_distance(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """1 − cos similarity on L2-normalised rows (ref:tools/distance.py:19-30)."""
    a = F.normalize(a, p=2, dim=1)
    b = F.normalize(b, p=2, dim=1)
    return 1.0 - a @ b.t()


def kl_distance(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """KL(softmax(b) ‖ softmax(a)) summed — FedSTIL task-token affinity
    (ref:tools/distance.py:33-36)."""
    return F.kl_div(F.log_softmax(a, dim=-1), F.softmax(b, dim=-1), reduction="sum")


def l2_normalize(x: torch.Tensor, dim: int = 1, eps: float = 1e-12) -> torch.Tensor:
    """Row-wise L2 normalisation (feature post-processing,
    ref:methods/fedavg.py:158-168)."""
    return F.normalize(x, p=2, dim=dim, eps=eps)


# ---------------------------------------------------------------------------
# losses (ref:criterions/)
# ---------------------------------------------------------------------------

def ce_label_smooth(score: torch.Tensor, target: torch.Tensor,
                    epsilon: float = 0.1) -> torch.Tensor:
    """Label-smoothed CE: mean over batch, summed over classes
    (ref:criterions/cross_entropy.py:29-41; one-hot stays on device here).
    """
    num_classes = score.size(1)
    log_probs = F.log_softmax(score, dim=1)
    with torch.no_grad():
        tgt = torch.zeros_like(log_probs).scatter_(1, target.view(-1, 1), 1.0)
        tgt = (1.0 - epsilon) * tgt + epsilon / num_classes
    return (-tgt * log_probs).mean(0).sum()


def _softmax_weights(dist: torch.Tensor, mask: torch.Tensor) -> torch.Tensor:
    max_v = torch.max(dist * mask, dim=1, keepdim=True)[0]
    diff = dist - max_v
    z = torch.sum(torch.exp(diff) * mask, dim=1, keepdim=True) + 1e-6
    return torch.exp(diff) * mask / z


def triplet_loss(feature: torch.Tensor, target: torch.Tensor,
                 margin: Optional[float] = 0.3, norm_feat: bool = False,
                 hard_mining: bool = True) -> torch.Tensor:
    """Batch-all triplet loss with batch-hard or softmax-weighted mining
    (ref:criterions/triplet_loss.py:12-127)."""
    if norm_feat:
        dist = pairwise_cosine_distance(feature, feature)
    else:
        dist = pairwise_sqeuclidean(feature, feature)

    n = dist.size(0)
    is_pos = target.view(n, 1).eq(target.view(1, n)).to(dist.dtype)
    is_neg = 1.0 - is_pos

    if hard_mining:
        dist_ap = torch.max(dist * is_pos, dim=1)[0]
        dist_an = torch.min(dist * is_neg + is_pos * 1e9, dim=1)[0]
    else:
        w_ap = _softmax_weights(dist * is_pos, is_pos)
        w_an = _softmax_weights(-dist * is_neg, is_neg)
        dist_ap = torch.sum(dist * is_pos * w_ap, dim=1)
        dist_an = torch.sum(dist * is_neg * w_an, dim=1)

    y = torch.ones_like(dist_an)
    if margin is not None and margin > 0:
        return F.margin_ranking_loss(dist_an, dist_ap, y, margin=margin)
    loss = F.soft_margin_loss(dist_an - dist_ap, y)
    if loss == float("inf"):
        loss = F.margin_ranking_loss(dist_an, dist_ap, y, margin=0.3)
    return loss


def icarl_distill_loss(score: torch.Tensor, target: torch.Tensor,
                       prev_logits: torch.Tensor) -> torch.Tensor:
    """iCaRL distillation step: classification BCE vs one-hot plus
    distillation BCE vs sigmoid of the stored pre-update logits over the
    first P (old-class) columns (ref:methods/icarl.py:216-236)."""
    from flreid_amd.tools.utils import get_one_hot

    p = prev_logits.shape[1]
    clf = F.binary_cross_entropy_with_logits(
        score.float(), get_one_hot(target, score.shape[1]))
    distill = F.binary_cross_entropy_with_logits(
        score[:, :p].float(), torch.sigmoid(prev_logits.float()))
    return clf + distill


def kd_loss(logits_student: torch.Tensor, logits_teacher: torch.Tensor,
            temperature: float = 4.0) -> torch.Tensor:
    """Temperature-scaled KL distillation (ref:criterions/kd_loss.py:10-27)."""
    t = temperature
    p_s = F.log_softmax(logits_student / t, dim=1)
    p_t = F.softmax(logits_teacher / t, dim=1)
    return F.kl_div(p_s, p_t, reduction="sum") * (t * t) / logits_student.size(0)


# ---------------------------------------------------------------------------
# lifelong-method math (ref:methods/ewc.py, mas.py, fedcurv.py, fedprox.py)
# ---------------------------------------------------------------------------

def importance_update(importance: Dict[str, torch.Tensor],
                      grads: Dict[str, torch.Tensor],
                      mode: str = "sq", scale: float = 1.0) -> None:
    """In-place accumulate per-parameter importance.

    mode='sq'  : F += g²·scale   (EWC Fisher, ref:methods/ewc.py:56-78)
    mode='abs' : F += |g|·scale  (MAS importance, ref:methods/mas.py:72-74)
    """
    for n, g in grads.items():
        if g is None:
            continue
        if mode == "sq":
            importance[n] += (g.detach().float() ** 2) * scale
        elif mode == "abs":
            importance[n] += g.detach().float().abs() * scale
        else:
            raise ValueError(mode)


def quadratic_penalty(params: Dict[str, torch.Tensor],
                      anchors: Dict[str, torch.Tensor],
                      importance: Optional[Dict[str, torch.Tensor]] = None,
                      ) -> torch.Tensor:
    """Σ F·(p − p_anchor)² — EWC/MAS/FedCurv penalty (ref:methods/ewc.py:80-85)
    and, with importance=None, the FedProx proximal term
    (ref:methods/fedprox.py:52-57)."""
    total = None
    for n, p in params.items():
        if n not in anchors:
            continue
        d = (p - anchors[n]) ** 2
        if importance is not None:
            d = importance[n] * d
        s = d.sum()
        total = s if total is None else total + s
    if total is None:
        total = torch.zeros((), device=next(iter(params.values())).device)
    return total


def l1_drift(pairs: Iterable[Tuple[torch.Tensor, torch.Tensor]]) -> torch.Tensor:
    """Σ |p − p₀| — FedSTIL sparsity/drift regulariser
    (ref:methods/fedstil.py:639-644)."""
    total = None
    for p, p0 in pairs:
        s = (p - p0).abs().sum()
        total = s if total is None else total + s
    return total


class _L1DriftFusedFn(torch.autograd.Function):
    """Horizontally-fused Σ|p − p₀| over a parameter list via _foreach ops
    (one fused sub + one fused L1-norm instead of 3 kernels per tensor;
    backward: grad·sign(p − p₀), matching torch.abs' subgradient-0-at-0)."""

    @staticmethod
    def forward(ctx, n_params, *tensors):
        params = list(tensors[:n_params])
        anchors = list(tensors[n_params:])
        diffs = torch._foreach_sub(params, anchors)
        ctx.n_params = n_params
        ctx.save_for_backward(*diffs)
        norms = torch._foreach_norm(diffs, 1)
        return torch.stack(norms).sum()

    @staticmethod
    def backward(ctx, grad_out):
        diffs = list(ctx.saved_tensors)
        signs = torch._foreach_sign(diffs)
        torch._foreach_mul_(signs, grad_out)
        return (None, *signs, *([None] * ctx.n_params))


def l1_drift_fused(pairs) -> torch.Tensor:
    pairs = list(pairs)
    params = [p for p, _ in pairs]
    anchors = [p0.detach() for _, p0 in pairs]
    return _L1DriftFusedFn.apply(len(params), *params, *anchors)


# ---------------------------------------------------------------------------
# adaptive-layer composition (FedSTIL; ref:methods/fedstil.py:84-92)
# ---------------------------------------------------------------------------

def adaptive_compose(global_weight: torch.Tensor, atten: torch.Tensor,
                     adaptive_weight: torch.Tensor) -> torch.Tensor:
    """θ = atten ⊙ W_glob + W_adapt, with atten broadcast over the LAST dim of
    the weight (atten.shape == (W.shape[-1],) — ref:methods/fedstil.py:66-67,84-92)."""
    return atten * global_weight + adaptive_weight


# ---------------------------------------------------------------------------
# CMC / mAP evaluation (ref:tools/evaluate.py)
# ---------------------------------------------------------------------------

@torch.no_grad()
def cmc_map(query_features: torch.Tensor, query_labels: torch.Tensor,
            gallery_features: torch.Tensor, gallery_labels: torch.Tensor,
            query_camera_labels: Optional[torch.Tensor] = None,
            gallery_camera_labels: Optional[torch.Tensor] = None,
            ) -> Tuple[torch.Tensor, float]:
    """Vectorised CMC curve + mAP.

    Replaces the reference's per-query python loop (ref:tools/evaluate.py:103-142)
    with one Q×G similarity GEMM, one sort, and vectorised rank statistics.
    Semantics match evaluate_with_index (ref:tools/evaluate.py:36-84):
      - junk = (gallery label == -1) ∪ (same id ∧ same camera)   [cam-aware mode]
      - good = same id (∧ different camera in cam-aware mode)
      - CMC counts the first good hit's junk-free rank
      - AP is the interpolated average precision over good hits
      - queries with no good hit contribute 0 but remain in the denominator
        (ref:tools/evaluate.py:137-142)
    Ties in similarity are broken by descending gallery index (matching
    np.argsort(sim)[::-1] stable-sort reversal).
    """
    sims = query_features.float() @ gallery_features.float().t()     # [Q, G]
    return cmc_map_from_sims(sims, query_labels, gallery_labels,
                             query_camera_labels, gallery_camera_labels)


@torch.no_grad()
def cmc_map_from_sims(sims: torch.Tensor, query_labels: torch.Tensor,
                      gallery_labels: torch.Tensor,
                      query_camera_labels: Optional[torch.Tensor] = None,
                      gallery_camera_labels: Optional[torch.Tensor] = None,
                      ) -> Tuple[torch.Tensor, float]:
    q, g = sims.shape
    device = sims.device
    ql = query_labels.to(device).view(q, 1)
    gl = gallery_labels.to(device).view(1, g)

    same_id = ql.eq(gl)                                              # [Q, G]
    if query_camera_labels is not None and gallery_camera_labels is not None:
        qc = query_camera_labels.to(device).view(q, 1)
        gc = gallery_camera_labels.to(device).view(1, g)
        same_cam = qc.eq(gc)
        junk = (gl.expand(q, g) == -1) | (same_id & same_cam)
        good = same_id & ~same_cam & (gl.expand(q, g) != -1)
    else:
        junk = torch.zeros_like(same_id)
        good = same_id

    # np.argsort(sim)[::-1] semantics: ascending stable sort then reverse —
    # on ties the HIGHER gallery index ranks first (ref:tools/evaluate.py:129)
    order = torch.flip(torch.argsort(sims, dim=1, descending=False, stable=True),
                       dims=[1])                                      # [Q, G]
    good_sorted = good.gather(1, order)
    valid_sorted = (~junk).gather(1, order)

    # junk-free rank of each ranked slot (0-based among valid slots)
    rank_in_valid = valid_sorted.long().cumsum(dim=1) - 1             # [Q, G]
    hit = good_sorted & valid_sorted

    # ---- CMC ----
    big = g + 1
    first_rank = torch.where(hit, rank_in_valid, torch.full_like(rank_in_valid, big))
    r0 = first_rank.min(dim=1)[0]                                     # [Q]
    has_good = r0 < big
    cmc = torch.zeros(g, dtype=torch.float64, device=device)
    if has_good.any():
        hist = torch.bincount(r0[has_good], minlength=g + 1)[:g].to(torch.float64)
        cmc = hist.cumsum(0)
    cmc = cmc / q

    # ---- mAP ----
    hit_idx = hit.long().cumsum(dim=1) - 1                            # i (0-based hit counter)
    r = rank_in_valid.to(torch.float64)
    i = hit_idx.to(torch.float64)
    precision = (i + 1.0) / (r + 1.0)
    old_precision = torch.where(r > 0, i / r.clamp(min=1), torch.ones_like(r))
    contrib = torch.where(hit, (precision + old_precision) * 0.5, torch.zeros_like(r))
    n_good = hit.sum(dim=1).to(torch.float64)                         # [Q]
    ap = contrib.sum(dim=1) / n_good.clamp(min=1)
    ap = torch.where(has_good, ap, torch.zeros_like(ap))
    return cmc.cpu(), float(ap.sum().item() / q)


# ---------------------------------------------------------------------------
# Swin window attention (ref:models/swin_transformer.py:255-286)
# ---------------------------------------------------------------------------

def window_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                     bias: torch.Tensor, mask: Optional[torch.Tensor],
                     scale: float, dropout=None) -> torch.Tensor:
    """Per-window MHSA: softmax(q·kᵀ·scale + rel-pos-bias [+ shift mask])·v.

    q/k/v: [B·nW, H, N, D]; bias: [H, N, N]; mask: [nW, N, N] or None.
    Eager composition — the fused CDNA4 kernel (K3) takes over on GPU.
    """
    bn, h, n, _d = q.shape
    attn = (q * scale) @ k.transpose(-2, -1) + bias.unsqueeze(0).to(q.dtype)
    if mask is not None:
        nw = mask.shape[0]
        attn = attn.view(bn // nw, nw, h, n, n) \
            + mask.unsqueeze(1).unsqueeze(0).to(attn.dtype)
        attn = attn.view(bn, h, n, n)
    attn = F.softmax(attn.float(), dim=-1).to(q.dtype)
    if dropout is not None:
        attn = dropout(attn)
    return attn @ v
