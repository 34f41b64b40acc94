"""Dispatch layer for flreid ops.

CPU tensors  -> pure-PyTorch reference implementations (ops/reference.py).
CUDA tensors -> hand-written HIP/CDNA4 kernels from the in-tree extension
                flreid_amd/ops/_flreid_hip.so (sources in ops/csrc, built by
                `python -m flreid_amd.ops.build` / __graft_entry__.build()).

On a GPU box the HIP path is mandatory: if the extension is missing the op
raises instead of silently falling back to eager PyTorch (set
FLREID_ALLOW_EAGER=1 to debug).  Numerics ground truth for every kernel is
the fp32 reference implementation (tests/test_ops_gpu.py).
"""

from __future__ import annotations

import os
from typing import Dict, Optional

import torch

from flreid_amd.ops import reference as ref

_EXT = None
_EXT_ERR: Optional[str] = None

_F32, _BF16 = 0, 1


def _load_extension():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from flreid_amd.ops import _flreid_hip  # built in-tree

        _EXT = _flreid_hip
    except Exception as e:  # pragma: no cover - GPU boxes only
        _EXT_ERR = f"{type(e).__name__}: {e}"
        _EXT = None
    return _EXT


def extension_available() -> bool:
    return _load_extension() is not None


def _ext_or_raise(opname: str):
    ext = _load_extension()
    if ext is not None:
        return ext
    if os.environ.get("FLREID_ALLOW_EAGER", "0") == "1":
        return None
    raise RuntimeError(
        f"flreid HIP extension missing for GPU op '{opname}' ({_EXT_ERR}). "
        "Build it with `python -m flreid_amd.ops.build` "
        "or set FLREID_ALLOW_EAGER=1 to debug with eager PyTorch."
    )


def _stream() -> int:
    return torch.cuda.current_stream().cuda_stream


def _dt(t: torch.Tensor) -> int:
    if t.dtype == torch.float32:
        return _F32
    if t.dtype == torch.bfloat16:
        return _BF16
    raise TypeError(f"unsupported dtype {t.dtype}")


# ---------------------------------------------------------------------------
# distances / similarity (K8)
# ---------------------------------------------------------------------------

def _pairwise_gpu(ext, a: torch.Tensor, b: torch.Tensor, mode: int) -> torch.Tensor:
    a = a.contiguous().float()
    b = b.contiguous().float()
    m, n, d = a.shape[0], b.shape[0], a.shape[1]
    out = torch.empty(m, n, device=a.device, dtype=torch.float32)
    if mode == 2:
        aa = torch.empty(m, device=a.device, dtype=torch.float32)
        bb = torch.empty(n, device=a.device, dtype=torch.float32)
        ext.rowsq(a.data_ptr(), aa.data_ptr(), m, d, _stream())
        ext.rowsq(b.data_ptr(), bb.data_ptr(), n, d, _stream())
    else:
        aa = bb = out  # unused
    ext.pairwise(a.data_ptr(), b.data_ptr(), aa.data_ptr(), bb.data_ptr(),
                 out.data_ptr(), m, n, d, mode, _stream())
    return out


def similarity_matrix(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """A·Bᵀ — the eval GEMM."""
    if a.is_cuda and not a.requires_grad and not b.requires_grad:
        ext = _ext_or_raise("pairwise")
        if ext is not None:
            return _pairwise_gpu(ext, a, b, 0)
    return a.float() @ b.float().t()


def pairwise_sqeuclidean(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    if a.is_cuda and not a.requires_grad and not b.requires_grad:
        ext = _ext_or_raise("pairwise")
        if ext is not None:
            return _pairwise_gpu(ext, a, b, 2)
    return ref.pairwise_sqeuclidean(a, b)


def pairwise_cosine_distance(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    if a.is_cuda and not a.requires_grad and not b.requires_grad:
        ext = _ext_or_raise("pairwise")
        if ext is not None:
            return _pairwise_gpu(ext, l2_normalize(a.float()),
                                 l2_normalize(b.float()), 1)
    return ref.pairwise_cosine_distance(a, b)


# ---------------------------------------------------------------------------
# rowwise L2 normalize (K11)
# ---------------------------------------------------------------------------

def l2_normalize(x: torch.Tensor, dim: int = 1) -> torch.Tensor:
    if (x.is_cuda and x.dim() == 2 and dim in (1, -1)
            and not x.requires_grad and x.dtype in (torch.float32, torch.bfloat16)):
        ext = _ext_or_raise("l2norm_rows")
        if ext is not None:
            x = x.contiguous()
            y = torch.empty_like(x)
            ext.l2norm_rows(x.data_ptr(), y.data_ptr(), x.shape[0], x.shape[1],
                            _dt(x), 1e-12, _stream())
            return y
    return ref.l2_normalize(x, dim=dim)


# ---------------------------------------------------------------------------
# fused label-smooth CE (K6)
# ---------------------------------------------------------------------------

class _CeLabelSmoothFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, score, target, epsilon, ext):
        score_c = score.contiguous()
        B, C = score_c.shape
        row_loss = torch.empty(B, device=score.device, dtype=torch.float32)
        grad = torch.empty_like(score_c)
        ext.ce_smooth(score_c.data_ptr(), target.contiguous().data_ptr(),
                      row_loss.data_ptr(), grad.data_ptr(), B, C,
                      _dt(score_c), float(epsilon), _stream())
        ctx.save_for_backward(grad)
        return row_loss.sum() * (1.0 / B)

    @staticmethod
    def backward(ctx, grad_out):
        (grad,) = ctx.saved_tensors
        return grad * grad_out, None, None, None


def ce_label_smooth(score: torch.Tensor, target: torch.Tensor,
                    epsilon: float = 0.1) -> torch.Tensor:
    if score.is_cuda and score.dim() == 2:
        ext = _ext_or_raise("ce_smooth")
        if ext is not None:
            return _CeLabelSmoothFn.apply(score, target.long(), epsilon, ext)
    return ref.ce_label_smooth(score, target, epsilon)


# ---------------------------------------------------------------------------
# FedSTIL composition (K1/K2 prologue, standalone form)
# ---------------------------------------------------------------------------

class _ComposeFn(torch.autograd.Function):
    """theta = atten ⊙_lastdim gw + aw.  gw/atten frozen in FedSTIL; grad
    flows to aw (identity) and, when atten trains (fedstil-atten), reduces
    over all dims but the last."""

    @staticmethod
    def forward(ctx, gw, atten, aw, ext):
        gw_c, aw_c = gw.contiguous(), aw.contiguous()
        at = atten.detach().contiguous().float()
        out = torch.empty_like(gw_c)
        ext.compose(gw_c.data_ptr(), at.data_ptr(), aw_c.data_ptr(),
                    out.data_ptr(), gw_c.numel(), gw_c.shape[-1], _dt(gw_c),
                    _stream())
        ctx.atten_requires = atten.requires_grad
        if ctx.atten_requires:
            ctx.save_for_backward(gw_c)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        g_atten = None
        if ctx.atten_requires:
            (gw,) = ctx.saved_tensors
            g_atten = (grad_out * gw).reshape(-1, gw.shape[-1]).sum(dim=0)
        return None, g_atten, grad_out, None


def adaptive_compose(global_weight: torch.Tensor, atten: torch.Tensor,
                     adaptive_weight: torch.Tensor) -> torch.Tensor:
    if (global_weight.is_cuda and not global_weight.requires_grad
            and global_weight.dtype in (torch.float32, torch.bfloat16)
            and adaptive_weight.dtype == global_weight.dtype):
        ext = _ext_or_raise("compose")
        if ext is not None:
            return _ComposeFn.apply(global_weight, atten, adaptive_weight, ext)
    return ref.adaptive_compose(global_weight, atten, adaptive_weight)


# ---------------------------------------------------------------------------
# EWC/MAS importance accumulation (K9)
# ---------------------------------------------------------------------------

def importance_update(importance: Dict[str, torch.Tensor],
                      grads: Dict[str, torch.Tensor], mode: str = "sq",
                      scale: float = 1.0) -> None:
    first = next(iter(importance.values()), None)
    if first is not None and first.is_cuda:
        ext = _ext_or_raise("importance")
        if ext is not None:
            for n, g in grads.items():
                if g is None:
                    continue
                F = importance[n]
                ext.importance(F.data_ptr(), g.contiguous().data_ptr(),
                               g.numel(), _dt(g), mode == "sq", scale,
                               _stream())
            return
    for n, g in grads.items():
        if g is None:
            continue
        if mode == "sq":
            importance[n] += (g.detach().float() ** 2) * scale
        else:
            importance[n] += g.detach().float().abs() * scale


# ---------------------------------------------------------------------------
# pass-throughs (small tensors / composed autograd paths)
# ---------------------------------------------------------------------------

kl_distance = ref.kl_distance
quadratic_penalty = ref.quadratic_penalty
l1_drift_fused = ref.l1_drift_fused
# l1_drift is defined below: fused HIP multi-tensor path on GPU


class _KdLossFn(torch.autograd.Function):
    """Fused temperature-softmax KL distillation (K7, kd.hip): one kernel
    computes loss + dL/dz_student; the eager chain is 5 kernels plus the
    autograd graph."""

    @staticmethod
    def forward(ctx, zs, zt, temperature, ext):
        zs_c = zs.detach().float().contiguous()
        zt_c = zt.detach().float().contiguous()
        B, C = zs_c.shape
        row_loss = torch.empty(B, device=zs.device, dtype=torch.float32)
        grad = torch.empty_like(zs_c)
        ext.kd_fwd(zs_c.data_ptr(), zt_c.data_ptr(), row_loss.data_ptr(),
                   grad.data_ptr(), B, C, float(temperature), _stream())
        ctx.save_for_backward(grad)
        ctx.zs_dtype = zs.dtype
        return row_loss.sum()

    @staticmethod
    def backward(ctx, grad_out):
        (grad,) = ctx.saved_tensors
        return (grad * grad_out).to(ctx.zs_dtype), None, None, None


def kd_loss(logits_student: torch.Tensor, logits_teacher: torch.Tensor,
            temperature: float = 4.0) -> torch.Tensor:
    if logits_student.is_cuda and logits_student.dim() == 2:
        ext = _ext_or_raise("kd_fwd")
        if ext is not None:
            return _KdLossFn.apply(logits_student, logits_teacher,
                                   temperature, ext)
    return ref.kd_loss(logits_student, logits_teacher, temperature)


class _IcarlDistillFn(torch.autograd.Function):
    """Fused iCaRL distillation step (K7, kd.hip): BOTH BCE-with-logits
    losses (one-hot classification + sigmoid-teacher distillation on the
    first P columns) and the combined gradient in ONE pass
    (ref:methods/icarl.py:216-236)."""

    @staticmethod
    def forward(ctx, score, target, prev, ext):
        score_c = score.detach().float().contiguous()
        prev_c = prev.detach().float().contiguous()
        B, C = score_c.shape
        P = prev_c.shape[1]
        row_loss = torch.empty(B, device=score.device, dtype=torch.float32)
        grad = torch.empty_like(score_c)
        ext.icarl_distill(score_c.data_ptr(),
                          target.long().contiguous().data_ptr(),
                          prev_c.data_ptr(), row_loss.data_ptr(),
                          grad.data_ptr(), B, C, P, _stream())
        ctx.save_for_backward(grad)
        ctx.score_dtype = score.dtype
        return row_loss.sum()

    @staticmethod
    def backward(ctx, grad_out):
        (grad,) = ctx.saved_tensors
        return (grad * grad_out).to(ctx.score_dtype), None, None, None


def icarl_distill_loss(score: torch.Tensor, target: torch.Tensor,
                       prev_logits: torch.Tensor) -> torch.Tensor:
    """clf BCE(score, onehot(target)) + distill BCE(score[:, :P],
    sigmoid(prev_logits)) — fused on GPU, eager reference elsewhere."""
    if (score.is_cuda and score.dim() == 2
            and prev_logits.shape[1] <= score.shape[1]):
        ext = _ext_or_raise("icarl_distill")
        if ext is not None:
            return _IcarlDistillFn.apply(score, target, prev_logits, ext)
    return ref.icarl_distill_loss(score, target, prev_logits)


class _TripletHardFn(torch.autograd.Function):
    """Fused batch-hard euclidean triplet (margin mode) — K5."""

    @staticmethod
    def forward(ctx, feature, target, margin, ext):
        f = feature.detach().contiguous().float()
        n, d = f.shape
        norms = torch.empty(n, device=f.device, dtype=torch.float32)
        ext.rowsq(f.data_ptr(), norms.data_ptr(), n, d, _stream())
        row_loss = torch.empty(n, device=f.device, dtype=torch.float32)
        p_idx = torch.empty(n, device=f.device, dtype=torch.int32)
        n_idx = torch.empty(n, device=f.device, dtype=torch.int32)
        ext.triplet_fwd(f.data_ptr(), norms.data_ptr(),
                        target.contiguous().data_ptr(), row_loss.data_ptr(),
                        p_idx.data_ptr(), n_idx.data_ptr(), n, d,
                        float(margin), _stream())
        ctx.save_for_backward(f, row_loss, p_idx, n_idx)
        ctx.ext = ext
        return row_loss.mean()

    @staticmethod
    def backward(ctx, grad_out):
        f, row_loss, p_idx, n_idx = ctx.saved_tensors
        n, d = f.shape
        grad = torch.zeros_like(f)
        # coeff = go * (1/N) * 2  (hinge mean + d(dist)/d(f) factor)
        coeff = float(grad_out) * 2.0 / n
        ctx.ext.triplet_bwd(f.data_ptr(), row_loss.data_ptr(),
                            p_idx.data_ptr(), n_idx.data_ptr(),
                            grad.data_ptr(), n, d, coeff, _stream())
        return grad, None, None, None


def triplet_loss(feature: torch.Tensor, target: torch.Tensor,
                 margin: Optional[float] = 0.3, norm_feat: bool = False,
                 hard_mining: bool = True) -> torch.Tensor:
    if (feature.is_cuda and hard_mining and not norm_feat
            and margin is not None and margin > 0 and feature.dim() == 2):
        ext = _ext_or_raise("triplet_fwd")
        if ext is not None:
            return _TripletHardFn.apply(feature.float(), target.long(),
                                        margin, ext)
    return ref.triplet_loss(feature, target, margin, norm_feat, hard_mining)


def cmc_map(query_features, query_labels, gallery_features, gallery_labels,
            query_camera_labels=None, gallery_camera_labels=None,
            query_chunk: int = 2048):
    """On GPU the Q×G similarity runs on the MFMA pairwise kernel; the rank
    statistics stay in device-side torch ops.  Queries are processed in
    chunks so million-image galleries (iCaRL config — K8/288 GB HBM) never
    materialise more than [chunk, G] similarities at once."""
    if not query_features.is_cuda:
        return ref.cmc_map(query_features, query_labels, gallery_features,
                           gallery_labels, query_camera_labels,
                           gallery_camera_labels)
    q = query_features.shape[0]
    g = gallery_features.shape[0]
    if q <= query_chunk:
        sims = similarity_matrix(query_features, gallery_features)
        return ref.cmc_map_from_sims(sims, query_labels, gallery_labels,
                                     query_camera_labels, gallery_camera_labels)
    total_cmc = torch.zeros(g, dtype=torch.float64)
    total_ap = 0.0
    for s0 in range(0, q, query_chunk):
        s1 = min(q, s0 + query_chunk)
        sims = similarity_matrix(query_features[s0:s1], gallery_features)
        cmc, mAP = ref.cmc_map_from_sims(
            sims, query_labels[s0:s1], gallery_labels,
            query_camera_labels[s0:s1] if query_camera_labels is not None else None,
            gallery_camera_labels)
        n = s1 - s0
        total_cmc += cmc * n
        total_ap += mAP * n
    return total_cmc / q, total_ap / q


def _window_attn_fwd_raw(ext, q, k, v, bias, mask, scale):
    bw, h, n, d = q.shape
    qc, kc, vc = q.contiguous(), k.contiguous(), v.contiguous()
    bias_c = bias.detach().contiguous().float()
    if mask is not None:
        mask_c = mask.detach().contiguous().float()
        nw = mask_c.shape[0]
        mask_ptr = mask_c.data_ptr()
    else:
        mask_c, nw, mask_ptr = None, 1, 0
    out = torch.empty_like(qc)
    ext.window_attn_fwd(qc.data_ptr(), kc.data_ptr(), vc.data_ptr(),
                        bias_c.data_ptr(), mask_ptr, out.data_ptr(),
                        bw, h, n, d, nw, float(scale), _dt(qc), _stream())
    return out, qc, kc, vc, bias_c, mask_c


class _WindowAttnFn(torch.autograd.Function):
    """Fused Swin window MHSA fwd+bwd (K3, window_attn.hip) — the TRAINING
    path: the backward recomputes S→P per (window, head) in LDS and emits
    dQ/dK/dV plus dS; the relative-position-bias gradient is the window
    reduction of dS (the bias-table gather backward stays in autograd)."""

    @staticmethod
    def forward(ctx, q, k, v, bias, mask, scale):
        ext = _ext_or_raise("window_attn_fwd")
        out, qc, kc, vc, bias_c, mask_c = _window_attn_fwd_raw(
            ext, q.detach(), k.detach(), v.detach(), bias, mask, scale)
        ctx.save_for_backward(qc, kc, vc, bias_c)
        ctx.mask_c = mask_c
        ctx.scale = float(scale)
        ctx.bias_requires = bias.requires_grad
        return out

    @staticmethod
    def backward(ctx, grad_out):
        ext = _ext_or_raise("window_attn_bwd")
        qc, kc, vc, bias_c = ctx.saved_tensors
        mask_c = ctx.mask_c
        bw, h, n, d = qc.shape
        nw = mask_c.shape[0] if mask_c is not None else 1
        go = grad_out.contiguous().to(qc.dtype)
        dq = torch.empty_like(qc)
        dk = torch.empty_like(kc)
        dv = torch.empty_like(vc)
        ds = torch.empty(bw, h, n, n, device=qc.device, dtype=torch.float32)
        ext.window_attn_bwd(qc.data_ptr(), kc.data_ptr(), vc.data_ptr(),
                            bias_c.data_ptr(),
                            mask_c.data_ptr() if mask_c is not None else 0,
                            go.data_ptr(), dq.data_ptr(), dk.data_ptr(),
                            dv.data_ptr(), ds.data_ptr(), bw, h, n, d, nw,
                            ctx.scale, _dt(qc), _stream())
        dbias = ds.sum(dim=0) if ctx.bias_requires else None
        return dq, dk, dv, dbias, None, None


def window_attention(q, k, v, bias, mask, scale, dropout=None):
    """Swin window MHSA (K3): fused HIP kernels on GPU — eval forward AND
    the training fwd+bwd (dropout-free, the Swin-ReID default); eager
    composition on CPU or when attention dropout is active."""
    drop_p = 0.0
    if dropout is not None:
        drop_p = float(getattr(dropout, "p", 0.0))
    if (q.is_cuda and q.shape[-2] <= 64 and q.shape[-1] <= 64
            and drop_p == 0.0 and extension_available()):
        if not torch.is_grad_enabled():
            ext = _ext_or_raise("window_attn_fwd")
            out, *_ = _window_attn_fwd_raw(ext, q, k, v, bias, mask, scale)
            return out
        return _WindowAttnFn.apply(q, k, v, bias, mask, scale)
    return ref.window_attention(q, k, v, bias, mask, scale, dropout)


class _PatchMergeLNFn(torch.autograd.Function):
    """Fused Swin PatchMerging gather + LayerNorm (K4, patch_merge.hip):
    the 2×2 strided concat tensor never materialises; forward emits the
    normalized [B, L/4, 4C] rows the reduction GEMM consumes, backward
    scatters dx and reduces per-block dgamma/dbeta partials."""

    @staticmethod
    def forward(ctx, x, gamma, beta, H, W, eps):
        ext = _ext_or_raise("patch_merge_ln_fwd")
        x_c = x.detach().contiguous()
        b, L, c = x_c.shape
        rows = b * (H // 2) * (W // 2)
        y = torch.empty(b, L // 4, 4 * c, device=x.device, dtype=x.dtype)
        mean = torch.empty(rows, device=x.device, dtype=torch.float32)
        rstd = torch.empty(rows, device=x.device, dtype=torch.float32)
        gamma_c = gamma.detach().float().contiguous()
        beta_c = beta.detach().float().contiguous()
        ext.patch_merge_ln_fwd(x_c.data_ptr(), gamma_c.data_ptr(),
                               beta_c.data_ptr(), y.data_ptr(),
                               mean.data_ptr(), rstd.data_ptr(), rows, c, H,
                               W, float(eps), _dt(x_c), _stream())
        ctx.save_for_backward(x_c, gamma_c, mean, rstd)
        ctx.hw = (H, W)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _ext_or_raise("patch_merge_ln_bwd")
        x_c, gamma_c, mean, rstd = ctx.saved_tensors
        H, W = ctx.hw
        b, L, c = x_c.shape
        rows = b * (H // 2) * (W // 2)
        nblk = (rows + 3) // 4
        dy_c = dy.contiguous().to(x_c.dtype)
        dx = torch.empty_like(x_c)
        dg_part = torch.empty(nblk, 4 * c, device=x_c.device,
                              dtype=torch.float32)
        db_part = torch.empty(nblk, 4 * c, device=x_c.device,
                              dtype=torch.float32)
        ext.patch_merge_ln_bwd(x_c.data_ptr(), gamma_c.data_ptr(),
                               dy_c.data_ptr(), mean.data_ptr(),
                               rstd.data_ptr(), dx.data_ptr(),
                               dg_part.data_ptr(), db_part.data_ptr(), rows,
                               c, H, W, _dt(x_c), _stream())
        dgamma = dg_part.sum(0) if ctx.needs_input_grad[1] else None
        dbeta = db_part.sum(0) if ctx.needs_input_grad[2] else None
        return dx, dgamma, dbeta, None, None, None


def patch_merge_ln(x: torch.Tensor, gamma: torch.Tensor, beta: torch.Tensor,
                   H: int, W: int, eps: float = 1e-5) -> Optional[torch.Tensor]:
    """Fused PatchMerging gather+LN; None when out of regime (caller runs
    the eager concat + LayerNorm)."""
    if (not x.is_cuda or x.dim() != 3 or H % 2 or W % 2
            or 4 * x.shape[2] > 3072 or x.shape[1] != H * W
            or not extension_available()):
        return None
    if x.dtype not in (torch.float32, torch.bfloat16):
        return None
    return _PatchMergeLNFn.apply(x, gamma, beta, H, W, eps)


def adaptive_linear_fwd(x: torch.Tensor, gw: torch.Tensor,
                        atten: Optional[torch.Tensor],
                        aw: Optional[torch.Tensor],
                        bias: Optional[torch.Tensor],
                        split_layout: int = 1) -> torch.Tensor:
    """Fused y = x · (atten⊙gw + aw)ᵀ + bias (K2) — bf16 MFMA, no-grad path.

    x bf16 [M, K]; gw/aw fp32 [N, K]; atten fp32 [K]; bias fp32 [N].
    """
    ext = _ext_or_raise("adaptive_linear_fwd")
    if ext is None:
        theta = ref.adaptive_compose(gw, atten, aw) if atten is not None else gw
        return (torch.nn.functional.linear(x.float(), theta,
                                           bias).to(x.dtype))
    m, k = x.shape
    n = gw.shape[0]
    x_c = x.contiguous()
    out = torch.empty(m, n, device=x.device, dtype=torch.bfloat16)
    ext.adaptive_linear_fwd(
        x_c.data_ptr(), gw.contiguous().data_ptr(),
        aw.contiguous().data_ptr() if aw is not None else 0,
        atten.contiguous().float().data_ptr() if atten is not None else 0,
        bias.contiguous().float().data_ptr() if bias is not None else 0,
        out.data_ptr(), m, n, k, split_layout, _stream())
    return out


class _AdaptiveLinearFn(torch.autograd.Function):
    """Autograd wrapper for the fused K2 GEMM: fwd composes θ in the weight
    fetch; bwd materialises θ once (compose kernel) for grad_x and computes
    grad_aw = gᵀ·x.  atten/gw are frozen in FedSTIL (no grads)."""

    @staticmethod
    def forward(ctx, x, gw, atten, aw, bias):
        x_bf = x.detach().to(torch.bfloat16).contiguous()
        out = adaptive_linear_fwd(x_bf, gw.detach(), atten, aw.detach(),
                                  bias.detach() if bias is not None else None)
        ctx.save_for_backward(x_bf, gw, atten, aw)
        ctx.has_bias = bias is not None
        ctx.x_dtype = x.dtype
        return out

    @staticmethod
    def backward(ctx, grad_out):
        x_bf, gw, atten, aw = ctx.saved_tensors
        g = grad_out.to(torch.bfloat16)
        ext = _load_extension()
        if ext is not None and gw.dtype == torch.float32:
            theta = compose_theta_bf16(ext, gw.contiguous(), atten,
                                       aw.contiguous())
        else:
            theta = adaptive_compose(gw, atten, aw).to(torch.bfloat16)
        grad_x = (g @ theta).to(ctx.x_dtype)
        grad_aw = (g.t() @ x_bf).to(aw.dtype)
        grad_bias = grad_out.sum(0).to(torch.float32) if ctx.has_bias else None
        return grad_x, None, None, grad_aw, grad_bias


def adaptive_linear(x: torch.Tensor, gw: torch.Tensor, atten: torch.Tensor,
                    aw: torch.Tensor, bias: Optional[torch.Tensor]):
    """Adaptive linear layer forward: fused MFMA path on GPU (frozen
    gw/atten), eager compose+linear elsewhere."""
    if (x.is_cuda and x.dim() == 2 and not gw.requires_grad
            and not atten.requires_grad and gw.shape[-1] % 32 == 0
            and atten.numel() == gw.shape[-1]      # kernel contract: atten[K]
            and extension_available()):
        return _AdaptiveLinearFn.apply(x, gw, atten, aw, bias)
    theta = adaptive_compose(gw, atten, aw)
    return torch.nn.functional.linear(x, theta, bias)


def bn_eval_2d(x: torch.Tensor, bn, relu: bool = False) -> Optional[torch.Tensor]:
    """Fused eval-mode BatchNorm2d (one coalesced pass; MIOpen's inference
    kernel measured ~0.3 TB/s on ReID shapes).  Returns None when the fused
    path does not apply (train mode, CPU, missing stats/extension)."""
    if (not x.is_cuda or x.requires_grad or bn.running_mean is None
            or x.dtype not in (torch.float32, torch.bfloat16)
            or x.dim() not in (2, 4)):
        return None
    ext = _load_extension()
    if ext is None:
        return None
    if x.dim() == 2:      # BatchNorm1d rows: NHWC layout with H·W == 1
        if not x.is_contiguous():
            x = x.contiguous()
        nhwc, hw = True, 1
    else:
        nhwc = x.is_contiguous(memory_format=torch.channels_last)
        if not nhwc and not x.is_contiguous():
            x = x.contiguous()
        hw = x.shape[2] * x.shape[3]
    out = torch.empty_like(x)
    c = x.shape[1]
    ext.bn_eval(x.data_ptr(), out.data_ptr(),
                bn.weight.detach().float().contiguous().data_ptr(),
                bn.bias.detach().float().contiguous().data_ptr(),
                bn.running_mean.float().contiguous().data_ptr(),
                bn.running_var.float().contiguous().data_ptr(),
                x.numel(), c, hw, float(bn.eps), int(nhwc), int(relu),
                _dt(x), _stream())
    return out


_DRIFT_CHUNK = 1 << 16
_DRIFT_META: Dict = {}


class _L1DriftHipFn(torch.autograd.Function):
    """Fused multi-tensor Σ|p − p₀| (ops/csrc/drift.hip): one read-only pass
    forward, one sign-write pass backward into a single flat buffer split
    into per-tensor grad views."""

    @staticmethod
    def forward(ctx, meta, n_params, *tensors):
        ext = _ext_or_raise("drift")
        ptrs, chunks, sizes, total = meta
        partials = torch.empty(chunks.shape[0], device=tensors[0].device,
                               dtype=torch.float32)
        ext.drift_fwd(ptrs.data_ptr(), chunks.data_ptr(), partials.data_ptr(),
                      chunks.shape[0], _stream())
        ctx.meta = meta
        ctx.n_params = n_params
        ctx.shapes = [t.shape for t in tensors[:n_params]]
        return partials.sum()

    @staticmethod
    def backward(ctx, grad_out):
        ext = _ext_or_raise("drift")
        ptrs, chunks, sizes, total = ctx.meta
        flat = torch.empty(total, device=grad_out.device, dtype=torch.float32)
        g = grad_out.to(torch.float32).contiguous()
        ext.drift_bwd(ptrs.data_ptr(), chunks.data_ptr(), g.data_ptr(),
                      flat.data_ptr(), chunks.shape[0], _stream())
        views = [v.view(s) for v, s in zip(flat.split(sizes), ctx.shapes)]
        return (None, None, *views, *([None] * ctx.n_params))


def _drift_meta(pairs):
    """(ptr_table, chunk_table, sizes, total) on device, cached on the
    parameter storages (in-place per-round updates keep pointers stable;
    a dispatch that rebinds storage changes the key and rebuilds)."""
    key = tuple((p.data_ptr(), a.data_ptr(), p.numel()) for p, a in pairs)
    meta = _DRIFT_META.get(key)
    if meta is not None:
        return meta
    dev = pairs[0][0].device
    ptr_rows, chunk_rows, sizes = [], [], []
    flat_off = 0
    for i, (p, a) in enumerate(pairs):
        ptr_rows.append((p.data_ptr(), a.data_ptr()))
        n = p.numel()
        sizes.append(n)
        off = 0
        while off < n:
            ln = min(_DRIFT_CHUNK, n - off)
            chunk_rows.append((i, off, ln, flat_off + off))
            off += ln
        flat_off += n
    ptrs = torch.tensor(ptr_rows, dtype=torch.int64, device=dev)
    chunks = torch.tensor(chunk_rows, dtype=torch.int32, device=dev)
    # entries are never evicted: a captured hipGraph may hold the table
    # pointers for its lifetime, and each entry is only ~KBs
    meta = (ptrs, chunks, sizes, flat_off)
    _DRIFT_META[key] = meta
    return meta


def l1_drift(pairs) -> torch.Tensor:
    """Σ |p − p₀| over drift pairs — fused HIP multi-tensor path on GPU
    (fp32, contiguous), _foreach fallback elsewhere."""
    pairs = [(p, a.detach()) for p, a in pairs]
    if (pairs and pairs[0][0].is_cuda and extension_available()
            and all(p.dtype == torch.float32 and p.is_contiguous()
                    and a.dtype == torch.float32 and a.is_contiguous()
                    and p.numel() == a.numel() for p, a in pairs)):
        meta = _drift_meta(pairs)
        params = [p for p, _ in pairs]
        anchors = [a for _, a in pairs]
        return _L1DriftHipFn.apply(meta, len(params), *params, *anchors)
    return ref.l1_drift_fused(pairs)


class _BnTrain2dFn(torch.autograd.Function):
    """Fused training BatchNorm2d over channels-last rows (bn_train.hip).

    One kernel per direction instead of MIOpen's 5-kernel chains +
    SubTensorOp casts + the autocast fp32 round-trip; running stats update
    inside the forward kernel (replay-safe for hipGraph capture)."""

    @staticmethod
    def forward(ctx, x, gamma, beta, running_mean, running_var, momentum, eps,
                nbt, relu):
        ext = _ext_or_raise("bn_train")
        c = x.shape[1]
        m = x.numel() // c
        y = torch.empty_like(x)
        smean = torch.empty(c, device=x.device, dtype=torch.float32)
        sinv = torch.empty(c, device=x.device, dtype=torch.float32)
        # split-row partial-sum workspace (overwritten densely every call —
        # no zeroing kernel needed)
        nslab = ext.bn_train_nslab(m, c)
        part = torch.empty(2, nslab, c, device=x.device, dtype=torch.float32)
        has_rs = running_mean is not None
        ext.bn_train_fwd(
            x.data_ptr(), y.data_ptr(),
            gamma.detach().contiguous().data_ptr(),
            beta.detach().contiguous().data_ptr(),
            running_mean.data_ptr() if has_rs else 0,
            running_var.data_ptr() if has_rs else 0,
            smean.data_ptr(), sinv.data_ptr(),
            part[0].data_ptr(), part[1].data_ptr(),
            nbt.data_ptr() if nbt is not None else 0, m, c,
            float(momentum), float(eps), float(m) / float(m - 1),
            int(relu), _dt(x), _stream())
        ctx.save_for_backward(x, gamma, beta, smean, sinv)
        ctx.relu = bool(relu)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, gamma, beta, smean, sinv = ctx.saved_tensors
        if x.dim() == 4:
            if not dy.is_contiguous(memory_format=torch.channels_last):
                dy = dy.contiguous(memory_format=torch.channels_last)
        elif not dy.is_contiguous():
            dy = dy.contiguous()
        if dy.dtype != x.dtype:
            dy = dy.to(x.dtype)
        ext = _ext_or_raise("bn_train")
        c = x.shape[1]
        m = x.numel() // c
        dx = torch.empty_like(x)
        dgamma = torch.empty(c, device=x.device, dtype=torch.float32)
        dbeta = torch.empty(c, device=x.device, dtype=torch.float32)
        nslab = ext.bn_train_nslab(m, c)
        part = torch.empty(2, nslab, c, device=x.device, dtype=torch.float32)
        ext.bn_train_bwd(x.data_ptr(), dy.data_ptr(), dx.data_ptr(),
                         gamma.detach().contiguous().data_ptr(),
                         beta.detach().contiguous().data_ptr(),
                         smean.data_ptr(), sinv.data_ptr(),
                         dgamma.data_ptr(), dbeta.data_ptr(),
                         part[0].data_ptr(), part[1].data_ptr(),
                         m, c, int(ctx.relu), _dt(x), _stream())
        return dx, dgamma, dbeta, None, None, None, None, None, None


def bn_train_2d(x: torch.Tensor, bn,
                relu: bool = False) -> Optional[torch.Tensor]:
    """Fused TRAINING-mode BatchNorm for the head-epoch regime.  Applies to
    CUDA channels-last 4-D tensors (BatchNorm2d) and contiguous 2-D tensors
    (BatchNorm1d — the BNNeck bottleneck) with C % 64 == 0 and a small row
    count (M = N·H·W ≤ 16384 — covers the cached-prototype batches AND
    batch-64 layer-4 full-image fine-tuning at 16×8 spatial; very large M
    stays on MIOpen's multi-block reduction).  The
    num_batches_tracked increment is fused into the forward kernel.  Returns
    None when the fused path does not apply."""
    if os.environ.get("FLREID_NO_FUSED_BN", "0") == "1":
        return None
    if (not x.is_cuda or x.dtype not in (torch.float32, torch.bfloat16)
            or bn.weight is None or bn.bias is None):
        return None
    if x.dim() == 4:
        if not x.is_contiguous(memory_format=torch.channels_last):
            return None
    elif x.dim() == 2:
        if not x.is_contiguous():
            return None
    else:
        return None
    c = x.shape[1]
    m = x.numel() // c
    if c % 64 != 0 or m < 2 or m > 16384:
        return None
    if bn.momentum is None:  # cumulative-average mode: keep torch semantics
        return None
    if not extension_available():
        return None
    nbt = None
    if bn.track_running_stats and bn.running_mean is not None:
        nbt = bn.num_batches_tracked
        rm, rv = bn.running_mean, bn.running_var
    else:
        rm = rv = None
    return _BnTrain2dFn.apply(x, bn.weight, bn.bias, rm, rv,
                              bn.momentum, bn.eps, nbt, relu)


def _cl(t: torch.Tensor) -> torch.Tensor:
    return t if t.is_contiguous(memory_format=torch.channels_last) \
        else t.contiguous(memory_format=torch.channels_last)


def compose_theta_bf16(ext, gw: torch.Tensor, atten: Optional[torch.Tensor],
                       aw: Optional[torch.Tensor]) -> torch.Tensor:
    """θ = atten⊙gw + aw composed DIRECTLY to bf16 in the weight's own
    physical layout (compose2 kernel) — one pass, no fp32 θ in HBM, no
    autocast cast kernel.  atten broadcasts over the LOGICAL last dim; for
    channels-last 4-D weights the physical index of that dim is
    (i / C) % kw."""
    gw_d = gw.detach()
    if gw_d.dim() == 4:
        assert gw_d.is_contiguous(memory_format=torch.channels_last)
        inner, L = gw_d.shape[1], gw_d.shape[-1]
    else:
        assert gw_d.is_contiguous()
        inner = 1
        L = atten.numel() if atten is not None else 1
    if atten is not None:
        assert atten.numel() == L
    out = torch.empty_like(gw_d, dtype=torch.bfloat16)
    ext.compose2(gw_d.data_ptr(),
                 atten.detach().float().contiguous().data_ptr() if atten is not None else 0,
                 aw.detach().data_ptr() if aw is not None else 0,
                 out.data_ptr(), gw_d.numel(), L, inner, _dt(gw_d), _BF16,
                 _stream())
    return out


class _AdaptiveLinear1x1Fn(torch.autograd.Function):
    """Default-on fused adaptive 1×1/linear path (K2 at M=N·H·W): θ is
    composed straight to bf16 (compose2 — no fp32 θ in HBM, no autocast
    cast pass) and the GEMMs run bf16 hipBLASLt.  aw's gradient is the
    bf16 wgrad GEMM cast once to fp32.  gw/atten frozen (FedSTIL)."""

    @staticmethod
    def forward(ctx, x, gw2d, atten, aw2d, bias):
        ext = _ext_or_raise("compose2")
        x_bf = x.detach().to(torch.bfloat16).contiguous()
        theta = compose_theta_bf16(ext, gw2d, atten, aw2d)
        y = x_bf @ theta.t()
        if bias is not None:
            y = y + bias.detach().to(torch.bfloat16)
        ctx.save_for_backward(x_bf, theta)
        ctx.x_dtype = x.dtype
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, grad_out):
        x_bf, theta = ctx.saved_tensors
        g = grad_out.to(torch.bfloat16)
        dx = (g @ theta).to(ctx.x_dtype)
        d_aw = (g.t() @ x_bf).to(torch.float32)
        d_bias = grad_out.sum(0).to(torch.float32) if ctx.has_bias else None
        return dx, None, None, d_aw, d_bias


def adaptive_linear_1x1(x2d: torch.Tensor, gw2d: torch.Tensor,
                        atten: Optional[torch.Tensor],
                        aw2d: Optional[torch.Tensor],
                        bias: Optional[torch.Tensor]) -> Optional[torch.Tensor]:
    """Dispatcher for the fused 1×1 route; None when out of regime."""
    if (not x2d.is_cuda or not extension_available()
            or os.environ.get("FLREID_NO_FUSED_1X1", "0") == "1"):
        return None
    if gw2d.requires_grad or (atten is not None and atten.requires_grad):
        return None
    if x2d.dtype != torch.bfloat16 and not torch.is_autocast_enabled():
        return None
    if gw2d.dtype != torch.float32 or (aw2d is not None
                                       and aw2d.dtype != torch.float32):
        return None
    return _AdaptiveLinear1x1Fn.apply(x2d, gw2d, atten, aw2d, bias)


def conv_theta_tile(ext, w: torch.Tensor, atten: Optional[torch.Tensor],
                    aw: Optional[torch.Tensor], mode: int) -> torch.Tensor:
    """θ = atten⊙w + aw emitted straight into the conv-tiled bf16 layout
    the hand-written conv consumes (mode 0: [C/32][9][K][32p]; mode 1: the
    flipped-transposed dgrad tile [K/32][9][C][32p]) — composition, cast
    and im2col-free weight re-layout in ONE pass."""
    w_d = w.detach()
    assert w_d.is_contiguous(memory_format=torch.channels_last)
    k, c = w_d.shape[0], w_d.shape[1]
    out = torch.empty(c * 9 * k, device=w.device, dtype=torch.bfloat16)
    ext.conv3x3_tile(
        w_d.data_ptr(),
        atten.detach().float().contiguous().data_ptr() if atten is not None else 0,
        aw.detach().data_ptr() if aw is not None else 0,
        out.data_ptr(), c, k, _dt(w_d), mode, _stream())
    return out


def _conv_impl() -> str:
    """Per-op conv engine policy.  'auto' (default) picks the MEASURED
    winner per op (profiles/README.md K1 table): the library conv
    currently beats the hand kernels on the ReID shapes (CK fwd 529 TF vs
    hand 428; wgrad 380 vs 103), so auto routes the conv math to the
    library while keeping the fused one-pass θ production (compose2 to
    bf16 — no fp32 θ, no autocast cast) and the fp32 weight-grad path.
    'hand' forces the hand-written kernels everywhere (the A/B switch the
    probe and profiles use); 'lib' forces the library."""
    return os.environ.get("FLREID_CONV_IMPL", "auto")


class _Conv3x3Fn(torch.autograd.Function):
    """3×3 s1p1 conv with the FedSTIL composition fused into the weight
    production (K1).

    fwd:   θ_bf16 tile/tensor from ONE compose2/tile pass over
           (gw, atten, aw); conv = hand halo kernel (conv3x3_img*.hip) or
           the library on θ, by policy (_conv_impl).
    dgrad: hand = the same kernel on the flip-transposed θ tile;
           lib = conv2d_input on θ.
    wgrad: hand = transpose-first MFMA reduction (fp32 out);
           lib = conv2d_weight (bf16, cast once).  For the adaptive layer
           dθ IS d(adaptive_weight) (identity composition).
    Only frozen gw/atten are supported (FedSTIL trains aw alone;
    plain nn.Conv2d passes the weight in slot 1)."""

    @staticmethod
    def forward(ctx, x, weight, atten, aw):
        ext = _ext_or_raise("conv3x3_img_fwd")
        x_bf = _cl(x.detach().to(torch.bfloat16))
        n, c, h, w = x_bf.shape
        k = weight.shape[0]
        hand = _conv_impl() == "hand"
        if hand:
            theta = compose_theta_bf16(ext, weight, atten, aw)  # plain CL θ
            y = torch.empty(n, k, h, w, device=x.device, dtype=torch.bfloat16,
                            memory_format=torch.channels_last)
            ext.conv3x3_img_fwd_ldsw(x_bf.data_ptr(), theta.data_ptr(),
                                     y.data_ptr(), n, h, w, c, k, _stream())
        else:
            theta = compose_theta_bf16(ext, weight, atten, aw)
            y = torch.nn.functional.conv2d(x_bf, theta, padding=1)
        # the lib dgrad reuses the forward's θ — composing twice per step
        # cost ~1.2 ms/round
        ctx.save_for_backward(x_bf, weight, atten, aw,
                              None if hand else theta)
        ctx.x_dtype = x.dtype
        ctx.w_dtype = weight.dtype
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _ext_or_raise("conv3x3_img_fwd")
        x_bf, weight, atten, aw, theta = ctx.saved_tensors
        n, c, h, w = x_bf.shape
        k = weight.shape[0]
        hand = _conv_impl() == "hand"
        dy_bf = _cl(dy.to(torch.bfloat16))
        dx = None
        if ctx.needs_input_grad[0]:
            if hand:
                # dgrad = the same kernel on the flip-transposed θ tile
                wt_tile = conv_theta_tile(ext, weight, atten, aw, mode=1)
                dx = torch.empty(n, c, h, w, device=dy.device,
                                 dtype=torch.bfloat16,
                                 memory_format=torch.channels_last)
                ext.conv3x3_img_fwd(dy_bf.data_ptr(), wt_tile.data_ptr(),
                                    dx.data_ptr(), n, h, w, k, c, _stream())
            else:
                if theta is None:
                    theta = compose_theta_bf16(ext, weight, atten, aw)
                dx = torch.nn.grad.conv2d_input((n, c, h, w), theta, dy_bf,
                                                padding=1)
            if ctx.x_dtype != torch.bfloat16:
                dx = dx.to(ctx.x_dtype)
        d_weight = d_aw = None
        if ctx.needs_input_grad[1] or ctx.needs_input_grad[3]:
            if hand:
                # wgrad on pre-transposed operands (the in-kernel
                # transposed scatter measured 75 LDS-conflict cycles/MFMA);
                # two exclusive M-split partials summed here
                m_rows = n * h * w
                # 128-elem pads both ends: the wgrad kernel loads
                # shifted/tail chunks unclamped (OOR lanes zero-masked)
                dyt = torch.empty(k * m_rows + 256, device=dy.device,
                                  dtype=torch.bfloat16)
                xt = torch.empty(c * m_rows + 256, device=dy.device,
                                 dtype=torch.bfloat16)
                ext.transpose_bf16(dy_bf.data_ptr(), dyt.data_ptr() + 256,
                                   m_rows, k, _stream())
                ext.transpose_bf16(x_bf.data_ptr(), xt.data_ptr() + 256,
                                   m_rows, c, _stream())
                part = torch.empty(2, k * 9 * c, device=dy.device,
                                   dtype=torch.float32)
                ext.conv3x3_wgrad(dyt.data_ptr() + 256, xt.data_ptr() + 256,
                                  part.data_ptr(), n, h, w, c, k, _stream())
                # [K][9][C] flat == channels-last [K,C,3,3] physical layout
                dw = (part[0] + part[1]).view(k, 3, 3, c).permute(0, 3, 1, 2)
            else:
                dw = torch.nn.grad.conv2d_weight(
                    x_bf, (k, c, 3, 3), dy_bf, padding=1).to(torch.float32)
            if ctx.needs_input_grad[3]:
                d_aw = dw              # identity composition, fp32 direct
            else:
                d_weight = dw if ctx.w_dtype == torch.float32 else dw.to(ctx.w_dtype)
        return dx, d_weight, None, d_aw


def conv3x3_try(x: torch.Tensor, weight: torch.Tensor,
                atten: Optional[torch.Tensor] = None,
                aw: Optional[torch.Tensor] = None) -> Optional[torch.Tensor]:
    """Dispatch guard for the hand-written 3×3 s1p1 conv: returns None when
    out of regime (caller falls back to the library conv).  Regime: the
    ReID layer-4 shapes — H·W ≤ 128 (one image per block), H·W % 16 == 0,
    C % 32, K % 32, channels-last weights, frozen gw/atten, bf16 compute
    (native or autocast)."""
    if not x.is_cuda or x.dim() != 4 or not extension_available():
        return None
    if os.environ.get("FLREID_NO_FUSED_CONV", "0") == "1":
        return None
    n, c, h, w = x.shape
    k = weight.shape[0]
    hw = h * w
    if hw > 128 or hw % 16 or h + 2 > 18 or w + 2 > 10:
        return None
    if c % 32 or k % 32:
        return None
    if x.dtype != torch.bfloat16 and not torch.is_autocast_enabled():
        return None
    if weight.requires_grad and (aw is not None or atten is not None):
        return None                      # composed path trains aw only
    if atten is not None and atten.requires_grad:
        return None
    if weight.dim() != 4 or not weight.is_contiguous(memory_format=torch.channels_last):
        return None
    if aw is not None and not aw.is_contiguous(memory_format=torch.channels_last):
        return None
    return _Conv3x3Fn.apply(x, weight, atten, aw)


def conv3x3_fwd_nhwc(x: torch.Tensor, weight: torch.Tensor) -> torch.Tensor:
    """Hand-written 3×3 s1 p1 NHWC bf16 conv forward (K1; frozen-backbone
    eval path).  x: channels-last bf16 [N, C, H, W]; weight: fp32
    channels-last [K, C, 3, 3].  Returns channels-last bf16 [N, K, H, W].
    Falls back to the library conv outside the kernel constraints."""
    n, c, h, w = x.shape
    k = weight.shape[0]
    if c % 32 != 0 or k % 16 != 0:
        return torch.nn.functional.conv2d(x, weight.to(x.dtype), padding=1)
    ext = _ext_or_raise("conv3x3_fwd")
    assert x.is_contiguous(memory_format=torch.channels_last)
    wcl = weight.detach()
    if not wcl.is_contiguous(memory_format=torch.channels_last):
        wcl = wcl.contiguous(memory_format=torch.channels_last)
    out = torch.empty(n, k, h, w, device=x.device,
                      dtype=torch.bfloat16).to(memory_format=torch.channels_last)
    ext.conv3x3_fwd(x.data_ptr(), wcl.float().data_ptr(), out.data_ptr(),
                    n, h, w, c, k, _stream())
    return out
