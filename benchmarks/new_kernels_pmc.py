"""PMC collection driver for the round-2 kernels (K7 KD/iCaRL, K3 window
attention bwd, K4 patch-merge LN, compose2): each kernel runs N times with
nothing interleaved so per-kernel counter rows are unambiguous."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from flreid_amd import ops


def main():
    reps = 20
    torch.manual_seed(0)

    # K7 kd_fwd (DistillKL shapes)
    zs = torch.randn(64, 8000, device="cuda")
    zt = torch.randn(64, 8000, device="cuda")
    for _ in range(reps):
        loss = ops.kd_loss(zs.requires_grad_(True), zt, 4.0)
    torch.cuda.synchronize()

    # K7 icarl_distill (iCaRL distillation pass shape)
    score = torch.randn(64, 8000, device="cuda", requires_grad=True)
    target = torch.randint(0, 8000, (64,), device="cuda")
    prev = torch.randn(64, 4000, device="cuda")
    for _ in range(reps):
        loss = ops.icarl_distill_loss(score, target, prev)
    torch.cuda.synchronize()

    # K3 training fwd+bwd (Swin stage-3 geometry)
    q = torch.randn(128, 12, 49, 32, device="cuda").bfloat16().requires_grad_(True)
    k = torch.randn_like(q).requires_grad_(True)
    v = torch.randn_like(q).requires_grad_(True)
    bias = torch.randn(12, 49, 49, device="cuda", requires_grad=True)
    for _ in range(reps):
        out = ops.window_attention(q, k, v, bias, None, 32 ** -0.5)
        out.sum().backward()
        for t in (q, k, v, bias):
            t.grad = None
    torch.cuda.synchronize()

    # K4 patch-merge LN fwd+bwd (stage-1 -> stage-2 transition)
    x = torch.randn(64, 56 * 56, 96, device="cuda").bfloat16().requires_grad_(True)
    gamma = torch.randn(4 * 96, device="cuda", requires_grad=True)
    beta = torch.randn(4 * 96, device="cuda", requires_grad=True)
    for _ in range(reps):
        y = ops.patch_merge_ln(x, gamma, beta, 56, 56)
        y.sum().backward()
        for t in (x, gamma, beta):
            t.grad = None
    torch.cuda.synchronize()

    # compose2 (the per-step θ production, classifier + conv shapes)
    ext = ops._load_extension()
    gw = torch.randn(8000, 2048, device="cuda")
    aw = torch.randn_like(gw)
    atten = torch.rand(2048, device="cuda")
    for _ in range(reps):
        ops.compose_theta_bf16(ext, gw, atten, aw)
    gwc = torch.randn(512, 512, 3, 3, device="cuda").to(
        memory_format=torch.channels_last)
    awc = torch.randn_like(gwc)
    att3 = torch.rand(3, device="cuda")
    for _ in range(reps):
        ops.compose_theta_bf16(ext, gwc, att3, awc)
    torch.cuda.synchronize()
    print("new-kernel pmc probe done")


if __name__ == "__main__":
    main()
