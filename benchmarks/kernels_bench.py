#!/usr/bin/env python3
"""Kernel microbenchmarks: fused HIP ops vs eager PyTorch composition.

Run on an MI355X box:
    python benchmarks/kernels_bench.py
Profile (separate runs per rocprofv3 constraints):
    rocprofv3 --kernel-trace --stats -d out -- python benchmarks/kernels_bench.py
    rocprofv3 --pmc SQ_INSTS_MFMA,SQ_INSTS_VALU,SQ_LDS_BANK_CONFLICT \
        -d out -- python benchmarks/kernels_bench.py --once
"""

import argparse
import sys
import time
import os

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from flreid_amd import ops
from flreid_amd.ops import reference as ref


def bench(fn, n=50, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--once", action="store_true",
                   help="single invocation per op (for PMC counter runs)")
    args = p.parse_args()
    n = 1 if args.once else 50
    assert torch.cuda.is_available() and ops.extension_available()
    results = {}

    # K8: eval similarity GEMM (MFMA f32) — Q×G at validation scale
    qf = ref.l2_normalize(torch.randn(2048, 2048, device="cuda"))
    gf = ref.l2_normalize(torch.randn(16384, 2048, device="cuda"))
    with torch.no_grad():
        results["pairwise_sim_mfma"] = bench(
            lambda: ops.similarity_matrix(qf, gf), n)
        results["pairwise_sim_eager"] = bench(lambda: qf @ gf.t(), n)
        flops = 2 * qf.shape[0] * gf.shape[0] * qf.shape[1]
        results["pairwise_sim_mfma_TF"] = round(
            flops / (results["pairwise_sim_mfma"] / 1e3) / 1e12, 1)

    # K6: fused label-smooth CE fwd+bwd vs eager
    score = torch.randn(64, 8000, device="cuda", requires_grad=True)
    target = torch.randint(0, 8000, (64,), device="cuda")

    def fused_ce():
        loss = ops.ce_label_smooth(score, target, 0.1)
        loss.backward()
        score.grad = None

    def eager_ce():
        loss = ref.ce_label_smooth(score, target, 0.1)
        loss.backward()
        score.grad = None

    results["ce_smooth_fused"] = bench(fused_ce, n)
    results["ce_smooth_eager"] = bench(eager_ce, n)

    # K3: fused window attention (Swin stage-1 tiny shape) vs eager
    q = torch.randn(4096, 3, 49, 32, device="cuda").bfloat16()
    k, v = torch.randn_like(q), torch.randn_like(q)
    bias = torch.randn(3, 49, 49, device="cuda")
    with torch.no_grad():
        results["window_attn_fused"] = bench(
            lambda: ops.window_attention(q, k, v, bias, None, 0.18), n)
        results["window_attn_eager"] = bench(
            lambda: ref.window_attention(q, k, v, bias, None, 0.18), n)

    # K11: rowwise L2 normalize
    x = torch.randn(16384, 2048, device="cuda")
    with torch.no_grad():
        results["l2norm_fused"] = bench(lambda: ops.l2_normalize(x), n)
        results["l2norm_eager"] = bench(lambda: ref.l2_normalize(x), n)
        gbps = 2 * x.numel() * 4 / (results["l2norm_fused"] / 1e3) / 1e9
        results["l2norm_fused_GBps"] = round(gbps, 0)

    # compose (FedSTIL θ) standalone
    gw = torch.randn(2048, 8000, device="cuda")
    aw = torch.randn_like(gw)
    atten = torch.full((8000,), 0.9, device="cuda")
    with torch.no_grad():
        results["compose_fused"] = bench(
            lambda: ops.adaptive_compose(gw, atten, aw), n)
        results["compose_eager"] = bench(
            lambda: ref.adaptive_compose(gw, atten, aw), n)

    # fused train-mode BN fwd+bwd vs MIOpen (head-epoch shape: M=2048, C=512)
    import torch.nn as nn
    for c in (512, 2048):
        bn = nn.BatchNorm2d(c).cuda().train()
        xb = (torch.randn(64, c, 8, 4, device="cuda").bfloat16()
              .to(memory_format=torch.channels_last))

        def fused_bn():
            x1 = xb.detach().requires_grad_(True)
            y = ops.bn_train_2d(x1, bn)
            y.backward(y.detach())

        def miopen_bn():
            x1 = xb.detach().requires_grad_(True)
            with torch.autocast("cuda", dtype=torch.bfloat16):
                y = nn.functional.batch_norm(
                    x1, bn.running_mean, bn.running_var, bn.weight, bn.bias,
                    True, 0.1, bn.eps)
            y.backward(y.detach())

        results[f"bn_train_c{c}_fused"] = bench(fused_bn, n)
        results[f"bn_train_c{c}_miopen"] = bench(miopen_bn, n)

    # fused multi-tensor L1 drift (FedSTIL per-step regulariser) vs _foreach
    shapes = [(512, 512, 3, 3), (2048, 512, 1, 1), (512, 2048, 1, 1),
              (8000, 2048)] * 3
    params = [torch.randn(s, device="cuda", requires_grad=True)
              for s in shapes]
    anchors = [torch.randn(s, device="cuda") for s in shapes]
    pairs = list(zip(params, anchors))

    def fused_drift():
        loss = ops.l1_drift(pairs)
        loss.backward()
        for p in params:
            p.grad = None

    def foreach_drift():
        loss = ref.l1_drift_fused(pairs)
        loss.backward()
        for p in params:
            p.grad = None

    results["drift_fused"] = bench(fused_drift, n)
    results["drift_foreach"] = bench(foreach_drift, n)

    for k in sorted(results):
        v = results[k]
        print(f"{k:28s} {v:10.3f}" + (" ms" if "_TF" not in k and "GBps" not in k else ""))


if __name__ == "__main__":
    main()
