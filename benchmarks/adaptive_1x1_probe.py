#!/usr/bin/env python3
"""Probe: fused adaptive GEMM (K2, compose-in-prologue) vs compose+hipBLASLt
at the head-epoch 1×1-conv shapes (M = 2048 feature rows).

Decides ROADMAP item 2 — whether AdaptiveConv2d's pointwise path should run
through ops.adaptive_linear instead of θ-compose + F.linear.
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from flreid_amd import ops


def bench(fn, n=100, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6     # µs


def main():
    assert ops.extension_available()
    torch.manual_seed(0)
    # layer4 1×1 shapes at 64×8×4 rows: (M, K_out, C_in)
    shapes = [(2048, 512, 1024), (2048, 512, 2048), (2048, 2048, 512),
              (2048, 2048, 1024), (64, 8000, 2048)]
    for m, n_out, c_in in shapes:
        x = torch.randn(m, c_in, device="cuda").bfloat16()
        gw = torch.randn(n_out, c_in, device="cuda") * 0.02
        aw = torch.randn_like(gw) * 0.02
        atten = torch.full((c_in,), 0.9, device="cuda")

        with torch.no_grad():
            t_fused = bench(lambda: ops.adaptive_linear_fwd(x, gw, atten, aw, None))

            def eager():
                theta = ops.adaptive_compose(gw, atten, aw)
                return torch.nn.functional.linear(x, theta.bfloat16())

            t_eager = bench(eager)

            y1 = ops.adaptive_linear_fwd(x, gw, atten, aw, None)
            theta = ops.adaptive_compose(gw, atten, aw)
            y2 = torch.nn.functional.linear(x, theta.bfloat16())
            ok = torch.allclose(y1.float(), y2.float(), atol=0.5, rtol=5e-2)
        tf = 2 * m * n_out * c_in / (t_fused / 1e6) / 1e12
        print(f"M={m:5d} N={n_out:5d} K={c_in:5d}  fused {t_fused:7.1f} us "
              f"({tf:5.1f} TF)  compose+blaslt {t_eager:7.1f} us  "
              f"{'OK' if ok else 'MISMATCH'}")


if __name__ == "__main__":
    main()
