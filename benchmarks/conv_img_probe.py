"""Microbench: hand-written conv3x3_img suite (K1) vs the library conv.

Shapes = the bench hot path (FedSTIL ResNet-50 layer4, batch 64, 16x8
spatial, last_stride=1) plus the 512-image eval-chunk shape.  Prints
ms and TFLOP/s for fwd / dgrad / wgrad of both paths.
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from flreid_amd import ops


def bench(fn, n=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1000.0


def run_shape(n, c, h, w, k):
    flops = 2.0 * n * h * w * c * k * 9
    x = torch.randn(n, c, h, w, device="cuda").bfloat16() \
        .to(memory_format=torch.channels_last)
    wt = ((torch.randn(k, c, 3, 3, device="cuda") / c)
          .to(memory_format=torch.channels_last))
    wt_bf = wt.bfloat16()
    dy = torch.randn(n, k, h, w, device="cuda").bfloat16() \
        .to(memory_format=torch.channels_last)

    ext = ops._load_extension()
    # pre-tiled weights (in production the compose kernel emits these
    # directly from gw/atten/aw in one pass per step)
    wt_tile = ops.conv_theta_tile(ext, wt_bf, None, None, mode=0)
    wt_tile_d = ops.conv_theta_tile(ext, wt_bf, None, None, mode=1)

    def ours_tile():
        return ops.conv_theta_tile(ext, wt_bf, None, None, mode=0)

    def ours_fwd():
        y = torch.empty(n, k, h, w, device="cuda", dtype=torch.bfloat16,
                        memory_format=torch.channels_last)
        ext.conv3x3_img_fwd(x.data_ptr(), wt_tile.data_ptr(), y.data_ptr(),
                            n, h, w, c, k, torch.cuda.current_stream().cuda_stream)
        return y

    def lib_fwd():
        return torch.nn.functional.conv2d(x, wt_bf, padding=1)

    def ours_dgrad():
        dx = torch.empty(n, c, h, w, device="cuda", dtype=torch.bfloat16,
                         memory_format=torch.channels_last)
        ext.conv3x3_img_fwd(dy.data_ptr(), wt_tile_d.data_ptr(), dx.data_ptr(),
                            n, h, w, k, c, torch.cuda.current_stream().cuda_stream)
        return dx

    def ours_wgrad():
        st = torch.cuda.current_stream().cuda_stream
        m_rows = n * h * w
        dyt = torch.empty(k * m_rows + 256, device="cuda", dtype=torch.bfloat16)
        xt = torch.empty(c * m_rows + 256, device="cuda", dtype=torch.bfloat16)
        ext.transpose_bf16(dy.data_ptr(), dyt.data_ptr() + 256, m_rows, k, st)
        ext.transpose_bf16(x.data_ptr(), xt.data_ptr() + 256, m_rows, c, st)
        part = torch.empty(2, k * 9 * c, device="cuda", dtype=torch.float32)
        ext.conv3x3_wgrad(dyt.data_ptr() + 256, xt.data_ptr() + 256,
                          part.data_ptr(), n, h, w, c, k, st)
        return (part[0] + part[1]).view(k, 3, 3, c).permute(0, 3, 1, 2)

    def ours_fwd_ldsw():
        y = torch.empty(n, k, h, w, device="cuda", dtype=torch.bfloat16,
                        memory_format=torch.channels_last)
        ext.conv3x3_img_fwd_ldsw(x.data_ptr(), wt_bf.data_ptr(), y.data_ptr(),
                                 n, h, w, c, k,
                                 torch.cuda.current_stream().cuda_stream)
        return y

    def lib_dgrad():
        return torch.nn.grad.conv2d_input((n, c, h, w), wt_bf, dy, padding=1)

    def lib_wgrad():
        return torch.nn.grad.conv2d_weight(x, (k, c, 3, 3), dy, padding=1)

    print(f"== {n}x{c}x{h}x{w} -> {k}  ({flops/1e9:.1f} GFLOP) ==")
    for name, fn in (("ours_tile", ours_tile),
                     ("ours_fwd", ours_fwd),
                     ("ours_fwd_ldsw", ours_fwd_ldsw), ("lib_fwd", lib_fwd),
                     ("ours_dgrad", ours_dgrad), ("lib_dgrad", lib_dgrad),
                     ("ours_wgrad", ours_wgrad), ("lib_wgrad", lib_wgrad)):
        try:
            ms = bench(fn)
            print(f"  {name:12s} {ms:8.3f} ms  {flops / ms / 1e9:8.1f} TF")
        except Exception as e:
            print(f"  {name:12s} FAILED: {e}")


if __name__ == "__main__":
    torch.backends.cudnn.benchmark = True
    run_shape(64, 512, 16, 8, 512)     # train-step hot shape
    run_shape(64, 256, 16, 8, 512)     # resnet18 layer4.0
    run_shape(512, 512, 16, 8, 512)    # eval chunk
