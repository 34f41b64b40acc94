"""Minimal kernel-only driver for rocprofv3 PMC collection on the conv
suite: runs each conv kernel N times on the bench hot shape with nothing
else in between, so per-kernel counter rows are unambiguous."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from flreid_amd import ops


def main():
    n, c, h, w, k = 64, 512, 16, 8, 512
    reps = 20
    x = torch.randn(n, c, h, w, device="cuda").bfloat16() \
        .to(memory_format=torch.channels_last)
    wt_bf = ((torch.randn(k, c, 3, 3, device="cuda") / c)
             .to(memory_format=torch.channels_last).bfloat16())
    dy = torch.randn(n, k, h, w, device="cuda").bfloat16() \
        .to(memory_format=torch.channels_last)
    ext = ops._load_extension()
    st = torch.cuda.current_stream().cuda_stream
    wt_tile = ops.conv_theta_tile(ext, wt_bf, None, None, mode=0)

    import sys
    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    y = torch.empty(n, k, h, w, device="cuda", dtype=torch.bfloat16,
                    memory_format=torch.channels_last)
    if which in ("all", "tile"):
        for _ in range(reps):
            ext.conv3x3_img_fwd(x.data_ptr(), wt_tile.data_ptr(),
                                y.data_ptr(), n, h, w, c, k, st)
    if which in ("all", "ldsw"):
        for _ in range(reps):
            ext.conv3x3_img_fwd_ldsw(x.data_ptr(), wt_bf.data_ptr(),
                                     y.data_ptr(), n, h, w, c, k, st)
    if which in ("all", "wgrad"):
        m_rows = n * h * w
        dyt = torch.empty(k * m_rows + 256, device="cuda",
                          dtype=torch.bfloat16)
        xt = torch.empty(c * m_rows + 256, device="cuda",
                         dtype=torch.bfloat16)
        ext.transpose_bf16(dy.data_ptr(), dyt.data_ptr() + 256, m_rows, k, st)
        ext.transpose_bf16(x.data_ptr(), xt.data_ptr() + 256, m_rows, c, st)
        part = torch.empty(2, k * 9 * c, device="cuda", dtype=torch.float32)
        for _ in range(reps):
            ext.conv3x3_wgrad(dyt.data_ptr() + 256, xt.data_ptr() + 256,
                              part.data_ptr(), n, h, w, c, k, st)
    torch.cuda.synchronize()
    print("pmc probe done")


if __name__ == "__main__":
    main()
