import sys, os
sys.path.insert(0, "/root/repo")
import torch
from flreid_amd import ops
torch.backends.cudnn.benchmark = True
n, c, h, w, k = 64, 512, 16, 8, 512
x = torch.randn(n, c, h, w, device="cuda").bfloat16().to(memory_format=torch.channels_last)
wt = (torch.randn(k, c, 3, 3, device="cuda") / c).to(memory_format=torch.channels_last)
with torch.no_grad():
    for _ in range(3):
        ops.conv3x3_fwd_nhwc(x, wt)
torch.cuda.synchronize()
print("done")
