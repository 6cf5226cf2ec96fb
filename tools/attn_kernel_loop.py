"""Tight loop over the fused attention kernels for PMC profiling.

Usage: python tools/attn_kernel_loop.py [fwd|bwd] [S] [iters]
Runs ONLY the selected kernel(s) so rocprofv3 --pmc rows are unambiguous.
"""
import math
import sys

import torch

sys.path.insert(0, sys.path[0].rsplit("/", 1)[0] if "/" in sys.path[0]
                else ".")
sys.path.insert(0, __file__.rsplit("/", 2)[0])

which = sys.argv[1] if len(sys.argv) > 1 else "fwd"
S = int(sys.argv[2]) if len(sys.argv) > 2 else 512
iters = int(sys.argv[3]) if len(sys.argv) > 3 else 100

from autodist_amd.ops import api

B, H, D = 8, 12, 64
scale = 1.0 / math.sqrt(D)
q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
k = torch.randn_like(q)
v = torch.randn_like(q)
o = api.ext().attn_fwd(q, k, v, scale)
dout = torch.randn_like(q)
torch.cuda.synchronize()
for _ in range(iters):
    if which == "fwd":
        api.ext().attn_fwd(q, k, v, scale)
    else:
        api.ext().attn_bwd(q, k, v, o, dout, scale)
torch.cuda.synchronize()
print("done", which, S, iters)
