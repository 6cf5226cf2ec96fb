"""MFMA attention BACKWARD vs fp32 autograd reference.

DRAFT-gated: the backward kernels are compile-checked but not yet
GPU-validated (round-1 GPU budget was exhausted after the forward landed);
set AUTODIST_EXPERIMENTAL=1 to run — this is round 2's first GPU task."""
import math
import os

import pytest
import torch

pytestmark = [
    pytest.mark.gpu,
    pytest.mark.skipif(os.environ.get("AUTODIST_EXPERIMENTAL") != "1",
                       reason="draft kernel: set AUTODIST_EXPERIMENTAL=1"),
]


@pytest.mark.parametrize("B,H,S", [(1, 1, 32), (2, 3, 128)])
def test_attn_bwd_matches_autograd(B, H, S):
    from autodist_amd.ops import api
    assert api.has_gpu_ops()
    torch.manual_seed(0)
    D = 64
    scale = 1.0 / math.sqrt(D)
    q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    dout = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    # fp32 autograd reference on the SAME bf16-quantized inputs
    qf = q.float().requires_grad_(True)
    kf = k.float().requires_grad_(True)
    vf = v.float().requires_grad_(True)
    s = (qf @ kf.transpose(-1, -2)) * scale
    p = torch.softmax(s, dim=-1)
    o_ref = p @ vf
    o_ref.backward(dout.float())
    o = api.ext().attn_fwd(q, k, v, scale)
    dq, dk, dv = api.ext().attn_bwd(q, k, v, o, dout, scale)
    tol = 6e-2
    for name, got, ref in (("dq", dq, qf.grad), ("dk", dk, kf.grad),
                           ("dv", dv, vf.grad)):
        err = (got.float() - ref).abs().max().item()
        assert err < tol, f"{name} max err {err}"
