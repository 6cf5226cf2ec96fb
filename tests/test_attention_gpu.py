"""MFMA attention forward vs fp32 torch reference (serving path)."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu


def _ref_attention(q, k, v, scale):
    s = (q.float() @ k.float().transpose(-1, -2)) * scale
    p = torch.softmax(s, dim=-1)
    return p @ v.float()


@pytest.mark.parametrize("B,H,S", [(1, 1, 32), (2, 4, 128), (1, 2, 96),
                                   (2, 12, 512)])
def test_attn_fwd_matches_reference(B, H, S):
    from autodist_amd.ops import api
    assert api.has_gpu_ops()
    torch.manual_seed(0)
    D = 64
    q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    scale = 1.0 / math.sqrt(D)
    o = api.ext().attn_fwd(q, k, v, scale)
    ref = _ref_attention(q, k, v, scale)
    err = (o.float() - ref).abs().max().item()
    assert err < 3e-2, f"max err {err}"


def test_attn_fwd_outlier_rows():
    """Force large max jumps across key tiles (online-softmax rescale path,
    guide rule 26: an input that FORCES the branch)."""
    from autodist_amd.ops import api
    torch.manual_seed(1)
    B, H, S, D = 1, 1, 128, 64
    q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    # spike a LATE key so every row's max jumps at the final tile
    k[0, 0, 120] = (q[0, 0, 5].float() * 4).to(torch.bfloat16)
    scale = 1.0 / math.sqrt(D)
    o = api.ext().attn_fwd(q, k, v, scale)
    ref = _ref_attention(q, k, v, scale)
    err = (o.float() - ref).abs().max().item()
    assert err < 5e-2, f"max err {err}"


def test_fused_sdpa_dispatch():
    """fused_sdpa uses the kernel for inference AND training (the backward
    was GPU-validated in round 2); masked/dropout cases fall back to SDPA."""
    from autodist_amd.ops.fused_attention import can_use_fused, fused_sdpa
    q = torch.randn(1, 2, 64, 64, device="cuda", dtype=torch.bfloat16)
    with torch.no_grad():
        assert can_use_fused(q, None, 0.0)
        o = fused_sdpa(q, q, q)
        ref = torch.nn.functional.scaled_dot_product_attention(
            q.float(), q.float(), q.float())
        assert (o.float() - ref).abs().max().item() < 3e-2
    qg = q.clone().requires_grad_(True)
    assert can_use_fused(qg, None, 0.0)       # training path fused
    assert can_use_fused(qg, None, 0.1)       # hash dropout supported
    pad = torch.ones(1, 1, 1, 64, device="cuda", dtype=torch.bool)
    assert can_use_fused(qg, pad, 0.1)        # key-padding mask supported
    full = torch.zeros(1, 2, 64, 64, device="cuda", dtype=torch.bfloat16)
    assert not can_use_fused(qg, full, 0.0)   # full S x S mask -> SDPA
    out = fused_sdpa(qg, qg, qg)
    assert out.grad_fn is not None


def test_bert_eval_uses_fused_path():
    """BERT eval forward with the fused attention stays close to the SDPA
    forward."""
    from autodist_amd.models.bert import bert_tiny
    torch.manual_seed(0)
    model = bert_tiny().to("cuda").eval()
    ids = torch.randint(0, 1000, (2, 32), device="cuda")
    with torch.no_grad(), torch.autocast("cuda", torch.bfloat16):
        logits, nsp = model(ids)
    assert torch.isfinite(logits).all() and torch.isfinite(nsp).all()
