"""Fused MFMA attention (gfx950) — forward + backward, mask + dropout.

`fused_sdpa(q, k, v, ...)` runs the hand-written CDNA4 MFMA attention
kernels (ops/csrc/attention.hip 4-wave LDS-tiled forward;
attention_bwd.hip FA2-style backward, GPU-validated vs fp32 autograd) when
applicable, else falls back to torch SDPA. Applicability: bf16, head_dim
64, S % 32 == 0, mask absent or a key-padding mask broadcastable to
[B, 1, 1, S], any dropout_p < 1 — the BERT/transformer hot path for both
serving and training. Dropout uses a counter-based hash of
(seed, bh, q, k) regenerated identically in the backward (flash-attention
style, no S x S state); the seed is drawn host-side per forward call.
"""
import math
import random
from typing import Optional

import torch

from autodist_amd.ops import api as ops_api


def _key_padding_mask(attn_mask, q) -> Optional[torch.Tensor]:
    """Convert a supported mask to the kernels' additive fp32 [B, S] form.
    Returns None when no mask, raises ValueError when unsupported."""
    if attn_mask is None:
        return None
    B, S = q.size(0), q.size(-2)
    m = attn_mask
    if m.dim() == 4 and m.size(1) == 1 and m.size(2) == 1:
        m = m[:, 0, 0, :]
    if m.dim() != 2 or m.shape != (B, S):
        raise ValueError("not a key-padding mask")
    if m.dtype == torch.bool:
        # True = attend (torch SDPA convention) -> additive 0 / -30000
        return torch.zeros(m.shape, dtype=torch.float32,
                           device=m.device).masked_fill_(~m, -30000.0)
    return m.to(torch.float32).contiguous()


def can_use_fused(q, attn_mask, dropout_p) -> bool:
    """Kernel applicability (training AND inference; reference context: the
    reference leaves attention to TF ops — hand-written CDNA4 here is the
    north-star requirement)."""
    if not (q.is_cuda and q.dtype == torch.bfloat16
            and q.size(-1) in (64, 128)
            and q.size(-2) % 32 == 0 and q.size(-2) >= 32
            and 0.0 <= dropout_p < 1.0 and ops_api.has_gpu_ops()):
        return False
    if attn_mask is None:
        return True
    if attn_mask.dim() == 4 and attn_mask.size(1) == 1 \
            and attn_mask.size(2) == 1:
        return True
    return attn_mask.dim() == 2 and attn_mask.shape == (q.size(0),
                                                        q.size(-2))


class FusedAttentionFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale, mask, p_drop, seed):
        o = ops_api.ext().attn_fwd(q, k, v, scale, mask, p_drop, seed)
        ctx.save_for_backward(q, k, v, o, *([mask] if mask is not None
                                            else []))
        ctx.scale = scale
        ctx.p_drop = p_drop
        ctx.seed = seed
        ctx.has_mask = mask is not None
        return o

    @staticmethod
    def backward(ctx, dout):
        if ctx.has_mask:
            q, k, v, o, mask = ctx.saved_tensors
        else:
            (q, k, v, o), mask = ctx.saved_tensors, None
        dq, dk, dv = ops_api.ext().attn_bwd(q, k, v, o, dout.contiguous(),
                                            ctx.scale, mask, ctx.p_drop,
                                            ctx.seed)
        return dq, dk, dv, None, None, None, None


def fused_sdpa(q, k, v, attn_mask: Optional[torch.Tensor] = None,
               dropout_p: float = 0.0, scale: Optional[float] = None):
    """Drop-in for torch SDPA on [B, H, S, D] tensors.

    Note: dropout_p is applied only in training-style calls; callers pass
    0.0 for eval (as the models do). The dropout mask differs from torch's
    RNG but is a valid iid Bernoulli(1-p) mask with exact fwd/bwd
    consistency.
    """
    if scale is None:
        scale = 1.0 / math.sqrt(q.size(-1))
    if can_use_fused(q, attn_mask, dropout_p):
        mask = _key_padding_mask(attn_mask, q)
        seed = random.getrandbits(31) if dropout_p > 0.0 else 0
        # kernels are stride-aware in B/H/S (head dim must be contiguous):
        # transposed views of the fused qkv projection pass through with
        # ZERO copies
        def _ok(t):
            return t if t.stride(-1) == 1 else t.contiguous()
        return FusedAttentionFn.apply(_ok(q), _ok(k), _ok(v), scale, mask,
                                      float(dropout_p), seed)
    return torch.nn.functional.scaled_dot_product_attention(
        q, k, v, attn_mask=attn_mask, dropout_p=dropout_p, scale=scale)
