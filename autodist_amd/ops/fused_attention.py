"""Fused MFMA attention (gfx950) — forward + backward.

`fused_sdpa(q, k, v, ...)` runs the hand-written CDNA4 MFMA attention
kernels (ops/csrc/attention.hip forward; attention_bwd.hip FA2-style
backward, GPU-validated vs fp32 autograd) when applicable, else falls back
to torch SDPA. Applicability: bf16, head_dim 64, S % 32 == 0, no mask, no
dropout — the BERT/transformer hot path for both serving and training.
Training support goes through FusedAttentionFn (autograd.Function): the
backward recomputes softmax stats tile-by-tile (no S x S materialization,
flash-attention-2 dataflow) in two atomic-free kernels.
"""
import math
from typing import Optional

import torch

from autodist_amd.ops import api as ops_api


def can_use_fused(q, attn_mask, dropout_p) -> bool:
    """Kernel applicability (training AND inference since the backward was
    hardware-validated; reference context: the reference leaves attention
    to TF ops — hand-written CDNA4 here is the north-star requirement)."""
    return (q.is_cuda and q.dtype == torch.bfloat16 and q.size(-1) == 64
            and q.size(-2) % 32 == 0 and q.size(-2) >= 32
            and attn_mask is None and dropout_p == 0.0
            and ops_api.has_gpu_ops())


class FusedAttentionFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale):
        o = ops_api.ext().attn_fwd(q, k, v, scale)
        ctx.save_for_backward(q, k, v, o)
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, dout):
        q, k, v, o = ctx.saved_tensors
        dq, dk, dv = ops_api.ext().attn_bwd(q, k, v, o, dout.contiguous(),
                                            ctx.scale)
        return dq, dk, dv, None


def fused_sdpa(q, k, v, attn_mask: Optional[torch.Tensor] = None,
               dropout_p: float = 0.0, scale: Optional[float] = None):
    """Drop-in for torch SDPA on [B, H, S, D] tensors."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.size(-1))
    if can_use_fused(q, attn_mask, dropout_p):
        return FusedAttentionFn.apply(q.contiguous(), k.contiguous(),
                                      v.contiguous(), scale)
    return torch.nn.functional.scaled_dot_product_attention(
        q, k, v, attn_mask=attn_mask, dropout_p=dropout_p, scale=scale)
