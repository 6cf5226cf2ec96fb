"""Fused MFMA attention (forward / serving path).

`fused_sdpa(q, k, v, scale)` runs the gfx950 MFMA attention kernel
(ops/csrc/attention.hip) when applicable, else falls back to torch SDPA.
Applicability: inference (no grad), bf16, head_dim 64, S % 32 == 0, no
mask/dropout — i.e. the BERT/transformer SERVING path. Training keeps torch
SDPA (the fused backward is round-2 work, docs/ROADMAP.md).
"""
import math
from typing import Optional

import torch

from autodist_amd.ops import api as ops_api


def can_use_fused(q, attn_mask, dropout_p) -> bool:
    return (q.is_cuda and not torch.is_grad_enabled()
            and q.dtype == torch.bfloat16 and q.size(-1) == 64
            and q.size(-2) % 32 == 0 and attn_mask is None
            and dropout_p == 0.0 and ops_api.has_gpu_ops())


def fused_sdpa(q, k, v, attn_mask: Optional[torch.Tensor] = None,
               dropout_p: float = 0.0, scale: Optional[float] = None):
    """Drop-in for torch SDPA on [B, H, S, D] tensors."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.size(-1))
    if can_use_fused(q, attn_mask, dropout_p):
        return ops_api.ext().attn_fwd(q.contiguous(), k.contiguous(),
                                      v.contiguous(), scale)
    return torch.nn.functional.scaled_dot_product_attention(
        q, k, v, attn_mask=attn_mask, dropout_p=dropout_p, scale=scale)
