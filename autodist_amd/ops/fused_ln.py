"""Fused bf16 LayerNorm (+ residual add) module for transformer streams.

torch's LayerNorm under bf16 autocast upcasts to fp32 and returns fp32,
so every transformer block pays bfloat16<->float32 copy kernels around
each LN and keeps an fp32 residual stream (measured: the cast/elementwise
cluster was ~13% of the BERT-base step). FusedLayerNorm keeps the
residual stream bf16 end to end: one gfx950 kernel reads bf16, folds the
residual add in, normalizes with fp32 row statistics, and writes bf16
(ops/csrc/ln_ops.hip; backward = dx kernel + chunked dgamma/dbeta
partials). Parameter names match nn.LayerNorm (weight/bias) so state
dicts are interchangeable.

Reference context: the reference delegates all normalization math to TF
ops (SURVEY §2.3 — zero native files); hand-written CDNA4 normalization
on the training hot path is the rebuild's north-star requirement, like
the fused BN trio (ops/csrc/bn_ops.hip) for the CNN side.
"""
from typing import Optional

import torch

from autodist_amd.ops import api as ops_api


class _FusedLNFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, res, weight, bias, eps):
        y, u, mean, rstd = ops_api.ext().ln_fwd(x, res, weight, bias, eps)
        ctx.save_for_backward(u, weight, mean, rstd)
        ctx.has_res = res is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        u, weight, mean, rstd = ctx.saved_tensors
        dx, dgamma, dbeta = ops_api.ext().ln_bwd(dy.contiguous(), u, weight,
                                                 mean, rstd)
        # d(x + res)/dx == d(x + res)/dres == identity
        return (dx, dx if ctx.has_res else None, dgamma, dbeta, None)


class FusedLayerNorm(torch.nn.Module):
    """Drop-in nn.LayerNorm with optional fused residual:
    forward(x, residual) == LayerNorm(x + residual).

    Under autocast the bf16 path keeps inputs AND outputs bf16 (fp32 row
    stats inside the kernel) — the standard bf16-residual-stream design —
    where torch LN would return fp32. CPU / non-bf16 inputs fall back to
    torch's layer_norm in fp32."""

    def __init__(self, hidden: int, eps: float = 1e-12):
        super().__init__()
        self.hidden = hidden
        self.eps = eps
        self.weight = torch.nn.Parameter(torch.ones(hidden))
        self.bias = torch.nn.Parameter(torch.zeros(hidden))

    def forward(self, x, residual: Optional[torch.Tensor] = None):
        if torch.is_autocast_enabled() and x.is_cuda:
            x = x.to(torch.bfloat16)
            if residual is not None:
                residual = residual.to(torch.bfloat16)
        if (x.is_cuda and x.dtype == torch.bfloat16
                and self.hidden <= 4096 and self.hidden % 2 == 0
                and ops_api.has_gpu_ops()):
            return _FusedLNFn.apply(
                x.contiguous(),
                residual.contiguous() if residual is not None else None,
                self.weight, self.bias, self.eps)
        t = x if residual is None else x + residual
        return torch.nn.functional.layer_norm(
            t, (self.hidden,), self.weight, self.bias, self.eps)

    def extra_repr(self):
        return f"{self.hidden}, eps={self.eps}, fused_bf16"
