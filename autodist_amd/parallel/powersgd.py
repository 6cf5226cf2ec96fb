"""PowerSGD low-rank gradient compression (Vogels et al., NeurIPS 2019).

Reference context: the reference ships PowerSGDCompressor as commented-out
stub code (autodist/kernel/synchronization/compressor.py:208-284). This is a
working MI355X implementation in the batched form: the bucket's flat
gradient is reshaped to a near-square matrix M, compressed as M ~= P @ Q^T
with rank-r factors, and only P/Q cross xGMI:

    M += error                  (error feedback)
    P = M @ Q ; all-reduce P ; orthonormalize P
    Q = M^T @ P ; all-reduce Q
    hat = P @ Q^T / world ; error = M - hat ; flat <- hat

Wire bytes drop from n*m to r*(n+m) (~50-100x for ResNet-50 buckets). The
GEMMs are tall-skinny (r ~ 4..32); they run on rocBLAS (plain library GEMMs
— the framework's own hand-written kernels cover the fused/elementwise hot
path). Q's warm start + Gram-Schmidt follow the paper; the Q seed derives
from the bucket name so every rank starts identically.
"""
import hashlib
import math
from typing import Optional

import torch
import torch.distributed as dist

from autodist_amd.parallel.compressor import Compressor


def _orthonormalize(p: torch.Tensor, eps=1e-8):
    """Column-wise Gram-Schmidt (matches torch DDP's PowerSGD hook)."""
    for i in range(p.shape[1]):
        col = p[:, i:i + 1]
        col.div_(col.norm() + eps)
        if i + 1 < p.shape[1]:
            rest = p[:, i + 1:]
            rest.sub_(col @ (col.t() @ rest))
    return p


class PowerSGDCompressor(Compressor):
    def __init__(self, var_name: str = "", rank: int = 4,
                 warm_start: bool = True):
        super().__init__(var_name)
        self.rank = rank
        self.warm_start = warm_start
        self._error: Optional[torch.Tensor] = None
        self._q: Optional[torch.Tensor] = None
        self._pad: Optional[torch.Tensor] = None
        self._side = 0

    def _use_hip(self, flat) -> bool:
        from autodist_amd.ops import api as ops_api
        return (flat.is_cuda and flat.dtype == torch.float32
                and self.rank <= 16 and ops_api.has_gpu_ops())

    def _init(self, flat: torch.Tensor):
        numel = flat.numel()
        side = int(math.ceil(math.sqrt(numel)))
        # gfx950 MFMA kernels tile in 64s; rounding up costs <3% padding
        self._side = (side + 63) // 64 * 64
        self._error = torch.zeros_like(flat)
        self._pad = torch.zeros(self._side * self._side, dtype=flat.dtype,
                                device=flat.device)
        seed = int(hashlib.md5(self.var_name.encode()).hexdigest()[:8], 16)
        g = torch.Generator(device="cpu").manual_seed(seed)
        q = torch.randn(self._side, self.rank, generator=g)
        self._q = _orthonormalize(q.to(flat.device, flat.dtype))
        if self._use_hip(flat):
            self._qpad = torch.zeros(self._side, 16, device=flat.device)
            self._ppad = torch.zeros(self._side, 16, device=flat.device)

    def reduce(self, flat, group=None, async_op=False, scale=1.0):
        if self._error is None:
            self._init(flat)
        n = flat.numel()
        m = self._pad.view(self._side, self._side)
        if self._use_hip(flat):
            from autodist_amd.ops import api as ops_api
            ext = ops_api.ext()
            ext.psgd_add_err_pad(flat, self._error, self._pad)
            self._qpad[:, :self.rank].copy_(self._q)
            p = ext.psgd_mq(m, self._qpad)[:, :self.rank].contiguous()
        else:
            flat.add_(self._error)
            self._pad[:n].copy_(flat)
            p = m @ self._q
        dist.all_reduce(p, op=dist.ReduceOp.SUM, group=group)
        _orthonormalize(p)
        if self._use_hip(flat):
            from autodist_amd.ops import api as ops_api
            ext = ops_api.ext()
            self._ppad[:, :self.rank].copy_(p)
            q = ext.psgd_mtp(m, self._ppad)[:, :self.rank].contiguous()
        else:
            q = m.t() @ p
        handle = dist.all_reduce(q, op=dist.ReduceOp.SUM, group=group,
                                 async_op=async_op)
        self._p = p
        self._scale = scale
        if self.warm_start:
            self._q = q
        return (handle, q)

    def finalize(self, flat, handle) -> None:
        h, q = handle
        if h is not None:
            h.wait()
        n = flat.numel()
        if self._use_hip(flat):
            from autodist_amd.ops import api as ops_api
            self._qpad.zero_()
            self._qpad[:, :self.rank].copy_(q)
            m = self._pad.view(self._side, self._side)
            ops_api.ext().psgd_decompress_ef(flat, self._error, m,
                                             self._ppad, self._qpad,
                                             self._scale)
            return
        hat = (self._p @ q.t()).mul_(self._scale).view(-1)
        self._error.copy_(flat).sub_(hat[:n])
        flat.copy_(hat[:n])
