"""Fused NHWC BatchNorm(+residual+ReLU) — python surface.

The hot elementwise chain of the ResNet benchmark:
    bn -> (+ residual) -> relu          (forward)
    relu-mask -> bn-bwd -> dres         (backward)
runs as hand-written gfx950 kernels (csrc/bn_ops.hip): bf16-native (no
autocast fp32 round-trip), residual-add and ReLU folded into the normalize /
gradient passes. On CPU (CI) the same math runs via torch ops so numerics
tests can compare.

`FusedBatchNorm2d` is state_dict-compatible with nn.BatchNorm2d (same
parameter/buffer names).
"""
from typing import Optional

import torch

from autodist_amd.ops import api as ops_api


class _FusedBNFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var,
                momentum, eps, relu, residual, ws=None):
        use_hip = x.is_cuda and ops_api.has_gpu_ops()
        ctx.ws_bwd = ws[1] if ws is not None else None
        if use_hip:
            y, save_mean, save_rstd = ops_api.ext().bn_fwd_train(
                x, weight, bias, running_mean, running_var, residual,
                eps, momentum, relu, ws[0] if ws is not None else None)
        else:
            xf = x.float()
            dims = (0, 2, 3)
            m = xf.numel() // xf.size(1)
            mean = xf.mean(dims)
            var = xf.var(dims, unbiased=False)
            save_mean = mean
            save_rstd = (var + eps).rsqrt()
            with torch.no_grad():
                running_mean.mul_(1 - momentum).add_(mean, alpha=momentum)
                unbiased = var * m / max(m - 1, 1)
                running_var.mul_(1 - momentum).add_(unbiased, alpha=momentum)
            y = (xf - mean[None, :, None, None]) * save_rstd[None, :, None, None]
            y = y * weight[None, :, None, None] + bias[None, :, None, None]
            if residual is not None:
                y = y + residual.float()
            if relu:
                y = torch.relu(y)
            y = y.to(x.dtype)
        ctx.save_for_backward(x, y, save_mean, save_rstd, weight)
        ctx.relu = relu
        ctx.has_res = residual is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, y, save_mean, save_rstd, weight = ctx.saved_tensors
        relu, has_res = ctx.relu, ctx.has_res
        use_hip = x.is_cuda and ops_api.has_gpu_ops()
        if use_hip:
            out = ops_api.ext().bn_bwd(x, dy.contiguous(
                memory_format=torch.channels_last), y, save_mean, save_rstd,
                weight, relu, has_res, ctx.ws_bwd)
            dx, dweight, dbias = out[0], out[1], out[2]
            dres = out[3] if has_res else None
        else:
            dyf = dy.float()
            if relu:
                dyf = dyf * (y > 0).float()
            dres = dyf.to(dy.dtype) if has_res else None
            m = x.numel() // x.size(1)
            xf = x.float()
            xhat = (xf - save_mean[None, :, None, None]) * \
                save_rstd[None, :, None, None]
            dims = (0, 2, 3)
            sum_dz = dyf.sum(dims)
            sum_dzxh = (dyf * xhat).sum(dims)
            dxf = (weight * save_rstd)[None, :, None, None] * (
                dyf - sum_dz[None, :, None, None] / m
                - xhat * sum_dzxh[None, :, None, None] / m)
            dx = dxf.to(x.dtype)
            dweight, dbias = sum_dzxh, sum_dz
        return (dx, dweight, dbias, None, None, None, None, None, dres, None)


def fused_bn_train(x, weight, bias, running_mean, running_var,
                   momentum=0.1, eps=1e-5, relu=False,
                   residual: Optional[torch.Tensor] = None, ws=None):
    return _FusedBNFunction.apply(x, weight, bias, running_mean, running_var,
                                  momentum, eps, relu, residual, ws)


class FusedBatchNorm2d(torch.nn.Module):
    """Drop-in BatchNorm2d with optional fused ReLU and residual add.

    forward(x) applies bn [+relu]; forward_add(x, residual) applies
    bn(x) + residual -> relu (the Bottleneck tail).
    """

    def __init__(self, num_features, eps=1e-5, momentum=0.1, relu=False):
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.relu = relu
        self.weight = torch.nn.Parameter(torch.ones(num_features))
        self.bias = torch.nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.register_buffer("num_batches_tracked",
                             torch.tensor(0, dtype=torch.long))
        self._ws = None  # persistent kernel workspaces (not in state_dict)

    def _check(self, x):
        if x.is_cuda and not x.is_contiguous(memory_format=torch.channels_last):
            x = x.contiguous(memory_format=torch.channels_last)
        return x

    def forward(self, x, residual: Optional[torch.Tensor] = None):
        x = self._check(x)
        if residual is not None:
            residual = self._check(residual)
        if self.training:
            self.num_batches_tracked += 1
            if x.is_cuda and (self._ws is None
                              or self._ws[0].device != x.device):
                # layouts match BN_GM_MAX=512 in csrc/ext.hip
                self._ws = (
                    torch.empty(2 * 512 + 4, self.num_features,
                                device=x.device),
                    torch.empty(2 * 512 + 3, self.num_features,
                                device=x.device))
            return fused_bn_train(x, self.weight, self.bias,
                                  self.running_mean, self.running_var,
                                  self.momentum, self.eps, self.relu, residual,
                                  ws=self._ws if x.is_cuda else None)
        # eval: running-stat normalize (+add+relu)
        scale = self.weight * (self.running_var + self.eps).rsqrt()
        shift = self.bias - self.running_mean * scale
        y = x * scale[None, :, None, None].to(x.dtype) + \
            shift[None, :, None, None].to(x.dtype)
        if residual is not None:
            y = y + residual
        return torch.relu(y) if self.relu else y

    def forward_add(self, x, residual):
        return self.forward(x, residual)

    def extra_repr(self):
        return f"{self.num_features}, relu={self.relu}"
