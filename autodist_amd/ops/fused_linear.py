"""Linear with a proper bias-gradient kernel for bf16 streams.

torch autograd computes a bias gradient as a generic dim-0 reduction over
the bf16 dY (reduce_kernel<BFloat16>, measured ~0.35 TB/s — ~13% of the
BERT-base step across its ~50 biased Linears). FusedLinear keeps the
forward on F.linear (hipBLASLt with the bias folded into the GEMM
epilogue) but takes over the backward: dX and dW are the same hipBLASLt
GEMMs autograd would run, and db uses the col_sum HIP kernel
(ops/csrc/ln_ops.hip — coalesced column-block streaming + small partial
sum). Parameter names match nn.Linear; CPU / non-bf16 paths fall back to
the stock implementation.

Reference context: the reference leaves dense layers to TF's runtime
(SURVEY §2.3); this is hot-path native coverage beyond reference parity.
"""
import os

import torch

from autodist_amd.ops import api as ops_api

# debug override for tools/graph_probe.py: force the fused path INSIDE
# hipGraph capture to bisect the composed-replay fault
_FORCE_IN_GRAPH = os.environ.get("AUTODIST_FUSED_LINEAR_IN_GRAPH") == "1"


class _LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b):
        ctx.save_for_backward(x, w)
        return torch.nn.functional.linear(x, w, b)

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1])
        x2 = x.reshape(-1, x.shape[-1])
        dx = (dy @ w) if ctx.needs_input_grad[0] else None
        dw = (dy2.t() @ x2) if ctx.needs_input_grad[1] else None
        db = None
        if ctx.needs_input_grad[2]:
            db = ops_api.ext().col_sum(dy2.contiguous()).to(dy.dtype)
        return dx, dw, db


def fused_linear(x, weight, bias):
    """F.linear with the custom backward when on the bf16 GPU path.

    Disabled under hipGraph stream capture: the composed BERT graph
    (engine buckets + LN + attention + this backward) replays with a GPU
    memory fault on the current ROCm stack even though every piece
    captures fine in isolation (tools/graph_probe.py) — so captured
    steps record the stock F.linear path and eager/multi-rank steps get
    the fused bias-gradient kernel."""
    if (x.is_cuda and bias is not None and ops_api.has_gpu_ops()
            and (_FORCE_IN_GRAPH
                 or not torch.cuda.is_current_stream_capturing())
            and (x.dtype == torch.bfloat16
                 or torch.is_autocast_enabled())):
        if x.dtype != torch.bfloat16:
            x = x.to(torch.bfloat16)
        # explicit recorded casts: grads flow back to the fp32 params
        # through the cast backward (fp32 accumulate), matching autocast
        w = weight.to(x.dtype)
        b = bias.to(x.dtype)
        return _LinearFn.apply(x, w, b)
    return torch.nn.functional.linear(x, weight, bias)


class FusedLinear(torch.nn.Linear):
    def forward(self, x):
        return fused_linear(x, self.weight, self.bias)
