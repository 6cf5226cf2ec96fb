"""Vocab-parallel output projection + cross-entropy.

Reference context: the LM1B example's 793k-row softmax weight is exactly
the class of variable the reference routes through the partitioner
(autodist/kernel/partitioner.py:577-602 rebuilds sharded reads); its
output projection stays a TF op over the partitioned variable. The
MI355X-native re-derivation shards the VOCABULARY dimension across ranks
and never materializes full [N, V] logits on any GPU:

  * rank r holds rows [start_r, end_r) of the projection weight (the same
    split_boundaries split as ShardedEmbedding, so a weight-tied LM shares
    one shard for input lookup and output projection),
  * each rank computes its logits shard [N, V_r] with one local GEMM
    (hipBLASLt bf16 under autocast),
  * the softmax statistics cross ranks with THREE tiny collectives
    (max, sum-exp, target-logit — all [N] floats over xGMI), instead of
    all-gathering [N, V] logits,
  * the backward is shard-local for dW/db (rows are exclusively owned —
    no gradient sync, optimizer state naturally sharded) plus one
    all-reduce of dHidden [N, D].

At world_size == 1 this degrades to a fused CE over the full weight.
"""
import os
from typing import Optional

import torch
import torch.distributed as dist

from autodist_amd.parallel.partitioner import split_boundaries


def _forced() -> bool:
    """AUTODIST_FORCE_COLLECTIVES: execute the sharded-CE collectives even
    at world 1 (1-GPU RCCL hardware validation; identity numerics)."""
    return os.environ.get("AUTODIST_FORCE_COLLECTIVES", "") in ("1", "True")


class _FusedCERows(torch.autograd.Function):
    """Per-row CE over bf16 logits via the gfx950 online-softmax kernels
    (ops/csrc/ce_ops.hip): forward streams the logits once keeping only
    per-row (max, log-sum-exp) — torch's log_softmax materializes and
    saves the full [N, V] log-probabilities (4 GB at LM1B shape)."""

    @staticmethod
    def forward(ctx, logits2d, targets):
        from autodist_amd.ops import api as ops_api
        loss_rows, m, l2s = ops_api.ext().ce_fwd(logits2d, targets)
        ctx.save_for_backward(logits2d, targets, m, l2s)
        return loss_rows

    @staticmethod
    def backward(ctx, dloss_rows):
        from autodist_amd.ops import api as ops_api
        logits2d, targets, m, l2s = ctx.saved_tensors
        dlogits = ops_api.ext().ce_bwd(logits2d, targets, m, l2s,
                                       dloss_rows.float())
        return dlogits, None


class _VocabParallelCE(torch.autograd.Function):
    """Mean cross-entropy over vocab-sharded logits."""

    @staticmethod
    def forward(ctx, hidden, shard_w, shard_b, targets, row_start, row_end,
                world, group):
        # local logits shard in the compute dtype (bf16 GEMM under autocast)
        h2 = hidden.reshape(-1, hidden.shape[-1])
        w = shard_w.to(h2.dtype)
        logits = h2 @ w.t()
        if shard_b is not None:
            logits = logits + shard_b.to(logits.dtype)
        comm = world > 1 or _forced()
        lf = logits.float()
        lmax = lf.max(dim=-1).values
        if comm:
            dist.all_reduce(lmax, op=dist.ReduceOp.MAX, group=group)
        sumexp = torch.exp(lf - lmax[:, None]).sum(dim=-1)
        t = targets.reshape(-1)
        local = (t >= row_start) & (t < row_end)
        idx = (t - row_start).clamp(0, row_end - row_start - 1)
        tgt_logit = torch.where(
            local, lf.gather(1, idx[:, None]).squeeze(1),
            torch.zeros((), dtype=lf.dtype, device=lf.device))
        if comm:
            dist.all_reduce(sumexp, op=dist.ReduceOp.SUM, group=group)
            dist.all_reduce(tgt_logit, op=dist.ReduceOp.SUM, group=group)
        loss = (torch.log(sumexp) + lmax - tgt_logit).mean()
        ctx.save_for_backward(h2, shard_w, logits, lmax, sumexp, t, local,
                              idx)
        ctx.has_bias = shard_b is not None
        ctx.world = world
        ctx.group = group
        ctx.hidden_shape = hidden.shape
        return loss

    @staticmethod
    def backward(ctx, dloss):
        h2, shard_w, logits, lmax, sumexp, t, local, idx = ctx.saved_tensors
        n = h2.shape[0]
        # dlogits = (softmax - onehot) * dloss / N, shard-local
        probs = torch.exp(logits.float() - lmax[:, None]) / sumexp[:, None]
        probs[local, idx[local]] -= 1.0
        probs = (probs * (dloss / n)).to(logits.dtype)
        dw = probs.t() @ h2                  # [V_r, D] exclusively owned
        db = probs.sum(dim=0) if ctx.has_bias else None
        dh = probs @ shard_w.to(probs.dtype)  # [N, D] needs cross-rank sum
        if ctx.world > 1 or _forced():
            dist.all_reduce(dh, op=dist.ReduceOp.SUM, group=ctx.group)
        dh = dh.view(ctx.hidden_shape)
        return (dh, dw.to(shard_w.dtype),
                db.to(shard_w.dtype) if db is not None else None,
                None, None, None, None, None)


class VocabParallelProjection(torch.nn.Module):
    """Row-sharded (vocab-dim) output projection with fused sharded CE.

    `tied_shard` reuses a ShardedEmbedding's shard parameter (weight
    tying); otherwise the module owns its shard. The shard (and bias
    shard) are marked exclusively-owned (`_autodist_shard_local`) so the
    engine applies updates locally with shard-local optimizer state and
    checkpoints reassemble the full tensor (SaveSliceInfo semantics)."""

    def __init__(self, vocab_size: int, dim: int, bias: bool = True,
                 tied_shard: Optional[torch.nn.Parameter] = None,
                 rank: Optional[int] = None,
                 world_size: Optional[int] = None, process_group=None):
        super().__init__()
        import os
        self.vocab_size = vocab_size
        self.dim = dim
        self.world_size = world_size if world_size is not None else int(
            os.environ.get("WORLD_SIZE", 1))
        self.rank = rank if rank is not None else int(
            os.environ.get("RANK", 0))
        self.process_group = process_group
        if vocab_size < self.world_size:
            raise ValueError(
                f"VocabParallelProjection: vocab {vocab_size} cannot shard "
                f"across {self.world_size} ranks")
        bounds = split_boundaries(vocab_size, self.world_size)
        self.row_start, self.row_end = bounds[self.rank]
        nrows = self.row_end - self.row_start
        if tied_shard is not None:
            assert tuple(tied_shard.shape) == (nrows, dim), \
                "tied shard must use the same split_boundaries split"
            self.weight = tied_shard   # shared Parameter (weight tying)
            self._tied = True
        else:
            w = torch.empty(nrows, dim)
            torch.nn.init.normal_(w, std=0.02)
            self.weight = torch.nn.Parameter(w)
            self.weight._autodist_shard_local = True
            self.weight._autodist_shard_range = (self.row_start,
                                                 self.row_end, vocab_size)
            self._tied = False
        if bias:
            self.bias = torch.nn.Parameter(torch.zeros(nrows))
            self.bias._autodist_shard_local = True
            self.bias._autodist_shard_range = (self.row_start, self.row_end,
                                               vocab_size)
        else:
            self.bias = None

    def loss(self, hidden: torch.Tensor, targets: torch.Tensor
             ) -> torch.Tensor:
        """Mean cross-entropy of the sharded projection (never builds full
        logits across ranks). world==1 routes through torch's fused CE
        (same math; avoids the custom path's fp32 logits materialization,
        measured 21% slower end-to-end on LM1B at 1 GPU)."""
        if self.world_size <= 1 and not _forced():
            h2 = hidden.reshape(-1, hidden.shape[-1])
            logits = torch.nn.functional.linear(
                h2, self.weight.to(h2.dtype),
                self.bias.to(h2.dtype) if self.bias is not None else None)
            t = targets.reshape(-1)
            from autodist_amd.ops import api as ops_api
            if (logits.is_cuda and logits.dtype == torch.bfloat16
                    and ops_api.has_gpu_ops()):
                return _FusedCERows.apply(logits.contiguous(), t).mean()
            return torch.nn.functional.cross_entropy(logits, t)
        return _VocabParallelCE.apply(hidden, self.weight, self.bias,
                                      targets, self.row_start, self.row_end,
                                      self.world_size, self.process_group)

    @torch.no_grad()
    def full_logits(self, hidden: torch.Tensor) -> torch.Tensor:
        """Materialized [**, V] logits (eval/serving; collective,
        non-differentiable — training uses .loss())."""
        h2 = hidden.reshape(-1, hidden.shape[-1])
        logits = h2 @ self.weight.to(h2.dtype).t()
        if self.bias is not None:
            logits = logits + self.bias.to(logits.dtype)
        if self.world_size > 1:
            from autodist_amd.parallel.comm import allgatherv
            parts = allgatherv(logits.t().contiguous(), self.world_size,
                               self.process_group)
            logits = torch.cat(parts, dim=0).t()
        return logits.view(*hidden.shape[:-1], self.vocab_size)

    def _save_to_state_dict(self, destination, prefix, keep_vars):
        if not self._tied:
            from autodist_amd.parallel.comm import allgatherv
            if self.world_size > 1:
                parts = allgatherv(self.weight.detach().contiguous(),
                                   self.world_size, self.process_group)
                destination[prefix + "weight"] = torch.cat(parts, 0).cpu()
            else:
                destination[prefix + "weight"] = self.weight.detach().cpu()
        if self.bias is not None:
            if self.world_size > 1:
                from autodist_amd.parallel.comm import allgatherv
                parts = allgatherv(self.bias.detach().contiguous(),
                                   self.world_size, self.process_group)
                destination[prefix + "bias"] = torch.cat(parts, 0).cpu()
            else:
                destination[prefix + "bias"] = self.bias.detach().cpu()

    def _load_from_state_dict(self, state_dict, prefix, local_metadata,
                              strict, missing_keys, unexpected_keys,
                              error_msgs):
        wkey, bkey = prefix + "weight", prefix + "bias"
        if not self._tied and wkey in state_dict:
            full = state_dict[wkey]
            with torch.no_grad():
                self.weight.copy_(full[self.row_start:self.row_end])
        if self.bias is not None and bkey in state_dict:
            with torch.no_grad():
                self.bias.copy_(
                    state_dict[bkey][self.row_start:self.row_end])

    def extra_repr(self):
        return (f"{self.vocab_size}, {self.dim}, rank={self.rank}/"
                f"{self.world_size}, rows=[{self.row_start},{self.row_end})"
                f"{', tied' if self._tied else ''}")
