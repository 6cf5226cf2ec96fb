"""Collective helpers: variable-length allgather and friends.

Reference context: sparse gradients synchronize via two collective_ops
all_gather calls on indices/values (all_reduce_synchronizer.py:132-173).
RCCL has no allgatherv; this implements size-exchange + padded allgather,
with the padding trimmed on unpack (the standard xGMI-friendly schedule:
one small int64 allgather + one large padded allgather).
"""
from typing import List, Tuple

import torch
import torch.distributed as dist


def allgatherv(tensor: torch.Tensor, world_size: int, group=None,
               force: bool = False) -> List[torch.Tensor]:
    """All-gather tensors whose dim-0 length differs per rank.

    Returns the per-rank tensors (views of one padded buffer, trimmed).
    `force` executes the real collectives even at world_size==1 (the 1-GPU
    RCCL validation mode, AUTODIST_FORCE_COLLECTIVES).
    """
    if world_size <= 1 and not force:
        return [tensor]
    n_local = torch.tensor([tensor.shape[0]], dtype=torch.int64,
                           device=tensor.device)
    sizes = torch.zeros(world_size, dtype=torch.int64, device=tensor.device)
    dist.all_gather_into_tensor(sizes, n_local, group=group)
    sizes_l = sizes.tolist()
    max_n = max(sizes_l) if sizes_l else 0
    if max_n == 0:
        return [tensor[:0] for _ in range(world_size)]
    pad_shape = (max_n,) + tuple(tensor.shape[1:])
    padded = torch.zeros(pad_shape, dtype=tensor.dtype, device=tensor.device)
    padded[:tensor.shape[0]] = tensor
    flat_in = padded.reshape(-1)
    out = torch.zeros(world_size * flat_in.numel(), dtype=tensor.dtype,
                      device=tensor.device)
    dist.all_gather_into_tensor(out, flat_in, group=group)
    out = out.view((world_size,) + pad_shape)
    return [out[r, :sizes_l[r]] for r in range(world_size)]


def allgather_sparse(indices: torch.Tensor, values: torch.Tensor,
                     world_size: int, group=None, force: bool = False
                     ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Gather a row-sparse gradient (indices [nnz], values [nnz, dim...])
    from all ranks; returns concatenated (indices, values)."""
    idx_parts = allgatherv(indices, world_size, group, force=force)
    val_parts = allgatherv(values, world_size, group, force=force)
    return torch.cat(idx_parts), torch.cat(val_parts)


def coalesce_rows(indices: torch.Tensor, values: torch.Tensor
                  ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Sum duplicate rows (the reference's sparse-accumulator dedup,
    ps_synchronizer.py:498-535). On gfx950 the segmented-reduce HIP kernel
    (ops/csrc/sparse_ops.hip) replaces the sort+index_add composite."""
    if indices.numel() == 0:
        return indices, values
    from autodist_amd.ops import api as ops_api
    if ops_api.has_gpu_ops() and indices.is_cuda:
        return ops_api.segment_coalesce(indices, values)
    uniq, inv = torch.unique(indices, sorted=True, return_inverse=True)
    out = torch.zeros((uniq.shape[0],) + tuple(values.shape[1:]),
                      dtype=values.dtype, device=values.device)
    out.index_add_(0, inv, values)
    return uniq, out
