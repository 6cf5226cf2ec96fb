"""ShardedEmbedding — row-sharded embedding table across ranks.

Reference context: the partitioner rebuilds sharded embeddings as
`embedding_lookup_v2` over a PartitionedVariable (partitioner.py:589-590),
with TF's runtime fetching remote shards. The MI355X-native equivalent keeps
shard s resident on rank s and exchanges ids/vectors with two all-to-alls
over xGMI (7 p2p links/GPU):

  fwd: route ids to owner ranks (all_to_all) -> local gather (HIP
       gather_rows kernel on gfx950) -> route vectors back (all_to_all)
  bwd: route output grads to owners -> local scatter-add into the shard
       gradient (HIP scatter_add_rows) — the shard gradient is LOCAL-ONLY
       (each rank exclusively owns its rows), so no further synchronization
       is needed and optimizer state is naturally sharded.

At world_size==1 this degrades to a plain local gather.
"""
from typing import Optional

import torch
import torch.distributed as dist

from autodist_amd.parallel.partitioner import split_boundaries


def _forced() -> bool:
    """AUTODIST_FORCE_COLLECTIVES: run the all-to-all exchange even at
    world 1 (1-GPU RCCL hardware validation; identity routing)."""
    import os
    return os.environ.get("AUTODIST_FORCE_COLLECTIVES", "") in ("1", "True")


def _alltoallv(parts, world, group, trailing_shape=()):
    """Exchange variable-length dim-0 chunks; parts[r] goes to rank r.
    Returns received chunks (list per source rank)."""
    send_sizes = torch.tensor([p.shape[0] for p in parts], dtype=torch.int64)
    dev = parts[0].device
    send_sizes = send_sizes.to(dev)
    recv_sizes = torch.zeros_like(send_sizes)
    dist.all_to_all_single(recv_sizes, send_sizes, group=group)
    recv_list = recv_sizes.tolist()
    send_list = [int(p.shape[0]) for p in parts]
    total_recv = sum(recv_list)
    flat_send = torch.cat(parts, dim=0)
    out = torch.empty((total_recv,) + tuple(trailing_shape),
                      dtype=flat_send.dtype, device=dev)
    dist.all_to_all_single(out, flat_send, recv_list, send_list, group=group)
    offs = [0]
    for n in recv_list:
        offs.append(offs[-1] + n)
    return [out[offs[r]:offs[r + 1]] for r in range(world)], recv_list


class _ShardedLookup(torch.autograd.Function):
    @staticmethod
    def forward(ctx, shard_weight, ids, module):
        world = module.world_size
        if world <= 1 and not _forced():
            ctx.save_for_backward(ids)
            ctx.module = module
            return _local_gather(shard_weight, ids.reshape(-1)).view(
                *ids.shape, module.dim)
        group = module.process_group
        flat_ids = ids.reshape(-1)
        owner = torch.bucketize(flat_ids, module.boundaries, right=True)
        order = torch.argsort(owner, stable=True)
        sorted_ids = flat_ids[order]
        counts = torch.bincount(owner, minlength=world)
        send_parts = list(torch.split(sorted_ids, counts.tolist()))
        recv_parts, recv_counts = _alltoallv(send_parts, world, group)
        # gather locally for each requester
        local_ids = torch.cat(recv_parts) - module.row_start
        vecs = _local_gather(shard_weight, local_ids)
        back_parts = list(torch.split(vecs, recv_counts))
        got_parts, _ = _alltoallv(back_parts, world, group,
                                  trailing_shape=(module.dim,))
        got = torch.cat(got_parts)
        # un-sort to original id order
        out = torch.empty_like(got)
        out[order] = got
        ctx.save_for_backward(flat_ids, owner, order, counts)
        ctx.module = module
        return out.view(*ids.shape, module.dim)

    @staticmethod
    def backward(ctx, grad_out):
        module = ctx.module
        world = module.world_size
        if world <= 1 and not _forced():
            (ids,) = ctx.saved_tensors
            g = torch.zeros_like(module.shard)
            _local_scatter_add(g, ids.reshape(-1),
                               grad_out.reshape(-1, module.dim))
            return g, None, None
        flat_ids, owner, order, counts = ctx.saved_tensors
        group = module.process_group
        g_flat = grad_out.reshape(-1, module.dim)
        g_sorted = g_flat[order]
        id_parts = list(torch.split(flat_ids[order], counts.tolist()))
        g_parts = list(torch.split(g_sorted, counts.tolist()))
        recv_ids, _ = _alltoallv(id_parts, world, group)
        recv_gs, _ = _alltoallv(g_parts, world, group,
                                trailing_shape=(module.dim,))
        g_shard = torch.zeros_like(module.shard)
        local_ids = torch.cat(recv_ids) - module.row_start
        _local_scatter_add(g_shard, local_ids, torch.cat(recv_gs))
        return g_shard, None, None


def _local_gather(weight, ids):
    if weight.is_cuda:
        from autodist_amd.ops import api as ops_api
        if ops_api.has_gpu_ops() and weight.dtype == torch.float32:
            return ops_api.ext().gather_rows(weight, ids)
    return weight.index_select(0, ids)


def _local_scatter_add(out, ids, vals):
    if out.is_cuda:
        from autodist_amd.ops import api as ops_api
        if ops_api.has_gpu_ops() and out.dtype == torch.float32 \
                and vals.dtype == torch.float32:
            ops_api.ext().scatter_add_rows(out, ids, vals.contiguous())
            return
    out.index_add_(0, ids, vals.to(out.dtype))


class ShardedEmbedding(torch.nn.Module):
    def __init__(self, num_embeddings: int, embedding_dim: int,
                 rank: Optional[int] = None, world_size: Optional[int] = None,
                 process_group=None):
        super().__init__()
        import os
        self.num_embeddings = num_embeddings
        self.dim = embedding_dim
        self.world_size = world_size if world_size is not None else int(
            os.environ.get("WORLD_SIZE", 1))
        self.rank = rank if rank is not None else int(
            os.environ.get("RANK", 0))
        self.process_group = process_group
        if num_embeddings < self.world_size:
            raise ValueError(
                f"ShardedEmbedding: {num_embeddings} rows cannot shard "
                f"across {self.world_size} ranks (use nn.Embedding for "
                f"tiny tables)")
        bounds = split_boundaries(num_embeddings, self.world_size)
        self.row_start, self.row_end = bounds[self.rank]
        # boundaries for bucketize: end of each shard except the last
        # (non-persistent: the state_dict carries the consolidated table only)
        self.register_buffer("boundaries", torch.tensor(
            [e for (_, e) in bounds[:-1]], dtype=torch.int64),
            persistent=False)
        shard = torch.empty(self.row_end - self.row_start, embedding_dim)
        torch.nn.init.normal_(shard, std=0.01)
        self.shard = torch.nn.Parameter(shard)
        # each rank exclusively owns its rows: no gradient sync needed
        self.shard._autodist_shard_local = True
        # checkpoint consolidation metadata (engine/saver use this to
        # reassemble full optimizer state — reference SaveSliceInfo analog)
        self.shard._autodist_shard_range = (self.row_start, self.row_end,
                                            num_embeddings)
        self.shard._autodist_shard_module = self

    # -- checkpoint integration: state_dict is nn.Embedding-compatible ------
    def _save_to_state_dict(self, destination, prefix, keep_vars):
        """Emit the CONSOLIDATED table as '<prefix>weight' so checkpoints are
        interchangeable with nn.Embedding (collective: every rank must be
        saving, which the AutoDist Saver guarantees)."""
        destination[prefix + "weight"] = self.full_weight().cpu()

    def _load_from_state_dict(self, state_dict, prefix, local_metadata,
                              strict, missing_keys, unexpected_keys,
                              error_msgs):
        key = prefix + "weight"
        if key in state_dict:
            full = state_dict[key]
            if tuple(full.shape) == (self.num_embeddings, self.dim):
                with torch.no_grad():
                    self.shard.copy_(full[self.row_start:self.row_end])
            else:
                error_msgs.append(
                    f"ShardedEmbedding {key}: expected "
                    f"{(self.num_embeddings, self.dim)}, got {tuple(full.shape)}")
            state_dict = dict(state_dict)
            del state_dict[key]
        elif strict:
            missing_keys.append(key)
        # also accept a raw per-rank 'shard' entry (legacy/per-rank ckpt)
        skey = prefix + "shard"
        if skey in state_dict and tuple(state_dict[skey].shape) == \
                tuple(self.shard.shape):
            with torch.no_grad():
                self.shard.copy_(state_dict[skey])

    def forward(self, ids):
        return _ShardedLookup.apply(self.shard, ids, self)

    def full_weight(self) -> torch.Tensor:
        """Assemble the full table (checkpoint/debug); collective call."""
        if self.world_size <= 1:
            return self.shard.detach()
        from autodist_amd.parallel.comm import allgatherv
        parts = allgatherv(self.shard.detach().contiguous(), self.world_size,
                           self.process_group)
        return torch.cat(parts, dim=0)

    def extra_repr(self):
        return (f"{self.num_embeddings}, {self.dim}, rank={self.rank}/"
                f"{self.world_size}, rows=[{self.row_start},{self.row_end})")
