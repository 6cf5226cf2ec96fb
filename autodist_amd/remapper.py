"""Feed/fetch remapping between the user's global-batch view and this rank.

Reference behavior: autodist/remapper.py:29-313 — feeds with a polymorphic
batch dim are np.array_split across replicas (109-123); train-op fetches fan
to all replicas; tensor fetches come from the master replica or are
concatenated (125-185).

Process-per-GPU translation: the "replica index" is the rank. Feeds are
split along dim 0; fetched 0-dim tensors are averaged across ranks; fetched
batched tensors are all-gathered and concatenated so every rank returns the
global-batch result, matching the reference's session semantics.
"""
from typing import Any, Dict

import numpy as np
import torch
import torch.distributed as dist

from autodist_amd.parallel.comm import allgatherv


class Remapper:
    def __init__(self, rank: int, world_size: int, device: torch.device,
                 process_group=None):
        self.rank = rank
        self.world_size = world_size
        self.device = device
        self.process_group = process_group
        # n_local/N of the last feed split — np.array_split produces UNEVEN
        # shards when N % world != 0; gradients and scalar fetches must then
        # be weighted by batch fraction, not 1/world (the reference asserts
        # the weighted average: tests/integration/cases/c0.py:92-119)
        self.batch_fraction: float = 1.0 / max(world_size, 1)

    # -- feeds -------------------------------------------------------------
    def remap_feed(self, value: Any) -> Any:
        """Split one feed along the batch dim for this rank
        (reference _remap_feed, remapper.py:81-123)."""
        if isinstance(value, np.ndarray):
            if self.world_size > 1 and value.ndim > 0:
                total = value.shape[0]
                value = np.array_split(value, self.world_size)[self.rank]
                if total:
                    self.batch_fraction = value.shape[0] / total
            t = torch.from_numpy(np.ascontiguousarray(value))
            return t.to(self.device)
        if isinstance(value, torch.Tensor):
            if self.world_size > 1 and value.dim() > 0:
                total = value.shape[0]
                value = torch.tensor_split(value, self.world_size)[self.rank]
                if total:
                    self.batch_fraction = value.shape[0] / total
            return value.to(self.device)
        return value  # scalars / python objects duplicated

    def remap_feed_dict(self, feed_dict: Dict[str, Any]) -> Dict[str, Any]:
        # reset each step: a feed-less step has no uneven split, so its
        # gradients weight 1/world (not the previous step's fraction)
        self.batch_fraction = 1.0 / max(self.world_size, 1)
        return {k: self.remap_feed(v) for k, v in (feed_dict or {}).items()}

    # -- fetches -----------------------------------------------------------
    def remap_fetch(self, value: Any) -> Any:
        """Merge one fetched value across ranks (reference _remap_fetch,
        remapper.py:125-185): scalars -> mean, batched tensors -> concat."""
        if not isinstance(value, torch.Tensor):
            return value
        value = value.detach()
        if self.world_size <= 1:
            return value
        if value.dim() == 0:
            # batch-fraction-weighted mean: with an even split this is the
            # plain mean; with uneven np.array_split shards it matches the
            # global-batch scalar (e.g. the loss over the whole batch)
            out = value.clone() * self.batch_fraction
            dist.all_reduce(out, op=dist.ReduceOp.SUM, group=self.process_group)
            return out
        parts = allgatherv(value.contiguous(), self.world_size,
                           self.process_group)
        return torch.cat(parts, dim=0)

    def remap_fetches(self, fetched: Any) -> Any:
        if isinstance(fetched, (list, tuple)):
            return type(fetched)(self.remap_fetch(v) for v in fetched)
        if isinstance(fetched, dict):
            return {k: self.remap_fetch(v) for k, v in fetched.items()}
        return self.remap_fetch(fetched)
