"""Gradient buckets: flat fusion buffers for the collective path.

Reference context: TF fuses per-variable CollectiveReduce ops with the
ScopedAllocator optimizer (autodist/runner.py:40-46, groups from
all_reduce_strategy.py:61-66). The MI355X-native equivalent is explicit:
each strategy `group` maps to pre-allocated flat HIP buffers; parameter
.grad tensors are VIEWS into the flat buffer, so gradient "fusion" costs
zero copies; one RCCL all-reduce per bucket is issued on a dedicated comm
stream as soon as the bucket's last gradient lands (overlapping backward),
sized for the xGMI mesh (7 p2p links x ~153 GB/s per GPU).
"""
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from autodist_amd.const import DEFAULT_BUCKET_BYTES
from autodist_amd.parallel.compressor import Compressor
from autodist_amd.utils import logging


class Bucket:
    """One fusion group: a flat buffer + member params whose .grad are views."""

    def __init__(self, bucket_id: int, dtype: torch.dtype, device: torch.device,
                 compressor: Compressor):
        self.id = bucket_id
        self.dtype = dtype
        self.device = device
        self.compressor = compressor
        self.params: List[torch.nn.Parameter] = []
        self.shapes: List[torch.Size] = []
        self.offsets: List[int] = []
        self.numel = 0
        self.flat: Optional[torch.Tensor] = None
        self._ready = 0
        self._handle = None
        self.done_event: Optional[torch.cuda.Event] = None
        self._issued = False

    def add(self, param: torch.nn.Parameter):
        self.params.append(param)
        self.shapes.append(param.shape)
        self.offsets.append(self.numel)
        self.numel += param.numel()

    def allocate(self):
        """Allocate the flat buffer and point every member's .grad at a view."""
        self.flat = torch.zeros(self.numel, dtype=self.dtype, device=self.device)
        for p, off, shape in zip(self.params, self.offsets, self.shapes):
            p.grad = self.flat[off:off + p.numel()].view(shape)

    @property
    def nbytes(self) -> int:
        return self.numel * self.flat.element_size() if self.flat is not None else 0

    # -- per-step protocol -------------------------------------------------
    def reset(self):
        self._ready = 0
        self._handle = None
        self._issued = False
        self.done_event = None

    def zero_(self):
        self.flat.zero_()

    def mark_ready_and_maybe_issue(self, engine) -> bool:
        """Called from a post-accumulate-grad hook; issues the collective when
        the bucket's last member gradient has been produced."""
        self._ready += 1
        if self._ready < len(self.params) or self._issued:
            return False
        self.issue(engine)
        return True

    def issue(self, engine):
        self._issued = True
        if engine.world_size <= 1:
            return
        if engine.device.type == "cuda":
            ev = torch.cuda.Event()
            ev.record()  # grads complete on the compute (current) stream
            engine.comm_stream.wait_event(ev)
            with torch.cuda.stream(engine.comm_stream):
                self._reduce(engine)
            self.done_event = torch.cuda.Event()
            self.done_event.record(engine.comm_stream)
        else:
            self._reduce(engine)

    def _reduce(self, engine):
        if not engine.avg_supported:
            self.flat.mul_(1.0 / engine.world_size)
        self._handle = self.compressor.reduce(
            self.flat, group=engine.process_group, async_op=True)

    def finalize(self, engine):
        """Make the compute stream depend on this bucket's reduced result."""
        if engine.world_size <= 1 or not self._issued:
            return
        if engine.device.type == "cuda":
            with torch.cuda.stream(engine.comm_stream):
                self.compressor.finalize(self.flat, self._handle)
                if engine.avg_supported_needs_post_div:
                    self.flat.mul_(1.0 / engine.world_size)
            self.done_event = torch.cuda.Event()
            self.done_event.record(engine.comm_stream)
            torch.cuda.current_stream().wait_event(self.done_event)
        else:
            self.compressor.finalize(self.flat, self._handle)
            if engine.avg_supported_needs_post_div:
                self.flat.mul_(1.0 / engine.world_size)


def build_buckets(items, device: torch.device,
                  bucket_bytes: int = DEFAULT_BUCKET_BYTES) -> List[Bucket]:
    """Group (param, group_id, compressor_type) triples into Buckets.

    Keeps the strategy's group ids (one ScopedAllocator-group == >=1 buckets),
    splitting any group larger than bucket_bytes so collectives overlap with
    backward instead of waiting for one giant buffer.

    Bucket order follows REVERSED registration order within each group, since
    autograd produces gradients roughly in reverse forward order — the first
    bucket to fill is the one holding the last layers.
    """
    by_group: Dict[tuple, list] = {}
    for param, group_id, comp_type in items:
        by_group.setdefault((group_id, param.dtype), []).append((param, comp_type))
    buckets: List[Bucket] = []
    for (group_id, dtype), members in sorted(by_group.items(),
                                             key=lambda kv: kv[0][0]):
        members = list(reversed(members))
        current = None
        elt = torch.empty((), dtype=dtype).element_size()
        for param, comp_type in members:
            if current is None or current.numel * elt >= bucket_bytes:
                comp = Compressor.create(comp_type, f"bucket{len(buckets)}")
                current = Bucket(len(buckets), dtype, device, comp)
                buckets.append(current)
            current.add(param)
        logging.debug("group %s -> %d bucket(s)", group_id, len(buckets))
    for b in buckets:
        b.allocate()
    return buckets
