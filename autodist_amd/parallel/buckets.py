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

from autodist_amd.const import DEFAULT_BUCKET_BYTES, DEFAULT_FIRST_BUCKET_BYTES
from autodist_amd.parallel.compressor import Compressor
from autodist_amd.utils import logging


class Bucket:
    """One fusion group: flat grad + flat param (+ flat optimizer state)
    buffers; member params and their .grad are VIEWS into the flats.

    Flattening params too means the whole bucket's optimizer update is ONE
    elementwise HIP kernel over contiguous HBM3E buffers — no multi-tensor
    metadata, perfectly coalesced 16 B/lane access."""

    def __init__(self, bucket_id: int, dtype: torch.dtype, device: torch.device,
                 compressor: Compressor, cls_name: str = "", hyper: dict = None):
        self.id = bucket_id
        self.dtype = dtype
        self.device = device
        self.compressor = compressor
        self.cls_name = cls_name          # optimizer class for this bucket
        self.hyper = hyper or {}
        self.params: List[torch.nn.Parameter] = []
        self.shapes: List[torch.Size] = []
        self.offsets: List[int] = []
        self.numel = 0
        self.flat: Optional[torch.Tensor] = None        # gradients
        self.flat_param: Optional[torch.Tensor] = None  # parameter values
        self.state: dict = {}                           # flat optimizer state
        self._ready = 0
        self._handle = None
        self.done_event: Optional[torch.cuda.Event] = None
        self._issued = False
        # deterministic cross-rank issue key (engine assigns the md5-derived
        # value from member var names, parallel/collective_key.py)
        self.instance_key = bucket_id
        self.group_key = 0

    def add(self, param: torch.nn.Parameter):
        self.params.append(param)
        self.shapes.append(param.shape)
        self.offsets.append(self.numel)
        self.numel += param.numel()

    def allocate(self):
        """Allocate flat buffers; repoint every member's .data and .grad at
        views of them (values preserved)."""
        self.flat = torch.zeros(self.numel, dtype=self.dtype, device=self.device)
        self.flat_param = torch.empty(self.numel, dtype=self.dtype,
                                      device=self.device)
        with torch.no_grad():
            for p, off, shape in zip(self.params, self.offsets, self.shapes):
                n = p.numel()
                self.flat_param[off:off + n].view(shape).copy_(p.data)
                p.data = self.flat_param[off:off + n].view(shape)
                p.grad = self.flat[off:off + n].view(shape)

    @property
    def nbytes(self) -> int:
        return self.numel * self.flat.element_size() if self.flat is not None else 0

    def ensure_views(self):
        """Re-install flat-buffer grad views if user code detached them.

        model.zero_grad(set_to_none=True) bypasses the engine's patched
        zero_grad and sets .grad = None; the next backward then allocates a
        FRESH grad tensor that does not alias the flat buffer. Without this
        check the engine would all-reduce and apply the stale flat contents
        (silent corruption). Values are preserved by copying the fresh grad
        into the view before re-pointing."""
        for p, off, shape in zip(self.params, self.offsets, self.shapes):
            n = p.numel()
            view = self.flat[off:off + n].view(shape)
            g = p.grad
            if g is None:
                p.grad = view
            elif g.data_ptr() != view.data_ptr():
                view.copy_(g.detach())
                p.grad = view

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
        self.ensure_views()
        if not engine.collectives_active:
            return
        if engine.device.type == "cuda":
            ev = torch.cuda.Event()
            ev.record()  # grads complete on the compute (current) stream
            engine.comm_stream.wait_event(ev)
            t0 = None
            if engine.comm_sampling:
                t0 = torch.cuda.Event(enable_timing=True)
                t0.record(engine.comm_stream)
            with torch.cuda.stream(engine.comm_stream):
                self._reduce(engine)
            self.done_event = torch.cuda.Event()
            self.done_event.record(engine.comm_stream)
            self._t0 = t0
        else:
            self._reduce(engine)

    def _reduce(self, engine):
        self._handle = self.compressor.reduce(
            self.flat, group=engine.process_group, async_op=True,
            scale=engine.grad_scale())

    def finalize(self, engine):
        """Make the compute stream depend on this bucket's reduced result."""
        if not engine.collectives_active or not self._issued:
            return
        if engine.device.type == "cuda":
            with torch.cuda.stream(engine.comm_stream):
                self.compressor.finalize(self.flat, self._handle)
            self.done_event = torch.cuda.Event(
                enable_timing=engine.comm_sampling)
            self.done_event.record(engine.comm_stream)
            torch.cuda.current_stream().wait_event(self.done_event)
            if engine.comm_sampling and getattr(self, "_t0", None) is not None:
                engine.record_comm_sample(self.nbytes, self._t0,
                                          self.done_event)
                self._t0 = None
        else:
            self.compressor.finalize(self.flat, self._handle)


def build_buckets(items, device: torch.device,
                  bucket_bytes: int = DEFAULT_BUCKET_BYTES,
                  first_bucket_bytes: int = DEFAULT_FIRST_BUCKET_BYTES
                  ) -> List[Bucket]:
    """Group (param, group_id, compressor_type, cls_name, hyper, hyper_key)
    tuples into Buckets.

    Keeps the strategy's group ids (one ScopedAllocator-group == >=1 buckets),
    splitting any group larger than bucket_bytes so collectives overlap with
    backward instead of waiting for one giant buffer. Buckets are homogeneous
    in (dtype, optimizer class, hyperparams, compressor) so the whole bucket
    updates with one fused kernel and one compressor instance.

    Bucket order follows REVERSED registration order within each group, since
    autograd produces gradients roughly in reverse forward order — the first
    bucket to fill is the one holding the last layers. That first bucket is
    capped at first_bucket_bytes (DDP-style) so its collective launches early
    in backward instead of waiting for a full-size buffer.
    """
    by_group: Dict[tuple, list] = {}
    for param, group_id, comp_type, cls_name, hyper, hyper_key in items:
        key = (group_id, param.dtype, cls_name, hyper_key, comp_type)
        by_group.setdefault(key, []).append((param, comp_type, hyper))
    buckets: List[Bucket] = []
    # sort key stringifies the hyper tuple: values of mixed types (None vs
    # bool from torch param-group internals) are not mutually orderable
    for (group_id, dtype, cls_name, _, comp_type), members in sorted(
            by_group.items(),
            key=lambda kv: (kv[0][0], kv[0][2], repr(kv[0][3]),
                            repr(kv[0][4]))):
        members = list(reversed(members))
        current = None
        cap = 0
        elt = torch.empty((), dtype=dtype).element_size()
        for param, comp_type, hyper in members:
            if current is None or current.numel * elt >= cap:
                cap = first_bucket_bytes if not buckets else bucket_bytes
                comp = Compressor.create(comp_type, f"bucket{len(buckets)}")
                current = Bucket(len(buckets), dtype, device, comp,
                                 cls_name=cls_name, hyper=hyper)
                buckets.append(current)
            current.add(param)
        logging.debug("group %s -> %d bucket(s)", group_id, len(buckets))
    for b in buckets:
        b.allocate()
    return buckets
