"""AllReduce synchronizer — bucketed RCCL collectives over xGMI.

Reference behavior: autodist/kernel/synchronization/all_reduce_synchronizer.py
(dense grads -> per-replica collective_ops.all_reduce via a Compressor,
102-130; sparse grads -> two collective all_gathers for indices/values,
132-173; between_graph_apply is a no-op, 199-201).

MI355X-native mapping:
  * dense whole vars   -> flat fusion Buckets (parallel/buckets.py), one
                          ncclAllReduce per bucket on the comm stream, issued
                          from post-accumulate-grad hooks (overlaps backward)
  * partitioned shards -> ShardReducer per shard (its own group => its own
                          collective, pipelining shard transfers)
  * sparse vars        -> allgatherv(indices) + allgatherv(values) +
                          segment-coalesce + rowwise apply (parallel/comm.py,
                          engine._sync_and_apply_sparse)
  * between-graph      -> nothing (collectives ARE the cross-replica sync)

This module re-exports the pieces so the reference's component map is
explicit; the mechanics live in buckets.py / engine.py.
"""
from autodist_amd.parallel.buckets import Bucket, build_buckets  # noqa: F401
from autodist_amd.parallel.comm import allgather_sparse, coalesce_rows  # noqa: F401


class AllReduceSynchronizer:
    """in_graph_apply: registration of a var into the collective machinery
    (done by DistributedEngine._build_buckets_and_hooks); between_graph_apply:
    no-op (reference all_reduce_synchronizer.py:199-201)."""

    @staticmethod
    def between_graph_apply(*_a, **_k):
        return None
