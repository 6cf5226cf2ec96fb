"""Random-axis partitioned AllReduce.

Reference behavior: random_axis_partition_all_reduce_strategy.py:118-141
(partition along a RANDOM axis whose dim > 1; sparse vars forced to axis 0 so
row-sharding keeps IndexedSlices index semantics).
"""
import random

from autodist_amd.proto.strategy_ir import Node
from autodist_amd.strategy.partitioned_all_reduce_strategy import PartitionedAR


def get_num_shards_and_axis(shape, max_shards, is_sparse, rng) -> tuple:
    """(num_shards, axis); axis random among dims > 1, sparse -> axis 0
    (reference random_axis_partition_all_reduce_strategy.py:118-141)."""
    if not shape:
        return 1, 0
    if is_sparse:
        axis = 0
    else:
        candidates = [i for i, d in enumerate(shape) if d > 1]
        if not candidates:
            return 1, 0
        axis = rng.choice(candidates)
    n = max(1, min(shape[axis], max_shards))
    return n, axis


class RandomAxisPartitionAR(PartitionedAR):
    """Random-axis partition + AR per shard."""

    def __init__(self, chunk_size=128, all_reduce_spec="RCCL",
                 compressor="NoneCompressor", max_shards=None,
                 min_partition_numel=16384, seed=1):
        super().__init__(chunk_size, all_reduce_spec, compressor,
                         max_shards, min_partition_numel)
        self._rng = random.Random(seed)

    def _gen_node_config(self, var_item, num_replicas: int) -> Node:
        if not var_item.shape or var_item.numel < self.min_partition_numel:
            return Node(var_name=var_item.name,
                        all_reduce_synchronizer=self._make_sync(self._next_group()))
        limit = self.max_shards or num_replicas
        n_shards, axis = get_num_shards_and_axis(
            var_item.shape, limit, var_item.is_sparse, self._rng)
        if n_shards <= 1:
            return Node(var_name=var_item.name,
                        all_reduce_synchronizer=self._make_sync(self._next_group()))
        counts = ["1"] * len(var_item.shape)
        counts[axis] = str(n_shards)
        parts = [Node(var_name=f"{var_item.name}/part_{i}",
                      all_reduce_synchronizer=self._make_sync(self._next_group()))
                 for i in range(n_shards)]
        return Node(var_name=var_item.name, partitioner=",".join(counts),
                    part_config=parts)
