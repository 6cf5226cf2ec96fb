"""Partitioned PS: axis-0 shard each variable, load-balance shards over PS.

Reference behavior: autodist/strategy/partitioned_ps_strategy.py:70-135
(num_shards = smallest divisor > 1 of the first-axis dim, bounded by the
number of PS devices; shards greedily load-balanced by bytes).

MI355X note: shard ownership across all 8 GPUs spreads reduction traffic over
all 7 xGMI links of every GPU and shards optimizer state (each owner holds
only its shard's momentum/Adam state).
"""
from autodist_amd.proto.strategy_ir import Node, PSSynchronizer
from autodist_amd.strategy.base import Strategy, StrategyBuilder
from autodist_amd.strategy.ps_lb_strategy import byte_size_load_fn


def get_num_shards(dim0: int, max_shards: int) -> int:
    """Smallest divisor of dim0 in (1, max_shards]; 1 if none
    (reference partitioned_ps_strategy.py:125-135)."""
    if dim0 < 2 or max_shards < 2:
        return 1
    for i in range(2, max_shards + 1):
        if dim0 % i == 0:
            return i
    return 1


class PartitionedPS(StrategyBuilder):
    """Axis-0 partition + load-balanced PS placement
    (reference partitioned_ps_strategy.py:42-123)."""

    def __init__(self, local_proxy_variable=False, sync=True, staleness=0):
        self._local_proxy_variable = local_proxy_variable
        self._sync = sync
        self._staleness = staleness
        if staleness > 0:
            assert sync, "staleness is only valid for sync training"
        self.loads = {}

    def build(self, graph_item, resource_spec) -> Strategy:
        strategy = Strategy()
        gpus = [k for k, _ in resource_spec.gpu_devices]
        strategy.graph_config.replicas = gpus or [
            k for k, _ in resource_spec.cpu_devices[:1]]
        ps_devices = gpus or [resource_spec.node_cpu_device(n).name_string()
                              for n in resource_spec.nodes]
        self.loads = {d: 0.0 for d in ps_devices}
        strategy.node_config = [
            self._gen_node_config(v, len(ps_devices))
            for v in graph_item.trainable_var_op_to_var.values()]
        return strategy

    def _pick_device(self, load: float) -> str:
        device = min(self.loads, key=self.loads.get)
        self.loads[device] += load
        return device

    def _num_shards_for(self, var_item, num_ps: int) -> int:
        if not var_item.shape:
            return 1
        return get_num_shards(var_item.shape[0], num_ps)

    def _gen_node_config(self, var_item, num_ps: int) -> Node:
        n_shards = self._num_shards_for(var_item, num_ps)
        if n_shards <= 1:
            return Node(var_name=var_item.name, ps_synchronizer=PSSynchronizer(
                reduction_destination=self._pick_device(byte_size_load_fn(var_item)),
                local_replication=self._local_proxy_variable,
                sync=self._sync, staleness=self._staleness))
        # partitioner string "n,1,1,..." = n shards along axis 0
        partitioner = ",".join([str(n_shards)] + ["1"] * (len(var_item.shape) - 1))
        shard_load = byte_size_load_fn(var_item) / n_shards
        parts = []
        for i in range(n_shards):
            parts.append(Node(
                var_name=f"{var_item.name}/part_{i}",
                ps_synchronizer=PSSynchronizer(
                    reduction_destination=self._pick_device(shard_load),
                    local_replication=self._local_proxy_variable,
                    sync=self._sync, staleness=self._staleness)))
        return Node(var_name=var_item.name, partitioner=partitioner,
                    part_config=parts)


class UnevenPartitionedPS(PartitionedPS):
    """Uneven shards: num_shards = smallest NON-divisor, so shard sizes differ
    (reference uneven_partition_ps_strategy.py:125-135: `if n % i > 0: return i`)."""

    def _num_shards_for(self, var_item, num_ps: int) -> int:
        if not var_item.shape:
            return 1
        dim0 = var_item.shape[0]
        if dim0 < 2 or num_ps < 2:
            return 1
        for i in range(2, num_ps + 1):
            if dim0 % i > 0:
                return i if i <= dim0 else 1
        return 1
