"""Parallax hybrid strategy: dense grads -> AllReduce, sparse grads -> PS.

Reference behavior: autodist/strategy/parallax_strategy.py:38-71
(arXiv 1808.02621: dense gradients over collectives, sparse IndexedSlices
gradients to load-balanced parameter servers).

MI355X note: sparse embedding gradients go to shard owners over xGMI p2p
(no variable-length allgather on the dense path), dense gradients take the
bucketed RCCL all-reduce pipeline.
"""
from autodist_amd.proto.strategy_ir import (AllReduceSpec,
                                            AllReduceSynchronizer,
                                            CompressorType, Node,
                                            PSSynchronizer)
from autodist_amd.strategy.base import Strategy
from autodist_amd.strategy.ps_lb_strategy import PSLoadBalancing, byte_size_load_fn


class Parallax(PSLoadBalancing):
    """Hybrid per-variable AR/PS mixing (reference parallax_strategy.py:38-71)."""

    def __init__(self, chunk_size=128, local_proxy_variable=False, sync=True,
                 staleness=0, all_reduce_spec="RCCL", compressor="NoneCompressor"):
        super().__init__(local_proxy_variable, sync, staleness)
        self.chunk_size = chunk_size
        self.all_reduce_spec = all_reduce_spec
        self.compressor = compressor

    def build(self, graph_item, resource_spec) -> Strategy:
        strategy = Strategy()
        gpus = [k for k, _ in resource_spec.gpu_devices]
        strategy.graph_config.replicas = gpus or [
            k for k, _ in resource_spec.cpu_devices[:1]]
        ps_devices = gpus or [resource_spec.node_cpu_device(n).name_string()
                              for n in resource_spec.nodes]
        self.loads = {d: 0.0 for d in ps_devices}
        spec = AllReduceSpec["NCCL" if self.all_reduce_spec == "NCCL"
                             else self.all_reduce_spec]
        compressor = CompressorType[self.compressor]
        configs = []
        dense_idx = 0
        for v in graph_item.trainable_var_op_to_var.values():
            if v.is_sparse:
                device = min(self.loads, key=self.loads.get)
                self.loads[device] += byte_size_load_fn(v)
                configs.append(Node(var_name=v.name, ps_synchronizer=PSSynchronizer(
                    reduction_destination=device,
                    local_replication=self._local_proxy_variable,
                    sync=self._sync, staleness=self._staleness)))
            else:
                configs.append(Node(
                    var_name=v.name,
                    all_reduce_synchronizer=AllReduceSynchronizer(
                        spec=spec, compressor=compressor,
                        group=dense_idx // self.chunk_size)))
                dense_idx += 1
        strategy.node_config = configs
        return strategy
