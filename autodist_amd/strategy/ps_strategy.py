"""PS strategy: every variable synchronized through one parameter server.

Reference behavior: autodist/strategy/ps_strategy.py:37-76 (all vars -> PS on
the chief's first CPU device). MI355X re-derivation: on a single xGMI node the
natural reduction destination is the chief's GPU 0 — gradients move over p2p
xGMI links (~153 GB/s/link) instead of staging through host memory, and the
update runs on-device. The CPU destination is kept for CPU-only topologies.
"""
from autodist_amd.proto.strategy_ir import Node, PSSynchronizer
from autodist_amd.strategy.base import Strategy, StrategyBuilder


class PS(StrategyBuilder):
    """All variables on a single PS (reference ps_strategy.py:37-56)."""

    def __init__(self, local_proxy_variable=False, sync=True, staleness=0):
        self._local_proxy_variable = local_proxy_variable
        self._sync = sync
        self._staleness = staleness
        if staleness > 0:
            assert sync, "staleness is only valid for sync training"

    def build(self, graph_item, resource_spec) -> Strategy:
        strategy = Strategy()
        gpus = resource_spec.gpu_devices
        strategy.graph_config.replicas = [k for k, _ in gpus] or [
            k for k, _ in resource_spec.cpu_devices[:1]]
        if gpus:
            reduction_device = gpus[0][0]       # chief GPU 0
        else:
            reduction_device = resource_spec.node_cpu_device(
                resource_spec.chief).name_string()
        strategy.node_config = [
            self._gen_ps_node_config(name, reduction_device)
            for name in graph_item.trainable_var_op_to_var]
        return strategy

    def _gen_ps_node_config(self, var_name, reduction_device) -> Node:
        """Reference _gen_ps_node_config (ps_strategy.py:58-76)."""
        return Node(var_name=var_name, ps_synchronizer=PSSynchronizer(
            reduction_destination=reduction_device,
            local_replication=self._local_proxy_variable,
            sync=self._sync, staleness=self._staleness))
