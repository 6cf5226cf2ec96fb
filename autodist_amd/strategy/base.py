"""Strategy wrapper, abstract builder, and compiler.

Reference behavior: autodist/strategy/base.py:28-168.
"""
import os
from datetime import datetime

from autodist_amd.const import DEFAULT_SERIALIZATION_DIR
from autodist_amd.proto.strategy_ir import GraphConfig, StrategyProto
from autodist_amd.utils import logging


class Strategy:
    """A built strategy: id + per-variable node configs + replica devices
    (reference base.py:28-99)."""

    def __init__(self, strategy_proto: StrategyProto = None):
        self._proto = strategy_proto or StrategyProto(
            id=datetime.utcnow().strftime("%Y%m%d%H%M%S%f"))

    @property
    def id(self):
        return self._proto.id

    @property
    def path(self):
        return self._proto.path

    @property
    def node_config(self):
        return self._proto.node_config

    @node_config.setter
    def node_config(self, value):
        self._proto.node_config = value

    @property
    def graph_config(self) -> GraphConfig:
        return self._proto.graph_config

    @graph_config.setter
    def graph_config(self, value):
        self._proto.graph_config = value

    @property
    def proto(self):
        return self._proto

    def copy(self) -> "Strategy":
        return Strategy(StrategyProto.from_dict(self._proto.to_dict()))

    def serialize(self, path=None) -> str:
        """Write to DEFAULT_SERIALIZATION_DIR/<id> (reference base.py:78-87)."""
        if path is None:
            os.makedirs(DEFAULT_SERIALIZATION_DIR, exist_ok=True)
            path = os.path.join(DEFAULT_SERIALIZATION_DIR, self._proto.id)
        self._proto.path = path
        with open(path, "w", encoding="utf-8") as f:
            f.write(self._proto.serialize_to_string())
        return path

    @classmethod
    def deserialize(cls, strategy_id=None, path=None) -> "Strategy":
        """Load by id from the serialization dir (reference base.py:89-99)."""
        if path is None:
            path = os.path.join(DEFAULT_SERIALIZATION_DIR, strategy_id)
        with open(path, "r", encoding="utf-8") as f:
            return cls(StrategyProto.parse_from_string(f.read()))

    def __str__(self):
        return str(self._proto)


class StrategyBuilder:
    """Abstract builder: (graph_item, resource_spec) -> Strategy
    (reference base.py:102-117)."""

    def build(self, graph_item, resource_spec) -> Strategy:
        raise NotImplementedError


class StrategyCompiler:
    """Resolve device strings and prune stateless vars
    (reference base.py:120-168)."""

    def __init__(self, graph_item):
        self._graph_item = graph_item
        self._device_resolver = None

    def set_device_resolver(self, resolver):
        self._device_resolver = resolver
        return self

    def _prune_nodes(self, strategy: Strategy) -> Strategy:
        """Drop node configs for variables absent from the graph or not
        trainable (reference _prune_nodes, base.py:156-161)."""
        known = set(self._graph_item.trainable_var_op_to_var.keys())
        pruned = [n for n in strategy.node_config if n.var_name in known]
        dropped = [n.var_name for n in strategy.node_config if n.var_name not in known]
        if dropped:
            logging.debug("strategy compiler pruned stateless vars: %s", dropped)
        strategy.node_config = pruned
        return strategy

    def _resolve_devices(self, strategy: Strategy) -> Strategy:
        if self._device_resolver is None:
            return strategy
        strategy.graph_config.replicas = [
            self._device_resolver(d) for d in strategy.graph_config.replicas]

        def _resolve_node(node):
            if node.ps_synchronizer and node.ps_synchronizer.reduction_destination:
                node.ps_synchronizer.reduction_destination = self._device_resolver(
                    node.ps_synchronizer.reduction_destination)
            for part in node.part_config:
                _resolve_node(part)

        for node in strategy.node_config:
            _resolve_node(node)
        return strategy

    def compile(self, strategy: Strategy) -> Strategy:
        strategy = self._prune_nodes(strategy.copy())
        strategy = self._resolve_devices(strategy)
        return strategy
