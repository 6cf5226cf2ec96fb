"""GraphTransformer — the compile seam between a Strategy and a running
engine.

Reference behavior: autodist/kernel/graph_transformer.py:55-92 orchestrates
Partition -> Replicate -> in-graph sync -> between-graph sync over a TF
GraphDef. The MI355X design has no serialized graph to rewrite — the
"transformation" is the construction of the engine's execution plan:

  phase 0 (original):      captured GraphItem (vars/optimizer/models)
  phase 1 (partition):     per-variable shard slices (parallel/partitioner)
  phase 2 (replication):   one process per GPU replaces in-graph replicas
                           (replicator.py:73-139): rank mapping + initial
                           parameter broadcast
  phase 3 (synchronizers): flat buckets + RCCL collectives, PS masters,
                           hooks — DistributedEngine.setup()

Each phase emits a plan snapshot (utils/visualization_util), mirroring the
reference's per-phase TensorBoard dumps (graph_transformer.py:62-90).
"""
from autodist_amd.parallel.engine import DistributedEngine
from autodist_amd.utils import logging, visualization_util


class GraphTransformer:
    def __init__(self, compiled_strategy, graph_item, rank=None,
                 world_size=None, device=None, dump_graphs=False):
        self._strategy = compiled_strategy
        self._graph_item = graph_item
        self._rank = rank
        self._world_size = world_size
        self._device = device
        self._dump = dump_graphs

    def transform(self) -> DistributedEngine:
        if self._dump:
            visualization_util.log_graph(
                "0-original", self._graph_item.serialize_to_string())
        engine = DistributedEngine(self._graph_item, self._strategy,
                                   rank=self._rank,
                                   world_size=self._world_size,
                                   device=self._device)
        engine.setup()
        if self._dump:
            visualization_util.log_graph(
                "3-transformed", visualization_util.describe_plan(engine))
        logging.info("graph transformed: %d buckets, %d var plans",
                     len(engine.buckets), len(engine.var_plans))
        return engine

    @property
    def num_local_replicas(self) -> int:
        """Process-per-GPU: one replica per rank (reference
        graph_transformer.py:113-118's per-worker replica count)."""
        return 1
