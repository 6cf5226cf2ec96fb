"""AutoStrategy: pick the best candidate strategy via the analytic cost model.

The reference's AutoSync work trains learned cost models offline
(autodist/simulator/dataset/README.md); this MI355X builder evaluates a
candidate set with the closed-form xGMI cost model and returns the cheapest —
the same "strategy is model-dependent" headline behavior
(docs/usage/performance.md:14) without requiring the offline dataset.
"""
from autodist_amd.simulator.cost_model import CostModel
from autodist_amd.strategy.all_reduce_strategy import AllReduce
from autodist_amd.strategy.base import Strategy, StrategyBuilder
from autodist_amd.strategy.parallax_strategy import Parallax
from autodist_amd.strategy.partitioned_all_reduce_strategy import PartitionedAR
from autodist_amd.strategy.partitioned_ps_strategy import PartitionedPS
from autodist_amd.strategy.ps_lb_strategy import PSLoadBalancing
from autodist_amd.utils import logging


class AutoStrategy(StrategyBuilder):
    """Evaluate candidates with the cost model, return the cheapest."""

    def __init__(self, candidates=None):
        self._candidates = candidates or [
            AllReduce(), PartitionedAR(), Parallax(),
            PSLoadBalancing(), PartitionedPS(),
        ]

    def build(self, graph_item, resource_spec) -> Strategy:
        model = CostModel(resource_spec)
        best, best_cost, best_name = None, float("inf"), ""
        for builder in self._candidates:
            try:
                s = builder.build(graph_item, resource_spec)
                cost = model.estimate(s, graph_item)
            except Exception as exc:  # noqa: BLE001 - candidate may not apply
                logging.debug("AutoStrategy candidate %s failed: %s",
                              type(builder).__name__, exc)
                continue
            logging.info("AutoStrategy candidate %s estimated %.3f ms/step",
                         type(builder).__name__, cost * 1e3)
            if cost < best_cost:
                best, best_cost, best_name = s, cost, type(builder).__name__
        if best is None:
            raise RuntimeError("no AutoStrategy candidate succeeded")
        logging.info("AutoStrategy selected %s (%.3f ms/step est.)",
                     best_name, best_cost * 1e3)
        return best
