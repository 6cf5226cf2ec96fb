"""Strategy builders (reference: autodist/strategy/)."""
from autodist_amd.strategy.all_reduce_strategy import AllReduce
from autodist_amd.strategy.auto_strategy import AutoStrategy
from autodist_amd.strategy.base import Strategy, StrategyBuilder, StrategyCompiler
from autodist_amd.strategy.parallax_strategy import Parallax
from autodist_amd.strategy.partitioned_all_reduce_strategy import PartitionedAR
from autodist_amd.strategy.partitioned_ps_strategy import (PartitionedPS,
                                                           UnevenPartitionedPS)
from autodist_amd.strategy.ps_lb_strategy import PSLoadBalancing
from autodist_amd.strategy.ps_strategy import PS
from autodist_amd.strategy.random_axis_partition_all_reduce_strategy import \
    RandomAxisPartitionAR

__all__ = [
    "AllReduce", "AutoStrategy", "Parallax", "PartitionedAR", "PartitionedPS",
    "PS", "PSLoadBalancing", "RandomAxisPartitionAR", "Strategy",
    "StrategyBuilder", "StrategyCompiler", "UnevenPartitionedPS",
]
