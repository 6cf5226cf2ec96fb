"""Strategy IR round-trip + builder behavior tests.

Reference: tests/test_strategy_base.py (proto round-trip) and the builder
semantics documented in SURVEY.md §2.1.
"""
import torch

from autodist_amd.graph_item import GraphItem
from autodist_amd.proto.strategy_ir import (AllReduceSpec, CompressorType,
                                            StrategyProto)
from autodist_amd.resource_spec import ResourceSpec
from autodist_amd.strategy import (AllReduce, AutoStrategy, Parallax,
                                   PartitionedAR, PartitionedPS, PS,
                                   PSLoadBalancing, RandomAxisPartitionAR,
                                   Strategy, UnevenPartitionedPS)


def _graph_item(shapes=((10, 4), (4,), (30, 8)), sparse_idx=None):
    g = GraphItem()

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            for i, s in enumerate(shapes):
                self.register_parameter(
                    f"p{i}", torch.nn.Parameter(torch.zeros(*s)))

    g.extend_model(M())
    names = list(g.variables)
    if sparse_idx is not None:
        g.mark_sparse(names[sparse_idx])
    return g


def test_strategy_serialize_roundtrip(tmp_path):
    g = _graph_item()
    rs = ResourceSpec()
    s = AllReduce(chunk_size=2).build(g, rs)
    path = s.serialize(str(tmp_path / "s1"))
    s2 = Strategy.deserialize(path=path)
    assert s2.proto.to_dict() == s.proto.to_dict()
    # JSON text form parses back equal
    s3 = Strategy(StrategyProto.parse_from_string(str(s.proto)))
    assert s3.proto.to_dict() == s.proto.to_dict()


def test_allreduce_groups():
    g = _graph_item(shapes=[(4, 4)] * 5)
    s = AllReduce(chunk_size=2).build(g, ResourceSpec())
    groups = [n.all_reduce_synchronizer.group for n in s.node_config]
    assert groups == [0, 0, 1, 1, 2]
    assert all(n.all_reduce_synchronizer.spec == AllReduceSpec.RCCL
               for n in s.node_config)


def test_allreduce_compressor():
    g = _graph_item()
    s = AllReduce(chunk_size=8, compressor="HorovodCompressorEF").build(
        g, ResourceSpec())
    assert all(n.all_reduce_synchronizer.compressor ==
               CompressorType.HorovodCompressorEF for n in s.node_config)


def test_ps_single_destination(tmp_gpu_resource_spec):
    g = _graph_item()
    rs = ResourceSpec(tmp_gpu_resource_spec)
    s = PS().build(g, rs)
    assert len(s.graph_config.replicas) == 8
    dests = {n.ps_synchronizer.reduction_destination for n in s.node_config}
    assert len(dests) == 1  # single PS
    assert all(n.ps_synchronizer.sync for n in s.node_config)


def test_ps_load_balancing(tmp_gpu_resource_spec):
    # Large vars should spread across multiple PS devices by byte size.
    g = _graph_item(shapes=[(100, 10), (100, 10), (100, 10), (100, 10)])
    rs = ResourceSpec(tmp_gpu_resource_spec)
    s = PSLoadBalancing().build(g, rs)
    dests = [n.ps_synchronizer.reduction_destination for n in s.node_config]
    assert len(set(dests)) == 4  # greedy least-loaded gives distinct devices


def test_partitioned_ps_shards(tmp_gpu_resource_spec):
    g = _graph_item(shapes=[(10, 4), (7, 2), (1,)])
    rs = ResourceSpec(tmp_gpu_resource_spec)
    s = PartitionedPS().build(g, rs)
    by_name = {n.var_name: n for n in s.node_config}
    n0 = by_name["p0"]  # dim0=10 -> smallest divisor 2
    assert n0.partitioner == "2,1"
    assert len(n0.part_config) == 2
    assert n0.part_config[0].var_name == "p0/part_0"
    n1 = by_name["p1"]  # dim0=7 prime, <= 8 devices: divisor 7
    assert n1.partition_count == 7
    n2 = by_name["p2"]  # scalar-ish: no partition
    assert not n2.partitioner and n2.ps_synchronizer is not None


def test_uneven_partitioned_ps(tmp_gpu_resource_spec):
    g = _graph_item(shapes=[(10, 4)])
    rs = ResourceSpec(tmp_gpu_resource_spec)
    s = UnevenPartitionedPS().build(g, rs)
    # smallest non-divisor of 10 is 3
    assert s.node_config[0].partitioner == "3,1"


def test_partitioned_ar(tmp_gpu_resource_spec):
    g = _graph_item(shapes=[(128, 256), (4,)])
    rs = ResourceSpec(tmp_gpu_resource_spec)
    s = PartitionedAR().build(g, rs)
    n0 = s.node_config[0]
    assert n0.partition_count == 8  # min(dim0, num_replicas)
    groups = [p.all_reduce_synchronizer.group for p in n0.part_config]
    assert len(set(groups)) == 8  # each shard its own group (pipelining)
    n1 = s.node_config[1]  # small var: not partitioned
    assert n1.all_reduce_synchronizer is not None


def test_random_axis_partition_ar(tmp_gpu_resource_spec):
    g = _graph_item(shapes=[(1, 64, 512)], sparse_idx=None)
    rs = ResourceSpec(tmp_gpu_resource_spec)
    s = RandomAxisPartitionAR(seed=3).build(g, rs)
    n = s.node_config[0]
    counts = [int(x) for x in n.partitioner.split(",")]
    assert counts[0] == 1  # axis 0 has dim 1, can't be chosen
    assert sum(c > 1 for c in counts) == 1


def test_random_axis_sparse_forced_axis0(tmp_gpu_resource_spec):
    g = _graph_item(shapes=[(100, 200)], sparse_idx=0)
    rs = ResourceSpec(tmp_gpu_resource_spec)
    s = RandomAxisPartitionAR(seed=3).build(g, rs)
    assert s.node_config[0].partition_axis == 0


def test_parallax_hybrid(tmp_gpu_resource_spec):
    g = _graph_item(shapes=[(50, 8), (100, 16), (4,)], sparse_idx=1)
    rs = ResourceSpec(tmp_gpu_resource_spec)
    s = Parallax().build(g, rs)
    by_name = {n.var_name: n for n in s.node_config}
    assert by_name["p0"].all_reduce_synchronizer is not None
    assert by_name["p1"].ps_synchronizer is not None  # sparse -> PS
    assert by_name["p2"].all_reduce_synchronizer is not None


def test_auto_strategy_picks_something(tmp_gpu_resource_spec):
    g = _graph_item(shapes=[(512, 512), (512,)])
    rs = ResourceSpec(tmp_gpu_resource_spec)
    s = AutoStrategy().build(g, rs)
    assert len(s.node_config) == 2


def test_compiler_device_resolver():
    """Device-string resolution hook (reference DeviceResolver,
    kernel/device/resolver.py:47-67: "ip:GPU:k" -> runtime device)."""
    from autodist_amd.strategy.base import StrategyCompiler
    g = _graph_item(shapes=[(8, 4)])
    rs = ResourceSpec()
    s = PS().build(g, rs)
    s.graph_config.replicas = ["10.0.0.1:GPU:0", "10.0.0.1:GPU:1"]
    s.node_config[0].ps_synchronizer.reduction_destination = "10.0.0.1:GPU:1"

    def resolver(dev):  # ip:GPU:k -> rank string
        return f"rank:{dev.rsplit(':', 1)[1]}"

    out = StrategyCompiler(g).set_device_resolver(resolver).compile(s)
    assert out.graph_config.replicas == ["rank:0", "rank:1"]
    assert out.node_config[0].ps_synchronizer.reduction_destination == "rank:1"
    # original strategy untouched (compile works on a copy)
    assert s.graph_config.replicas[0] == "10.0.0.1:GPU:0"


def test_compiler_prunes_unknown_vars():
    from autodist_amd.strategy.base import StrategyCompiler
    g = _graph_item(shapes=[(4, 4)])
    s = AllReduce().build(g, ResourceSpec())
    from autodist_amd.proto.strategy_ir import Node
    s.node_config.append(Node(var_name="ghost"))
    out = StrategyCompiler(g).compile(s)
    assert [n.var_name for n in out.node_config] == ["p0"]
