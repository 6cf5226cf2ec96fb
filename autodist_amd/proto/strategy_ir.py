"""Strategy IR — serializable per-variable synchronization plan.

Field-for-field re-derivation of the reference's protobuf schema
(autodist/proto/strategy.proto:29-69 and synchronizers.proto:26-57) as
dataclasses with a stable JSON wire format (protoc is not shipped in the
image; the message/field names are preserved so strategies remain
human-readable and round-trippable).

A Strategy assigns, per variable: a synchronizer (PS or AllReduce over RCCL),
an optional axis-partitioner, and the set of replica devices (one rank per
MI355X GPU).
"""
import dataclasses
import enum
import json
from typing import List, Optional


class AllReduceSpec(enum.IntEnum):
    """Collective implementation (synchronizers.proto:38-42). RCCL == NCCL on ROCm."""
    AUTO = 0
    RCCL = 1
    RING = 2
    # alias kept for reference-strategy compatibility
    NCCL = 1


class CompressorType(enum.IntEnum):
    """Gradient compression applied before all-reduce (synchronizers.proto:44-50)."""
    NoneCompressor = 0
    HorovodCompressor = 1
    HorovodCompressorEF = 2
    PowerSGDCompressor = 3


@dataclasses.dataclass
class PSSynchronizer:
    """Parameter-server sync config (synchronizers.proto:26-31)."""
    reduction_destination: str = ""   # DeviceSpec string of the PS owner
    local_replication: bool = False   # proxy-variable caching on each worker
    sync: bool = True                 # synchronous (True) vs async (False)
    staleness: int = 0                # bounded staleness k (sync must be True)

    def to_dict(self):
        return {"reduction_destination": self.reduction_destination,
                "local_replication": self.local_replication,
                "sync": self.sync, "staleness": self.staleness}

    @classmethod
    def from_dict(cls, d):
        return cls(**d)


@dataclasses.dataclass
class AllReduceSynchronizer:
    """All-reduce sync config (synchronizers.proto:36-57)."""
    spec: AllReduceSpec = AllReduceSpec.AUTO
    compressor: CompressorType = CompressorType.NoneCompressor
    group: int = 0                    # bucket/fusion group id

    def to_dict(self):
        return {"spec": int(self.spec), "compressor": int(self.compressor),
                "group": self.group}

    @classmethod
    def from_dict(cls, d):
        return cls(spec=AllReduceSpec(d.get("spec", 0)),
                   compressor=CompressorType(d.get("compressor", 0)),
                   group=d.get("group", 0))


@dataclasses.dataclass
class Node:
    """Per-variable (or per-shard) config (strategy.proto:44-63)."""
    var_name: str = ""
    ps_synchronizer: Optional[PSSynchronizer] = None
    all_reduce_synchronizer: Optional[AllReduceSynchronizer] = None
    partitioner: str = ""             # e.g. "2,1" = 2 shards along axis 0
    part_config: List["Node"] = dataclasses.field(default_factory=list)

    @property
    def synchronizer(self):
        return self.ps_synchronizer or self.all_reduce_synchronizer

    @property
    def partition_count(self) -> int:
        if not self.partitioner:
            return 0
        counts = [int(x) for x in self.partitioner.split(",") if x]
        n = 1
        for c in counts:
            n *= c
        return n

    @property
    def partition_axis(self) -> int:
        if not self.partitioner:
            return 0
        counts = [int(x) for x in self.partitioner.split(",") if x]
        for ax, c in enumerate(counts):
            if c > 1:
                return ax
        return 0

    def to_dict(self):
        d = {"var_name": self.var_name}
        if self.ps_synchronizer is not None:
            d["ps_synchronizer"] = self.ps_synchronizer.to_dict()
        if self.all_reduce_synchronizer is not None:
            d["all_reduce_synchronizer"] = self.all_reduce_synchronizer.to_dict()
        if self.partitioner:
            d["partitioner"] = self.partitioner
        if self.part_config:
            d["part_config"] = [p.to_dict() for p in self.part_config]
        return d

    @classmethod
    def from_dict(cls, d):
        return cls(
            var_name=d.get("var_name", ""),
            ps_synchronizer=PSSynchronizer.from_dict(d["ps_synchronizer"])
            if "ps_synchronizer" in d else None,
            all_reduce_synchronizer=AllReduceSynchronizer.from_dict(
                d["all_reduce_synchronizer"])
            if "all_reduce_synchronizer" in d else None,
            partitioner=d.get("partitioner", ""),
            part_config=[cls.from_dict(p) for p in d.get("part_config", [])],
        )


@dataclasses.dataclass
class GraphConfig:
    """Graph-level config: the data-parallel replica devices (strategy.proto:65-67)."""
    replicas: List[str] = dataclasses.field(default_factory=list)

    def to_dict(self):
        return {"replicas": list(self.replicas)}

    @classmethod
    def from_dict(cls, d):
        return cls(replicas=list(d.get("replicas", [])))


@dataclasses.dataclass
class StrategyProto:
    """Top-level strategy message (strategy.proto:29-42)."""
    id: str = ""
    path: str = ""
    node_config: List[Node] = dataclasses.field(default_factory=list)
    graph_config: GraphConfig = dataclasses.field(default_factory=GraphConfig)

    def to_dict(self):
        return {"id": self.id, "path": self.path,
                "node_config": [n.to_dict() for n in self.node_config],
                "graph_config": self.graph_config.to_dict()}

    @classmethod
    def from_dict(cls, d):
        return cls(id=d.get("id", ""), path=d.get("path", ""),
                   node_config=[Node.from_dict(n) for n in d.get("node_config", [])],
                   graph_config=GraphConfig.from_dict(d.get("graph_config", {})))

    def serialize_to_string(self) -> str:
        return json.dumps(self.to_dict(), indent=1, sort_keys=True)

    @classmethod
    def parse_from_string(cls, s: str) -> "StrategyProto":
        return cls.from_dict(json.loads(s))

    def __str__(self):
        return self.serialize_to_string()
