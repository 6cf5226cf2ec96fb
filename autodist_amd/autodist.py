"""AutoDist — the single user-facing entry object.

Reference behavior: autodist/autodist.py:60-322. API surface kept:
  AutoDist(resource_spec_file, strategy_builder) -> .scope() ->
  .create_distributed_session() / .function(fn) / .build_strategy()

MI355X-native pipeline behind it (reference _build, autodist.py:139-150):
  1. GraphItem captures models/optimizers built under scope()   (§graph_item)
  2. chief builds the Strategy; workers load it by id            (§strategy)
  3. StrategyCompiler prunes/resolves                            (§strategy.base)
  4. DistributedEngine installs RCCL buckets / PS shards         (§parallel.engine)
  5. WrappedSession remaps feeds/fetches per rank                (§runner)

Process model: one rank per MI355X GPU. If launched under torchrun, ranks
already exist; otherwise the chief re-executes the user script once per
remaining GPU (runtime/coordinator.py), mirroring the reference's
chief/worker env protocol (AUTODIST_WORKER / AUTODIST_STRATEGY_ID,
reference autodist.py:40-41, 100-109).
"""
import os
from typing import Optional

from autodist_amd import const
from autodist_amd.const import ENV, is_chief
from autodist_amd.graph_item import GraphItem
from autodist_amd.parallel.engine import DistributedEngine
from autodist_amd.patch import PatchTorch
from autodist_amd.remapper import Remapper
from autodist_amd.resource_spec import ResourceSpec
from autodist_amd.runner import WrappedSession
from autodist_amd.runtime.coordinator import Coordinator, find_free_port
from autodist_amd.strategy.base import Strategy, StrategyCompiler
from autodist_amd.utils import logging

_default_autodist = None


def get_default_autodist() -> Optional["AutoDist"]:
    return _default_autodist


class _Scope:
    def __init__(self, ad):
        self._ad = ad
        self._gi_scope = None

    def __enter__(self):
        self._gi_scope = self._ad.graph_item.as_default()
        self._gi_scope.__enter__()
        PatchTorch.patch()
        return self._ad

    def __exit__(self, *exc):
        PatchTorch.unpatch()
        self._gi_scope.__exit__(*exc)
        return False


class AutoDist:
    """One AutoDist per process (reference set_default_autodist,
    autodist.py:46-51)."""

    def __init__(self, resource_spec_file: Optional[str] = None,
                 strategy_builder=None, world_size: Optional[int] = None):
        global _default_autodist
        if _default_autodist is not None and not ENV.AUTODIST_IS_TESTING.val:
            raise RuntimeError("Only one AutoDist instance is allowed per "
                               "process")
        _default_autodist = self
        spec_file = resource_spec_file or ENV.AUTODIST_RESOURCE_SPEC.val or None
        self._resource_spec = ResourceSpec(spec_file)
        if strategy_builder is None:
            from autodist_amd.strategy.ps_lb_strategy import PSLoadBalancing
            strategy_builder = PSLoadBalancing()
        self._strategy_builder = strategy_builder
        self.graph_item = GraphItem()
        self._requested_world = world_size
        self._coordinator = None
        self._session: Optional[WrappedSession] = None
        self._engine: Optional[DistributedEngine] = None
        self._cached_fn = None

    # -- public API --------------------------------------------------------
    def scope(self):
        """Capture scope (reference autodist.py:309-322)."""
        return _Scope(self)

    def build_strategy(self) -> Strategy:
        """Build (don't deploy) the strategy (reference build_strategy,
        autodist.py:91-98)."""
        self.graph_item.prepare()
        return self._strategy_builder.build(self.graph_item, self._resource_spec)

    def create_distributed_session(self) -> WrappedSession:
        """Compile + deploy + return the session (reference
        create_distributed_session, autodist.py:191-198)."""
        if self._session is None:
            self._build()
        return self._session

    def function(self, fn):
        """TF2-style stepping (reference autodist.py:269-289): wraps a step
        fn; feeds split per rank, fetches merged."""
        if self._cached_fn is not None:
            raise RuntimeError("only one autodist.function per scope "
                               "(reference autodist.py:281-283)")

        def run_fn(*args, **kwargs):
            sess = self.create_distributed_session()
            feeds = [sess.remapper.remap_feed(a) for a in args]
            out = fn(*feeds, **kwargs)
            return sess.remapper.remap_fetches(out)

        self._cached_fn = run_fn
        return run_fn

    @property
    def engine(self) -> Optional[DistributedEngine]:
        return self._engine

    @property
    def resource_spec(self) -> ResourceSpec:
        return self._resource_spec

    # -- build pipeline ----------------------------------------------------
    def _decide_world(self) -> int:
        if "WORLD_SIZE" in os.environ:
            return int(os.environ["WORLD_SIZE"])
        if self._requested_world is not None:
            return self._requested_world
        return max(self._resource_spec.num_gpus, 1)

    def _build_or_load_strategy(self) -> Strategy:
        """Chief builds + serializes; spawned workers load by id; torchrun
        workers rebuild deterministically (reference autodist.py:100-109)."""
        self.graph_item.prepare()
        if ENV.AUTODIST_STRATEGY_ID.val:
            return Strategy.deserialize(ENV.AUTODIST_STRATEGY_ID.val)
        strategy = self._strategy_builder.build(
            self.graph_item, self._resource_spec)
        if is_chief():
            strategy.serialize()
        return strategy

    def _compile_strategy(self, strategy: Strategy) -> Strategy:
        """Prune + device-resolve (reference _compile_strategy,
        autodist.py:111-118)."""
        compiled = StrategyCompiler(self.graph_item).compile(strategy)
        logging.debug("compiled strategy:\n%s", compiled)
        return compiled

    def _setup(self, strategy: Strategy, world: int):
        """Launch workers if this chief owns the launch (reference _setup,
        autodist.py:120-128). Single node: local subprocesses via the
        Coordinator; multi node: SSH launch via the Cluster with the strategy
        file shipped to every node (reference coordinator.py:84-88)."""
        if "RANK" in os.environ or world <= 1:
            return
        if not is_chief():
            return
        if self._resource_spec.num_nodes > 1:
            import sys
            from autodist_amd.runtime.cluster import SSHCluster
            cluster = SSHCluster(self._resource_spec)
            os.environ.update({
                "RANK": "0", "LOCAL_RANK": "0",
                "WORLD_SIZE": str(cluster.world_size),
                "MASTER_ADDR": cluster.master_addr,
                "MASTER_PORT": str(cluster.master_port),
            })
            strategy_path = strategy.serialize()
            for addr in self._resource_spec.nodes:
                if addr != self._resource_spec.chief:
                    cluster.remote_copy(addr, strategy_path,
                                        os.path.dirname(strategy_path))
            cluster.start([sys.executable] + sys.argv,
                          extra_env={"AUTODIST_STRATEGY_ID": strategy.id})
            self._coordinator = cluster
            return
        port = find_free_port()
        os.environ.update({
            "RANK": "0", "LOCAL_RANK": "0", "WORLD_SIZE": str(world),
            "MASTER_ADDR": const.DEFAULT_MASTER_ADDR,
            "MASTER_PORT": str(port),
        })
        self._coordinator = Coordinator(strategy, self._resource_spec)
        self._coordinator.launch_clients(world, port)

    def _build(self):
        from autodist_amd.parallel.graph_transformer import GraphTransformer
        strategy = self._build_or_load_strategy()
        world = self._decide_world()
        self._setup(strategy, world)
        compiled = self._compile_strategy(strategy)
        self._engine = GraphTransformer(compiled, self.graph_item).transform()
        remapper = Remapper(self._engine.rank, self._engine.world_size,
                            self._engine.device, self._engine.process_group)
        self._session = WrappedSession(self._engine, remapper, self.graph_item)
        logging.info("distributed session ready: rank=%d world=%d device=%s",
                     self._engine.rank, self._engine.world_size,
                     self._engine.device)


def _reset_default_autodist_for_tests():
    global _default_autodist
    _default_autodist = None
