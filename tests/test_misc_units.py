"""Small unit tests: collective keys, ENV flags, cost model, failure
detection (reference §5.2/§5.3: determinism by construction, fail-fast
coordinator)."""
import os
import subprocess
import sys
import textwrap

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_collective_keys_deterministic():
    from autodist_amd.parallel.collective_key import CollectiveKey
    k1 = CollectiveKey()
    k2 = CollectiveKey()
    # instance keys are content-addressed: identical across processes/ranks
    assert k1.generate_instance_key("layer1.weight") == \
        k2.generate_instance_key("layer1.weight")
    assert k1.generate_instance_key("a") != k1.generate_instance_key("b")
    # group keys increment per distinct device set
    g1 = k1.generate_group_key(["gpu0", "gpu1"])
    g2 = k1.generate_group_key(["gpu1", "gpu0"])  # order-insensitive
    assert g1 == g2
    assert k1.generate_group_key(["gpu0"]) != g1


def test_env_flags(monkeypatch):
    from autodist_amd.const import ENV, is_chief
    monkeypatch.delenv("AUTODIST_WORKER", raising=False)
    monkeypatch.delenv("RANK", raising=False)
    assert is_chief()
    monkeypatch.setenv("AUTODIST_WORKER", "10.0.0.2")
    assert not is_chief()
    assert ENV.AUTODIST_WORKER.val == "10.0.0.2"
    monkeypatch.delenv("AUTODIST_WORKER", raising=False)
    monkeypatch.setenv("RANK", "3")
    assert not is_chief()
    monkeypatch.setenv("AUTODIST_IS_TESTING", "True")
    assert ENV.AUTODIST_IS_TESTING.val is True


def test_cost_model_fit():
    from autodist_amd.simulator.cost_model import CostModel
    cm = CostModel()
    t1 = cm.allreduce_time(100e6, 8)
    # feed synthetic measurements implying lower efficiency
    samples = [(100e6, 8, cm.allreduce_time(100e6, 8) * 2)]
    cm.fit(samples)
    t2 = cm.allreduce_time(100e6, 8)
    assert t2 > t1
    # ps model: sharding owners speeds up the round trip
    assert cm.ps_round_trip_time(100e6, 8, owners=8) < \
        cm.ps_round_trip_time(100e6, 8, owners=1)


def test_engine_stats():
    import torch
    from autodist_amd.graph_item import GraphItem
    from autodist_amd.parallel.engine import DistributedEngine
    from autodist_amd.resource_spec import ResourceSpec
    from autodist_amd.strategy import Parallax
    torch.manual_seed(0)
    emb = torch.nn.Embedding(10, 4, sparse=True)
    lin = torch.nn.Linear(4, 2)
    model = torch.nn.ModuleDict({"e": emb, "l": lin})
    g = GraphItem()
    g.extend_model(model)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    g.extend_optimizer_info(opt)
    engine = DistributedEngine(g, Parallax().build(g, ResourceSpec()),
                               rank=0, world_size=1,
                               device=torch.device("cpu")).setup()
    s = engine.stats()
    assert s["n_buckets"] >= 1
    assert s["allreduce_bytes_per_step"] == (4 * 2 + 2) * 4  # lin w+b fp32
    assert s["sparse_vars"] == 1
    assert not s["fallback_user_optimizer"]
    engine.teardown()


@pytest.mark.integration
def test_coordinator_fail_fast(tmp_path):
    """A crashing worker must bring down the chief (reference
    coordinator.py:98-110: non-zero exit -> os._exit(1))."""
    script = tmp_path / "crash.py"
    script.write_text(textwrap.dedent(f"""
        import os, sys, time
        sys.path.insert(0, {REPO!r})
        import torch
        from autodist_amd import AutoDist
        from autodist_amd.strategy import AllReduce
        if os.environ.get("AUTODIST_WORKER"):
            sys.exit(3)   # worker dies before joining the process group
        ad = AutoDist(strategy_builder=AllReduce(), world_size=2)
        with ad.scope():
            model = torch.nn.Linear(2, 1)
            opt = torch.optim.SGD(model.parameters(), lr=0.1)
        sess = ad.create_distributed_session()  # blocks in rendezvous
        print("SHOULD NOT REACH")
    """))
    r = subprocess.run([sys.executable, str(script)], capture_output=True,
                       text=True, timeout=120)
    assert r.returncode != 0
    assert "SHOULD NOT REACH" not in r.stdout


def test_cost_model_fit_latency_bandwidth():
    """fit() recovers both terms from synthetic t = a + wire/bw samples."""
    from autodist_amd.simulator.cost_model import CostModel
    cm = CostModel(calibration=None)
    lat, bw = 30e-6, 400e9
    samples = []
    for nbytes in (1 << 16, 1 << 20, 1 << 24, 1 << 27):
        wire = 2.0 * 7 / 8 * nbytes
        samples.append((nbytes, 8, lat + wire / bw))
    cm.fit(samples)
    assert abs(cm.coll_latency - lat) / lat < 0.05
    got_bw = cm.allreduce_eff * cm.links * cm.link_gbps * 1e9
    assert abs(got_bw - bw) / bw < 0.05


def test_cost_model_calibration_roundtrip(tmp_path):
    from autodist_amd.simulator.cost_model import CostModel
    cm = CostModel(calibration=None)
    cm.allreduce_eff = 0.42
    cm.coll_latency = 33e-6
    path = str(tmp_path / "cal.json")
    cm.save_calibration(path, measured_on="unit test",
                        samples=[[1024, 8, 1e-4]])
    cm2 = CostModel(calibration=path)
    assert cm2.allreduce_eff == 0.42
    assert cm2.coll_latency == 33e-6
    assert cm2.calibrated_from == "unit test"


def test_cost_model_fit_latency_only():
    from autodist_amd.simulator.cost_model import CostModel
    cm = CostModel(calibration=None)
    cm.fit_latency([(1024, 1, 25e-6), (2048, 1, 27e-6), (4096, 1, 29e-6)])
    assert abs(cm.coll_latency - 27e-6) < 1e-9
