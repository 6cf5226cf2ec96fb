"""Runtime-layer tests: cluster spec, coordinator env protocol, AutoDist
end-to-end API (reference tests/test_autodist.py + cluster/coordinator
behavior)."""
import os
import subprocess
import sys

import numpy as np
import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_cluster_rank_table(tmp_path):
    from autodist_amd.resource_spec import ResourceSpec
    from autodist_amd.runtime.cluster import Cluster
    p = tmp_path / "r.yml"
    p.write_text(
        "nodes:\n"
        "  - address: 10.0.0.1\n    gpus: [0, 1]\n    chief: true\n"
        "  - address: 10.0.0.2\n    gpus: [0, 1, 2]\n")
    c = Cluster(ResourceSpec(str(p)))
    assert c.world_size == 5
    assert c.master_addr == "10.0.0.1"
    assert c.rank_table[0] == ("10.0.0.1", 0, 0)
    assert c.rank_table[-1] == ("10.0.0.2", 2, 4)
    assert c.cluster_spec["worker"][0] == "10.0.0.1:0"


def test_cluster_remote_exec_dry_run(tmp_path, monkeypatch):
    from autodist_amd.resource_spec import ResourceSpec
    from autodist_amd.runtime.cluster import SSHCluster
    monkeypatch.setenv("AUTODIST_DEBUG_REMOTE", "True")
    p = tmp_path / "r.yml"
    p.write_text(
        "nodes:\n"
        "  - address: 10.0.0.1\n    gpus: [0]\n    chief: true\n"
        "  - address: 10.0.0.2\n    gpus: [0]\n    ssh_config: c\n"
        "ssh:\n  c:\n    username: u\n    key_file: /k\n    port: 2222\n")
    c = SSHCluster(ResourceSpec(str(p)))
    # dry-run: no process spawned, no exception
    assert c.remote_exec("10.0.0.2", ["python", "x.py"],
                         {"RANK": "1"}) is None
    c.remote_copy("10.0.0.2", __file__, "/tmp/autodist_amd")


def test_autodist_end_to_end_world1(monkeypatch):
    """Full API flow in-process: scope capture -> strategy -> session.run."""
    monkeypatch.setenv("AUTODIST_IS_TESTING", "True")
    from autodist_amd import AutoDist
    from autodist_amd.autodist import _reset_default_autodist_for_tests
    from autodist_amd.strategy import AllReduce
    _reset_default_autodist_for_tests()
    ad = AutoDist(strategy_builder=AllReduce(), world_size=1)
    with ad.scope():
        torch.manual_seed(0)
        model = torch.nn.Linear(4, 2)
        opt = torch.optim.SGD(model.parameters(), lr=0.1)
    # scope captured model + optimizer without explicit registration
    ad.graph_item.prepare()
    assert "weight" in list(ad.graph_item.variables)[0] or \
        len(ad.graph_item.variables) == 2
    assert ad.graph_item.optimizer is opt

    def train_step(x, y):
        opt.zero_grad()
        loss = torch.nn.functional.mse_loss(model(x), y)
        loss.backward()
        opt.step()
        return loss

    sess = ad.create_distributed_session()
    xs = np.random.RandomState(0).randn(16, 4).astype(np.float32)
    ys = np.random.RandomState(1).randn(16, 2).astype(np.float32)
    l0 = float(sess.run(train_step, feed_dict={"x": xs, "y": ys}))
    for _ in range(20):
        l1 = float(sess.run(train_step, feed_dict={"x": xs, "y": ys}))
    assert l1 < l0
    sess.close()
    _reset_default_autodist_for_tests()


def test_graph_mutation_detected(monkeypatch):
    """Adding params after build is caught under AUTODIST_IS_TESTING
    (reference autodist.py:152-165)."""
    monkeypatch.setenv("AUTODIST_IS_TESTING", "True")
    from autodist_amd import AutoDist
    from autodist_amd.autodist import _reset_default_autodist_for_tests
    from autodist_amd.strategy import AllReduce
    _reset_default_autodist_for_tests()
    ad = AutoDist(strategy_builder=AllReduce(), world_size=1)
    with ad.scope():
        model = torch.nn.Sequential(torch.nn.Linear(2, 2))
        opt = torch.optim.SGD(model.parameters(), lr=0.1)
    sess = ad.create_distributed_session()

    def step(x):
        opt.zero_grad()
        model(x).square().mean().backward()
        opt.step()
        return torch.tensor(0.0)

    sess.run(step, feed_dict={"x": np.ones((2, 2), np.float32)})
    model.append(torch.nn.Linear(2, 2))  # mutate after build
    with pytest.raises(RuntimeError, match="mutated"):
        sess.run(step, feed_dict={"x": np.ones((2, 2), np.float32)})
    _reset_default_autodist_for_tests()


def test_one_autodist_per_process(monkeypatch):
    """Reference invariant (autodist.py:46-51 / tests/test_autodist.py)."""
    monkeypatch.delenv("AUTODIST_IS_TESTING", raising=False)
    from autodist_amd import AutoDist
    from autodist_amd.autodist import _reset_default_autodist_for_tests
    _reset_default_autodist_for_tests()
    AutoDist(world_size=1)
    with pytest.raises(RuntimeError):
        AutoDist(world_size=1)
    _reset_default_autodist_for_tests()


def test_autodist_function_single_cache(monkeypatch):
    monkeypatch.setenv("AUTODIST_IS_TESTING", "True")
    from autodist_amd import AutoDist
    from autodist_amd.autodist import _reset_default_autodist_for_tests
    _reset_default_autodist_for_tests()
    ad = AutoDist(world_size=1)
    with ad.scope():
        model = torch.nn.Linear(2, 1)
        opt = torch.optim.SGD(model.parameters(), lr=0.1)

    @ad.function
    def step(x):
        opt.zero_grad()
        loss = model(x).square().mean()
        loss.backward()
        opt.step()
        return loss

    out = step(np.ones((4, 2), dtype=np.float32))
    assert float(out) >= 0
    with pytest.raises(RuntimeError):
        ad.function(lambda x: x)  # only one function per scope
    _reset_default_autodist_for_tests()


@pytest.mark.integration
def test_coordinator_spawns_workers(tmp_path):
    """Chief re-executes the script for rank 1 (reference
    coordinator.py:46-90); verified by running linear_regression with
    world-size 2 on CPU via subprocess."""
    script = tmp_path / "tiny_dist.py"
    script.write_text(f"""
import os, sys
sys.path.insert(0, {REPO!r})
import numpy as np, torch
from autodist_amd import AutoDist
from autodist_amd.strategy import AllReduce
ad = AutoDist(strategy_builder=AllReduce(), world_size=2)
with ad.scope():
    torch.manual_seed(0)
    model = torch.nn.Linear(2, 1)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
def step(x, y):
    opt.zero_grad()
    loss = torch.nn.functional.mse_loss(model(x), y)
    loss.backward(); opt.step(); return loss
sess = ad.create_distributed_session()
rng = np.random.RandomState(0)
xs = rng.randn(64, 2).astype(np.float32)
ys = (xs @ np.array([[2.0], [1.0]], dtype=np.float32))
for _ in range(40):
    loss = sess.run(step, feed_dict={{"x": xs, "y": ys}})
print("RANK", os.environ.get("RANK", "0"), "FINAL", float(loss))
assert float(loss) < 0.05
sess.close()
""")
    r = subprocess.run([sys.executable, str(script)], capture_output=True,
                       text=True, timeout=240)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "FINAL" in r.stdout
