"""AUTODIST_FORCE_COLLECTIVES: world-1 process group executing every real
collective (sum over one rank == identity). CPU/gloo version of the 1-GPU
RCCL hardware validation (tests/test_rccl_world1_gpu.py) — training under
forced collectives must match plain torch bit-for-bit."""
import os
import socket

import pytest
import torch
import torch.distributed as dist

from autodist_amd.graph_item import GraphItem
from autodist_amd.parallel.engine import DistributedEngine
from autodist_amd.resource_spec import ResourceSpec


@pytest.fixture()
def world1_pg(monkeypatch):
    monkeypatch.setenv("AUTODIST_FORCE_COLLECTIVES", "1")
    if not dist.is_initialized():
        with socket.socket() as s:
            s.bind(("127.0.0.1", 0))
            port = s.getsockname()[1]
        dist.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{port}",
            rank=0, world_size=1)
    yield
    if dist.is_initialized():
        dist.destroy_process_group()


def _train(model, opt, data, engine=None):
    for x, y in data:
        opt.zero_grad()
        torch.nn.functional.mse_loss(model(x), y).backward()
        opt.step()
    if engine is not None:
        engine.drain()


CASES = ["AllReduce", "PS", "PartitionedPS", "PartitionedAR", "Parallax"]


@pytest.mark.parametrize("builder_name", CASES)
def test_forced_collectives_match_torch(world1_pg, builder_name):
    from autodist_amd import strategy as strat
    data = [(torch.randn(8, 6), torch.randn(8, 4)) for _ in range(4)]

    torch.manual_seed(3)
    model_t = torch.nn.Sequential(torch.nn.Linear(6, 16), torch.nn.Tanh(),
                                  torch.nn.Linear(16, 4))
    opt_t = torch.optim.SGD(model_t.parameters(), lr=0.05, momentum=0.9)
    _train(model_t, opt_t, data)

    torch.manual_seed(3)
    model_e = torch.nn.Sequential(torch.nn.Linear(6, 16), torch.nn.Tanh(),
                                  torch.nn.Linear(16, 4))
    g = GraphItem()
    g.extend_model(model_e)
    opt_e = torch.optim.SGD(model_e.parameters(), lr=0.05, momentum=0.9)
    g.extend_optimizer_info(opt_e)
    strategy = getattr(strat, builder_name)().build(g, ResourceSpec())
    engine = DistributedEngine(g, strategy, rank=0, world_size=1,
                               device=torch.device("cpu"))
    assert engine._force_collectives
    engine.setup()
    assert engine.collectives_active
    _train(model_e, opt_e, data, engine)
    engine.teardown()
    for pt, pe in zip(model_t.parameters(), model_e.parameters()):
        assert torch.allclose(pt, pe, atol=1e-6), \
            f"{builder_name}: forced collectives changed numerics " \
            f"({(pt - pe).abs().max()})"


def test_forced_sparse_path(world1_pg):
    from autodist_amd.strategy import Parallax
    torch.manual_seed(5)
    emb_t = torch.nn.Embedding(16, 4, sparse=True)
    opt_t = torch.optim.SGD(emb_t.parameters(), lr=0.5)
    ids = torch.tensor([1, 5, 5, 9])
    for _ in range(2):
        opt_t.zero_grad()
        emb_t(ids).sum().backward()
        opt_t.step()

    torch.manual_seed(5)
    emb_e = torch.nn.Embedding(16, 4, sparse=True)
    g = GraphItem()
    g.extend_model(emb_e)
    opt_e = torch.optim.SGD(emb_e.parameters(), lr=0.5)
    g.extend_optimizer_info(opt_e)
    strategy = Parallax().build(g, ResourceSpec())
    engine = DistributedEngine(g, strategy, rank=0, world_size=1,
                               device=torch.device("cpu")).setup()
    for _ in range(2):
        opt_e.zero_grad()
        emb_e(ids).sum().backward()
        opt_e.step()
    engine.drain()
    engine.teardown()
    assert torch.allclose(emb_t.weight, emb_e.weight, atol=1e-6)
