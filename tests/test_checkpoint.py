"""Checkpoint tests (reference tests/checkpoint/: shards must save/restore as
the ORIGINAL unpartitioned tensor; restore validated with VANILLA torch to
prove single-node compatibility, cases/c0.py:125-133)."""
import copy

import pytest
import torch

from autodist_amd.checkpoint.saver import Saver
from autodist_amd.graph_item import GraphItem
from autodist_amd.parallel.engine import DistributedEngine
from autodist_amd.resource_spec import ResourceSpec
from autodist_amd.strategy import AllReduce, PartitionedPS, PS
from tests.dist_utils import run_distributed


def _setup(builder, seed=0, lr=0.05, momentum=0.9, world=1, rank=0):
    torch.manual_seed(seed)
    model = torch.nn.Sequential(torch.nn.Linear(6, 12), torch.nn.Tanh(),
                                torch.nn.Linear(12, 3))
    g = GraphItem()
    g.extend_model(model)
    opt = torch.optim.SGD(model.parameters(), lr=lr, momentum=momentum)
    g.extend_optimizer_info(opt)
    strategy = builder.build(g, ResourceSpec())
    if world > 1:
        strategy.graph_config.replicas = [
            f"127.0.0.1:CPU:{r}" for r in range(world)]
    engine = DistributedEngine(g, strategy, rank=rank, world_size=world,
                               device=torch.device("cpu")).setup()
    return model, opt, g, engine


def _train(model, opt, steps, seed=10):
    for s in range(steps):
        torch.manual_seed(seed + s)
        x, y = torch.randn(8, 6), torch.randn(8, 3)
        opt.zero_grad()
        torch.nn.functional.mse_loss(model(x), y).backward()
        opt.step()


@pytest.mark.parametrize("builder_cls", [AllReduce, PS, PartitionedPS])
def test_save_restore_vanilla_torch_roundtrip(builder_cls, tmp_path):
    """Train 3 steps under the engine, save, load into VANILLA torch, train 2
    more steps both ways: must match a pure-torch run exactly."""
    model, opt, g, engine = _setup(builder_cls())
    saver = Saver(graph_item=g)
    _train(model, opt, 3)
    path = saver.save(str(tmp_path / "ckpt"))
    engine.teardown()

    # vanilla torch continuation from checkpoint
    torch.manual_seed(0)
    vmodel = torch.nn.Sequential(torch.nn.Linear(6, 12), torch.nn.Tanh(),
                                 torch.nn.Linear(12, 3))
    ckpt = torch.load(path, weights_only=False)
    vmodel.load_state_dict(ckpt["model"])
    vopt = torch.optim.SGD(vmodel.parameters(), lr=0.05, momentum=0.9)
    vopt.load_state_dict(ckpt["optimizer"])
    _train(vmodel, vopt, 2, seed=20)

    # pure torch all along
    torch.manual_seed(0)
    rmodel = torch.nn.Sequential(torch.nn.Linear(6, 12), torch.nn.Tanh(),
                                 torch.nn.Linear(12, 3))
    ropt = torch.optim.SGD(rmodel.parameters(), lr=0.05, momentum=0.9)
    _train(rmodel, ropt, 3)
    _train(rmodel, ropt, 2, seed=20)

    for pv, pr in zip(vmodel.parameters(), rmodel.parameters()):
        assert torch.allclose(pv, pr, atol=1e-6), (pv - pr).abs().max()


@pytest.mark.parametrize("builder_cls", [AllReduce, PS, PartitionedPS])
def test_restore_into_engine(builder_cls, tmp_path):
    """Engine -> checkpoint -> fresh engine: training continues identically."""
    model, opt, g, engine = _setup(builder_cls())
    saver = Saver(graph_item=g)
    _train(model, opt, 3)
    path = saver.save(str(tmp_path / "ckpt"))
    _train(model, opt, 2, seed=20)
    expected = [p.detach().clone() for p in model.parameters()]
    engine.teardown()

    model2, opt2, g2, engine2 = _setup(builder_cls(), seed=99)
    saver2 = Saver(graph_item=g2)
    saver2.restore(path)
    _train(model2, opt2, 2, seed=20)
    for p2, pe in zip(model2.parameters(), expected):
        assert torch.allclose(p2, pe, atol=1e-6), (p2 - pe).abs().max()
    engine2.teardown()


def _dist_ckpt_case(rank, world, tmpdir, builder_name):
    from autodist_amd import strategy as strat
    builder = getattr(strat, builder_name)()
    model, opt, g, engine = _setup(builder, world=world, rank=rank)
    saver = Saver(graph_item=g)
    for s in range(3):
        torch.manual_seed(50 + 10 * s + rank)
        x, y = torch.randn(8, 6), torch.randn(8, 3)
        opt.zero_grad()
        torch.nn.functional.mse_loss(model(x), y).backward()
        opt.step()
    path = saver.save(tmpdir + "/ckpt_dist")
    import torch.distributed as dist
    dist.barrier()
    # checkpoint must load into a fresh single-process model
    ckpt = torch.load(path, weights_only=False)
    vmodel = torch.nn.Sequential(torch.nn.Linear(6, 12), torch.nn.Tanh(),
                                 torch.nn.Linear(12, 3))
    vmodel.load_state_dict(ckpt["model"])
    vopt = torch.optim.SGD(vmodel.parameters(), lr=0.05, momentum=0.9)
    vopt.load_state_dict(ckpt["optimizer"])
    # model params in ckpt match this rank's replica (replicas are in sync)
    for pv, p in zip(vmodel.parameters(), model.parameters()):
        assert torch.allclose(pv, p.detach(), atol=1e-6)
    # momentum state exists for every param
    assert len(ckpt["optimizer"]["state"]) == len(list(model.parameters()))
    engine.teardown()


@pytest.mark.integration
@pytest.mark.parametrize("builder_name", ["PSLoadBalancing", "PartitionedPS",
                                          "AllReduce"])
def test_distributed_checkpoint(builder_name, tmp_path):
    run_distributed(_dist_ckpt_case, world_size=2,
                    args=(str(tmp_path), builder_name))


def test_saved_model_builder(tmp_path):
    from autodist_amd.checkpoint import SavedModelBuilder
    model, opt, g, engine = _setup(AllReduce())
    saver = Saver(graph_item=g)
    _train(model, opt, 1)
    b = SavedModelBuilder(str(tmp_path / "export"))
    b.add_meta_graph_and_variables(saver, tags=["serve"],
                                   example_inputs=torch.randn(2, 6))
    out = b.save()
    import os
    assert os.path.exists(os.path.join(out, "model.pt"))
    assert os.path.exists(os.path.join(out, "metadata.json"))
    assert os.path.exists(os.path.join(out, "model_traced.pt"))
    traced = torch.jit.load(os.path.join(out, "model_traced.pt"))
    x = torch.randn(3, 6)
    model.eval()
    with torch.no_grad():
        assert torch.allclose(traced(x), model(x), atol=1e-5)
    engine.teardown()
