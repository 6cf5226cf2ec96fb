"""Model-family coverage (reference examples/benchmark: bert, ncf, lm1b) —
tiny configs train through the engine on CPU; GPU runs use the full sizes."""
import pytest
import torch

from autodist_amd.graph_item import GraphItem
from autodist_amd.parallel.engine import DistributedEngine
from autodist_amd.resource_spec import ResourceSpec
from autodist_amd.strategy import (AllReduce, Parallax, PartitionedAR,
                                   PartitionedPS)
from tests.dist_utils import run_distributed


def _engine_for(model, opt_fn, builder, world=1, rank=0):
    g = GraphItem()
    g.extend_model(model)
    opt = opt_fn(model.parameters())
    g.extend_optimizer_info(opt)
    strategy = builder.build(g, ResourceSpec())
    if world > 1:
        strategy.graph_config.replicas = [
            f"127.0.0.1:CPU:{r}" for r in range(world)]
    engine = DistributedEngine(g, strategy, rank=rank, world_size=world,
                               device=torch.device("cpu")).setup()
    return opt, engine


def test_bert_tiny_trains():
    from autodist_amd.models.bert import bert_tiny
    torch.manual_seed(0)
    model = bert_tiny()
    opt, engine = _engine_for(model, lambda p: torch.optim.AdamW(p, lr=1e-3),
                              Parallax())
    losses = []
    for s in range(3):
        torch.manual_seed(s)
        ids = torch.randint(0, 1000, (2, 32))
        labels = ids.clone()
        labels[:, ::2] = -100
        opt.zero_grad()
        loss = model.loss(ids, labels)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert all(v == v for v in losses)
    assert losses[-1] < losses[0]
    engine.teardown()


def test_ncf_sparse_trains():
    from autodist_amd.models.ncf import NeuMF
    torch.manual_seed(0)
    model = NeuMF(100, 200, mf_dim=8, mlp_dims=(16, 8), sparse=True)
    opt, engine = _engine_for(model, lambda p: torch.optim.SGD(p, lr=0.05),
                              Parallax())
    g0 = None
    for s in range(4):
        torch.manual_seed(s)
        u = torch.randint(0, 100, (32,))
        i = torch.randint(0, 200, (32,))
        y = torch.randint(0, 2, (32,))
        opt.zero_grad()
        loss = model.loss(u, i, y)
        loss.backward()
        opt.step()
        if g0 is None:
            g0 = loss.item()
    assert loss.item() < g0
    engine.teardown()


def test_lm1b_partitioned_ar_trains():
    from autodist_amd.models.lm1b import lm1b_small
    torch.manual_seed(0)
    model = lm1b_small(vocab_size=500)
    opt, engine = _engine_for(model, lambda p: torch.optim.Adam(p, lr=1e-3),
                              PartitionedAR(min_partition_numel=1024))
    first = None
    for s in range(3):
        torch.manual_seed(s)
        toks = torch.randint(0, 500, (4, 16))
        tgts = torch.randint(0, 500, (4, 16))
        opt.zero_grad()
        loss = model.loss(toks, tgts)
        loss.backward()
        opt.step()
        if first is None:
            first = loss.item()
    assert loss.item() < first
    engine.teardown()


def test_ncf_partitioned_ps():
    from autodist_amd.models.ncf import NeuMF
    torch.manual_seed(1)
    model = NeuMF(64, 64, mf_dim=8, mlp_dims=(16, 8), sparse=False)
    opt, engine = _engine_for(model, lambda p: torch.optim.SGD(p, lr=0.05),
                              PartitionedPS())
    for s in range(2):
        torch.manual_seed(s)
        u = torch.randint(0, 64, (16,))
        i = torch.randint(0, 64, (16,))
        y = torch.randint(0, 2, (16,))
        opt.zero_grad()
        model.loss(u, i, y).backward()
        opt.step()
    engine.drain()
    for p in model.parameters():
        assert torch.isfinite(p).all()
    engine.teardown()


@pytest.mark.parametrize("maker,size", [
    ("vgg16", 64), ("densenet121", 64), ("inception_v3", 299)])
def test_imagenet_cnn_families_train(maker, size):
    """VGG16 / DenseNet121 / InceptionV3 (reference benchmark models,
    examples/benchmark/imagenet.py) run a training step through the engine."""
    from autodist_amd.models.densenet import densenet121
    from autodist_amd.models.inception import inception_v3
    from autodist_amd.models.vgg import vgg16
    makers = {"vgg16": lambda: vgg16(num_classes=10, batch_norm=True),
              "densenet121": lambda: densenet121(num_classes=10),
              "inception_v3": lambda: inception_v3(num_classes=10)}
    torch.manual_seed(0)
    model = makers[maker]()
    opt, engine = _engine_for(model, lambda p: torch.optim.SGD(p, lr=0.01),
                              AllReduce())
    x = torch.randn(2, 3, size, size)
    y = torch.randint(0, 10, (2,))
    opt.zero_grad()
    loss = torch.nn.functional.cross_entropy(model(x), y)
    loss.backward()
    opt.step()
    assert torch.isfinite(loss)
    for p in model.parameters():
        assert torch.isfinite(p).all()
    engine.teardown()


# ----------------------------------------------------- sharded embedding

def _sharded_embedding_case(rank, world):
    import torch.distributed as dist
    from autodist_amd.parallel.sharded_embedding import ShardedEmbedding

    torch.manual_seed(0)  # same on both ranks
    ref = torch.nn.Embedding(50, 8)
    se = ShardedEmbedding(50, 8, rank=rank, world_size=world)
    # install reference rows into the shard
    with torch.no_grad():
        se.shard.copy_(ref.weight[se.row_start:se.row_end])
    torch.manual_seed(100 + rank)
    ids = torch.randint(0, 50, (7, 3))
    out = se(ids)
    ref_out = ref(ids)
    assert torch.allclose(out, ref_out, atol=1e-6), \
        (out - ref_out).abs().max()
    torch.manual_seed(300 + rank)
    g = torch.randn_like(out)
    out.backward(g)
    # reference: the shard grad accumulates EVERY rank's contributions to the
    # local rows (grads were routed to owners in backward)
    for r in range(world):
        torch.manual_seed(100 + r)
        ids_r = torch.randint(0, 50, (7, 3))
        torch.manual_seed(300 + r)
        g_r = torch.randn(7, 3, 8)
        ref(ids_r).backward(g_r)
    ref_grad = ref.weight.grad[se.row_start:se.row_end]
    assert torch.allclose(se.shard.grad, ref_grad, atol=1e-6), \
        (se.shard.grad - ref_grad).abs().max()
    # full_weight assembles the original table
    full = se.full_weight()
    assert torch.allclose(full, ref.weight.detach(), atol=1e-6)
    dist.barrier()


@pytest.mark.integration
def test_sharded_embedding_gloo():
    run_distributed(_sharded_embedding_case, world_size=2)


def _sharded_ncf_case(rank, world, tmpdir):
    from autodist_amd.checkpoint.saver import Saver
    from autodist_amd.models.ncf import NeuMF
    torch.manual_seed(3)
    model = NeuMF(40, 60, mf_dim=8, mlp_dims=(16, 8), sharded=True)
    g = None
    opt, engine = _engine_for(model, lambda p: torch.optim.SGD(
        p, lr=0.05, momentum=0.9), AllReduce(), world=world, rank=rank)
    saver = Saver(graph_item=engine.graph_item)
    for s in range(3):
        torch.manual_seed(200 + 10 * s + rank)
        u = torch.randint(0, 40, (16,))
        i = torch.randint(0, 60, (16,))
        y = torch.randint(0, 2, (16,))
        opt.zero_grad()
        model.loss(u, i, y).backward()
        opt.step()
    import torch.distributed as dist
    # dense tower params stay replica-consistent
    for name, p in model.named_parameters():
        if getattr(p, "_autodist_shard_local", False):
            continue
        lst = [torch.zeros_like(p) for _ in range(world)]
        dist.all_gather(lst, p.detach())
        assert torch.allclose(lst[0], lst[1], atol=1e-6), name
    # checkpoint: consolidated, nn.Embedding-compatible
    path = saver.save(tmpdir + "/sharded_ckpt")
    dist.barrier()
    ckpt = torch.load(path, weights_only=False)
    assert tuple(ckpt["model"]["mf_user.weight"].shape) == (40, 8)
    vanilla = NeuMF(40, 60, mf_dim=8, mlp_dims=(16, 8), sparse=True)
    vanilla.load_state_dict(ckpt["model"])
    assert torch.allclose(vanilla.mf_user.weight.detach(),
                          model.mf_user.full_weight().cpu(), atol=1e-6)
    # restore into a fresh sharded model: shard + momentum state round-trip
    torch.manual_seed(99)
    model2 = NeuMF(40, 60, mf_dim=8, mlp_dims=(16, 8), sharded=True)
    opt2, engine2 = _engine_for(model2, lambda p: torch.optim.SGD(
        p, lr=0.05, momentum=0.9), AllReduce(), world=world, rank=rank)
    saver2 = Saver(graph_item=engine2.graph_item)
    saver2.restore(path)
    assert torch.allclose(model2.mf_user.shard.detach(),
                          model.mf_user.shard.detach(), atol=1e-6)
    engine.teardown()
    engine2.teardown()


@pytest.mark.integration
def test_sharded_ncf_gloo(tmp_path):
    run_distributed(_sharded_ncf_case, world_size=2, args=(str(tmp_path),))
