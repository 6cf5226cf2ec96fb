"""Strategy x model matrix (reference tests/integration/test_all.py:20-46:
cartesian product of model cases x resource specs x strategy instances, each
isolated in its own process)."""
import pytest
import torch

from tests.dist_utils import run_distributed

pytestmark = pytest.mark.integration

STRATEGIES = [
    ("PS", {}),
    ("PS", {"sync": False}),
    ("PSLoadBalancing", {}),
    ("PartitionedPS", {}),
    ("UnevenPartitionedPS", {}),
    ("AllReduce", {"chunk_size": 2}),
    ("PartitionedAR", {"min_partition_numel": 1}),
    ("RandomAxisPartitionAR", {"min_partition_numel": 1}),
    ("Parallax", {}),
    ("AutoStrategy", {}),
]

MODELS = ["mlp", "emb_mix"]


def _make_model(kind):
    torch.manual_seed(11)
    if kind == "mlp":
        return torch.nn.Sequential(
            torch.nn.Linear(6, 24), torch.nn.ReLU(),
            torch.nn.Linear(24, 24), torch.nn.Tanh(),
            torch.nn.Linear(24, 3))
    if kind == "transformer":
        from autodist_amd.models.bert import bert_tiny
        return bert_tiny()
    if kind == "lm1b_sharded":
        from autodist_amd.models.lm1b import LM1BModel
        return LM1BModel(vocab_size=48, emb_dim=12, hidden=24, proj=12,
                         dropout=0.0, sharded_softmax=True)
    emb = torch.nn.Embedding(30, 6, sparse=True)
    lin = torch.nn.Linear(6, 3)
    return torch.nn.ModuleDict({"emb": emb, "lin": lin})


def _loss(kind, model, seed):
    torch.manual_seed(seed)
    if kind == "mlp":
        x, y = torch.randn(8, 6), torch.randn(8, 3)
        return torch.nn.functional.mse_loss(model(x), y)
    if kind == "transformer":
        ids = torch.randint(0, 1000, (2, 32))
        labels = ids.clone()
        labels[:, ::2] = -100
        return model.loss(ids, labels)
    if kind == "lm1b_sharded":
        tokens = torch.randint(0, 48, (3, 5))
        targets = torch.randint(0, 48, (3, 5))
        return model.loss(tokens, targets)
    ids = torch.randint(0, 30, (8, 4))
    y = torch.randn(8, 3)
    out = model["lin"](model["emb"](ids).mean(1))
    return torch.nn.functional.mse_loss(out, y)


def _case(rank, world, kind, strat_name, strat_kwargs):
    import torch.distributed as dist
    from autodist_amd import strategy as strat
    from autodist_amd.graph_item import GraphItem
    from autodist_amd.parallel.engine import DistributedEngine
    from autodist_amd.resource_spec import ResourceSpec

    model = _make_model(kind)
    g = GraphItem()
    g.extend_model(model)
    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    g.extend_optimizer_info(opt)
    builder = getattr(strat, strat_name)(**strat_kwargs)
    strategy = builder.build(g, ResourceSpec())
    strategy.graph_config.replicas = [f"127.0.0.1:CPU:{r}" for r in range(world)]
    engine = DistributedEngine(g, strategy, rank=rank, world_size=world,
                               device=torch.device("cpu")).setup()
    for s in range(3):
        opt.zero_grad()
        _loss(kind, model, 77 + 10 * s + rank).backward()
        opt.step()
    engine.drain()
    for name, p in model.named_parameters():
        assert torch.isfinite(p).all(), name
        if getattr(p, "_autodist_shard_local", False):
            continue  # exclusively-owned rows: shapes differ per rank
        lst = [torch.zeros_like(p) for _ in range(world)]
        dist.all_gather(lst, p.detach())
        for other in lst[1:]:
            assert torch.allclose(lst[0], other, atol=1e-5), \
                f"{name}: replicas diverged by {(lst[0]-other).abs().max()}"
    engine.teardown()


@pytest.mark.parametrize("kind", MODELS)
@pytest.mark.parametrize("strat_name,strat_kwargs", STRATEGIES)
def test_matrix(kind, strat_name, strat_kwargs):
    run_distributed(_case, world_size=2, args=(kind, strat_name, strat_kwargs))


@pytest.mark.parametrize("kind,strat_name", [
    ("transformer", "PS"), ("transformer", "AllReduce"),
    ("transformer", "Parallax"), ("transformer", "AutoStrategy"),
    ("lm1b_sharded", "AllReduce"), ("lm1b_sharded", "Parallax"),
])
def test_matrix_model_families(kind, strat_name):
    """Transformer (fused-LN/linear CPU fallbacks) and the sharded-softmax
    LM through the strategy matrix."""
    run_distributed(_case, world_size=2, args=(kind, strat_name, {}))


@pytest.mark.parametrize("strat_name", ["PartitionedPS", "PSLoadBalancing",
                                        "AllReduce", "Parallax"])
def test_matrix_world4(strat_name):
    """Spot-check at world_size=4 (multiple PS owners per var; deeper rank
    fan-out than the standard 2-rank cases)."""
    run_distributed(_case, world_size=4, args=("mlp", strat_name, {}))
