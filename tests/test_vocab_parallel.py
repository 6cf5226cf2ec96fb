"""Vocab-parallel projection + sharded cross-entropy (the LM1B 793k-vocab
seam — reference partitioner.py:577-602). Losses and gradients must match
the unsharded computation exactly."""
import pytest
import torch

from autodist_amd.parallel.vocab_parallel import VocabParallelProjection
from tests.dist_utils import run_distributed


def _ref_loss_grads(w, b, hidden, targets):
    wf = w.clone().requires_grad_(True)
    bf = b.clone().requires_grad_(True)
    hf = hidden.clone().requires_grad_(True)
    logits = hf.reshape(-1, hf.shape[-1]) @ wf.t() + bf
    loss = torch.nn.functional.cross_entropy(logits, targets.reshape(-1))
    loss.backward()
    return loss.detach(), hf.grad, wf.grad, bf.grad


def test_world1_matches_cross_entropy():
    torch.manual_seed(0)
    V, D, N = 50, 8, 12
    proj = VocabParallelProjection(V, D, rank=0, world_size=1)
    hidden = torch.randn(N, D, requires_grad=True)
    targets = torch.randint(0, V, (N,))
    loss = proj.loss(hidden, targets)
    loss.backward()
    ref, dh, dw, db = _ref_loss_grads(proj.weight.detach(),
                                      proj.bias.detach(), hidden.detach(),
                                      targets)
    assert torch.allclose(loss, ref, atol=1e-6)
    assert torch.allclose(hidden.grad, dh, atol=1e-6)
    assert torch.allclose(proj.weight.grad, dw, atol=1e-6)
    assert torch.allclose(proj.bias.grad, db, atol=1e-6)


def _world2_case(rank, world):
    torch.manual_seed(0)
    V, D, N = 50, 8, 12
    full_w = torch.randn(V, D)
    full_b = torch.randn(V)
    hidden = torch.randn(N, D)
    targets = torch.randint(0, V, (N,))
    proj = VocabParallelProjection(V, D, rank=rank, world_size=world)
    with torch.no_grad():
        proj.weight.copy_(full_w[proj.row_start:proj.row_end])
        proj.bias.copy_(full_b[proj.row_start:proj.row_end])
    h = hidden.clone().requires_grad_(True)
    loss = proj.loss(h, targets)
    loss.backward()
    ref, dh, dw, db = _ref_loss_grads(full_w, full_b, hidden, targets)
    assert torch.allclose(loss, ref, atol=1e-5), (loss.item(), ref.item())
    assert torch.allclose(h.grad, dh, atol=1e-5), \
        f"dh err {(h.grad - dh).abs().max()}"
    assert torch.allclose(proj.weight.grad,
                          dw[proj.row_start:proj.row_end], atol=1e-5)
    assert torch.allclose(proj.bias.grad,
                          db[proj.row_start:proj.row_end], atol=1e-5)
    # full_logits reassembles the unsharded projection
    logits = proj.full_logits(hidden)
    ref_logits = hidden @ full_w.t() + full_b
    assert torch.allclose(logits, ref_logits, atol=1e-5)


@pytest.mark.integration
def test_world2_sharded_ce_matches_full():
    run_distributed(_world2_case, world_size=2)


def _lm1b_engine_case(rank, world):
    """Full LM1B sharded-softmax model trains through the engine (tied
    ShardedEmbedding shard updates locally; LSTM syncs via AR buckets)."""
    from autodist_amd.graph_item import GraphItem
    from autodist_amd.models.lm1b import LM1BModel
    from autodist_amd.parallel.engine import DistributedEngine
    from autodist_amd.resource_spec import ResourceSpec
    from autodist_amd.strategy import AllReduce

    torch.manual_seed(7)
    model = LM1BModel(vocab_size=64, emb_dim=16, hidden=32, proj=16,
                      dropout=0.0, sharded_softmax=True)
    g = GraphItem()
    g.extend_model(model)
    opt = torch.optim.Adagrad(model.parameters(), lr=0.05)
    g.extend_optimizer_info(opt)
    strategy = AllReduce().build(g, ResourceSpec())
    strategy.graph_config.replicas = [
        f"127.0.0.1:CPU:{r}" for r in range(world)]
    engine = DistributedEngine(g, strategy, rank=rank, world_size=world,
                               device=torch.device("cpu")).setup()
    shard0 = model.emb.shard.detach().clone()
    torch.manual_seed(100)  # same FIXED batch on all ranks/steps so the
    tokens = torch.randint(0, 64, (4, 6))  # loss must strictly improve
    targets = torch.randint(0, 64, (4, 6))
    losses = []
    for _ in range(5):
        opt.zero_grad()
        loss = model.loss(tokens, targets)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    engine.drain()
    assert losses[-1] < losses[0], losses
    assert not torch.allclose(model.emb.shard.detach(), shard0)
    # LSTM weights stay replica-consistent
    import torch.distributed as dist
    w = next(model.lstm.parameters()).detach().clone()
    wsum = w.clone()
    dist.all_reduce(wsum)
    assert torch.allclose(wsum / world, w, atol=1e-6)
    engine.teardown()


@pytest.mark.integration
def test_lm1b_sharded_softmax_engine_world2():
    run_distributed(_lm1b_engine_case, world_size=2)


def _ckpt_case(rank, world):
    """state_dict of the sharded LM1B reassembles full tensors; loading a
    full checkpoint re-shards (SaveSliceInfo semantics)."""
    import io
    torch.manual_seed(11)
    from autodist_amd.models.lm1b import LM1BModel
    m = LM1BModel(vocab_size=64, emb_dim=16, hidden=32, proj=16,
                  dropout=0.0, sharded_softmax=True)
    sd = m.state_dict()
    assert tuple(sd["emb.weight"].shape) == (64, 16)
    assert tuple(sd["out.bias"].shape) == (64,)
    buf = io.BytesIO()
    torch.save(sd, buf)
    buf.seek(0)
    # a re-sharded fresh model loads the full tensors back into its shard
    torch.manual_seed(99)
    m2 = LM1BModel(vocab_size=64, emb_dim=16, hidden=32, proj=16,
                   dropout=0.0, sharded_softmax=True)
    m2.load_state_dict(torch.load(buf, weights_only=False))
    assert torch.allclose(m2.emb.shard, m.emb.shard)
    assert torch.allclose(m2.out.bias, m.out.bias)


@pytest.mark.integration
def test_vocab_parallel_checkpoint_world2():
    run_distributed(_ckpt_case, world_size=2)
