"""Engine numerics at world_size=1: engine-applied updates must match plain
torch.optim training bit-for-bit (the engine replaces optimizer.step, so this
is the ResourceApply* parity check — reference op_info.py:24-68 table)."""
import copy

import pytest
import torch

from autodist_amd.graph_item import GraphItem
from autodist_amd.parallel.engine import DistributedEngine
from autodist_amd.resource_spec import ResourceSpec
from autodist_amd.strategy import AllReduce, PartitionedAR, PS, PartitionedPS


def make_model(seed=0):
    torch.manual_seed(seed)
    return torch.nn.Sequential(
        torch.nn.Linear(6, 16), torch.nn.Tanh(), torch.nn.Linear(16, 4))


def torch_train(model, make_opt, data, steps):
    opt = make_opt(model.parameters())
    for x, y in data:
        opt.zero_grad()
        loss = torch.nn.functional.mse_loss(model(x), y)
        loss.backward()
        opt.step()
    return model


def engine_train(model, make_opt, data, steps, builder):
    g = GraphItem()
    g.extend_model(model)
    opt = make_opt(model.parameters())
    g.extend_optimizer_info(opt)
    strategy = builder.build(g, ResourceSpec())
    engine = DistributedEngine(g, strategy, rank=0, world_size=1,
                               device=torch.device("cpu"))
    engine.setup()
    for x, y in data:
        opt.zero_grad()  # routed to engine
        loss = torch.nn.functional.mse_loss(model(x), y)
        loss.backward()
        opt.step()       # routed to engine
    engine.drain()
    engine.teardown()
    return model


OPTS = [
    ("sgd", lambda ps: torch.optim.SGD(ps, lr=0.05)),
    ("sgd_mom", lambda ps: torch.optim.SGD(ps, lr=0.05, momentum=0.9,
                                           weight_decay=1e-4)),
    ("sgd_nesterov", lambda ps: torch.optim.SGD(ps, lr=0.05, momentum=0.9,
                                                nesterov=True)),
    ("adam", lambda ps: torch.optim.Adam(ps, lr=1e-2)),
    ("adamw", lambda ps: torch.optim.AdamW(ps, lr=1e-2, weight_decay=0.05)),
    ("adagrad", lambda ps: torch.optim.Adagrad(ps, lr=1e-2)),
    ("rmsprop", lambda ps: torch.optim.RMSprop(ps, lr=1e-3, momentum=0.9)),
    ("adamax", lambda ps: torch.optim.Adamax(ps, lr=1e-2, weight_decay=1e-3)),
    ("nadam", lambda ps: torch.optim.NAdam(ps, lr=1e-2)),
    ("nadam_wd", lambda ps: torch.optim.NAdam(ps, lr=1e-2, weight_decay=0.02,
                                              decoupled_weight_decay=True)),
    ("radam", lambda ps: torch.optim.RAdam(ps, lr=1e-2)),
    ("adadelta", lambda ps: torch.optim.Adadelta(ps, lr=0.5)),
    ("asgd", lambda ps: torch.optim.ASGD(ps, lr=1e-2)),
    ("rprop", lambda ps: torch.optim.Rprop(ps, lr=1e-2)),
]


def _data(steps=5, seed=7):
    torch.manual_seed(seed)
    return [(torch.randn(8, 6), torch.randn(8, 4)) for _ in range(steps)]


@pytest.mark.parametrize("name,make_opt", OPTS)
@pytest.mark.parametrize("builder_cls", [AllReduce, PS, PartitionedPS,
                                         PartitionedAR])
def test_world1_matches_torch(name, make_opt, builder_cls):
    data = _data()
    m_ref = make_model()
    m_eng = copy.deepcopy(m_ref)
    torch_train(m_ref, make_opt, data, len(data))
    engine_train(m_eng, make_opt, data, len(data), builder_cls())
    for (n1, p1), (n2, p2) in zip(m_ref.named_parameters(),
                                  m_eng.named_parameters()):
        assert torch.allclose(p1, p2, atol=1e-6), \
            f"{name}/{builder_cls.__name__}: param {n1} diverged " \
            f"(max {((p1 - p2).abs().max())})"


def test_multi_param_group_hypers():
    """Different per-group hyperparams must bucket separately and match
    torch exactly."""
    data = _data()
    torch.manual_seed(0)
    m_ref = make_model()
    m_eng = copy.deepcopy(m_ref)

    def mk(model):
        params = list(model.parameters())
        return torch.optim.SGD([
            {"params": params[:2], "lr": 0.1, "momentum": 0.9},
            {"params": params[2:], "lr": 0.01, "weight_decay": 1e-3},
        ], lr=0.05)

    torch_train(m_ref, lambda ps: mk(m_ref), data, len(data))
    engine_train(m_eng, lambda ps: mk(m_eng), data, len(data), AllReduce())
    for p1, p2 in zip(m_ref.parameters(), m_eng.parameters()):
        assert torch.allclose(p1, p2, atol=1e-6)


def test_gradient_accumulation_no_sync():
    """2 micro-batches accumulated + 1 step == torch on the summed grads."""
    data = _data(steps=4)
    m_ref = make_model()
    m_eng = copy.deepcopy(m_ref)
    # torch reference: accumulate 2 micro-batches per step
    opt_r = torch.optim.SGD(m_ref.parameters(), lr=0.05, momentum=0.9)
    for i in range(0, 4, 2):
        opt_r.zero_grad()
        for x, y in data[i:i + 2]:
            torch.nn.functional.mse_loss(m_ref(x), y).backward()
        opt_r.step()
    # engine with no_sync
    from autodist_amd.graph_item import GraphItem
    from autodist_amd.parallel.engine import DistributedEngine
    from autodist_amd.resource_spec import ResourceSpec
    g = GraphItem()
    g.extend_model(m_eng)
    opt_e = torch.optim.SGD(m_eng.parameters(), lr=0.05, momentum=0.9)
    g.extend_optimizer_info(opt_e)
    engine = DistributedEngine(g, AllReduce().build(g, ResourceSpec()),
                               rank=0, world_size=1,
                               device=torch.device("cpu")).setup()
    for i in range(0, 4, 2):
        opt_e.zero_grad()
        with engine.no_sync():
            x, y = data[i]
            torch.nn.functional.mse_loss(m_eng(x), y).backward()
        x, y = data[i + 1]
        torch.nn.functional.mse_loss(m_eng(x), y).backward()
        opt_e.step()
    engine.teardown()
    for p1, p2 in zip(m_ref.parameters(), m_eng.parameters()):
        assert torch.allclose(p1, p2, atol=1e-6), (p1 - p2).abs().max()


class _CustomOpt(torch.optim.SGD):
    """A user-defined optimizer class unknown to the engine applier."""


def test_unsupported_optimizer_fallback():
    """Unknown optimizer classes: pure-AR strategies fall back to the user
    optimizer after gradient sync."""
    data = _data()
    m_ref = make_model()
    m_eng = copy.deepcopy(m_ref)
    mk = lambda ps: _CustomOpt(ps, lr=1e-2, momentum=0.9)  # noqa: E731
    torch_train(m_ref, mk, data, len(data))
    engine_train(m_eng, mk, data, len(data), AllReduce())
    for p1, p2 in zip(m_ref.parameters(), m_eng.parameters()):
        assert torch.allclose(p1, p2, atol=1e-6)


def test_unsupported_optimizer_with_ps_raises():
    g = GraphItem()
    m = make_model()
    g.extend_model(m)
    opt = _CustomOpt(m.parameters(), lr=1e-2)
    g.extend_optimizer_info(opt)
    strategy = PS().build(g, ResourceSpec())
    engine = DistributedEngine(g, strategy, rank=0, world_size=1,
                               device=torch.device("cpu"))
    with pytest.raises(NotImplementedError):
        engine.setup()
