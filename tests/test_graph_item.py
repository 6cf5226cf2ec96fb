"""GraphItem capture tests (reference tests/test_graph_item.py:56-120 —
update-op discovery per optimizer and proto round-trip)."""
import pytest
import torch

from autodist_amd.graph_item import GraphItem, get_default_graph_item


class TinyNet(torch.nn.Module):
    def __init__(self):
        super().__init__()
        self.fc1 = torch.nn.Linear(4, 8)
        self.fc2 = torch.nn.Linear(8, 2)
        self.emb = torch.nn.Embedding(10, 4, sparse=True)


OPTIMIZERS = [
    lambda ps: torch.optim.SGD(ps, lr=0.1),
    lambda ps: torch.optim.SGD(ps, lr=0.1, momentum=0.9),
    lambda ps: torch.optim.SGD(ps, lr=0.1, momentum=0.9, nesterov=True),
    lambda ps: torch.optim.Adam(ps, lr=1e-3),
    lambda ps: torch.optim.AdamW(ps, lr=1e-3),
    lambda ps: torch.optim.Adamax(ps, lr=1e-3),
    lambda ps: torch.optim.Adagrad(ps, lr=1e-2),
    lambda ps: torch.optim.Adadelta(ps),
    lambda ps: torch.optim.RMSprop(ps, lr=1e-3),
    lambda ps: torch.optim.RMSprop(ps, lr=1e-3, momentum=0.9),
    lambda ps: torch.optim.NAdam(ps, lr=1e-3),
    lambda ps: torch.optim.RAdam(ps, lr=1e-3),
    lambda ps: torch.optim.ASGD(ps, lr=1e-2),
    lambda ps: torch.optim.Rprop(ps, lr=1e-2),
]


@pytest.mark.parametrize("make_opt", OPTIMIZERS)
def test_optimizer_capture(make_opt):
    """Every optimizer's params map 1:1 onto captured variables
    (reference test_update_ops_for_optimizers, test_graph_item.py:56-84)."""
    g = GraphItem()
    net = TinyNet()
    g.extend_model(net)
    opt = make_opt(net.parameters())
    g.extend_optimizer_info(opt)
    assert len(g.optimizers) == 1
    item = g.optimizers[0]
    trainables = list(g.trainable_var_op_to_var)
    # one captured var per trainable param, all named by the optimizer
    assert sorted(item.param_names) == sorted(trainables)
    assert g.optimizer is opt


def test_sparse_detection():
    g = GraphItem()
    g.extend_model(TinyNet())
    assert g.variables["emb.weight"].is_sparse
    assert not g.variables["fc1.weight"].is_sparse


def test_scope_semantics():
    """as_default scope sets/restores the default item
    (reference test_graph_item.py:86-97)."""
    g = GraphItem()
    assert get_default_graph_item() is None
    with g.as_default():
        assert get_default_graph_item() is g
        g2 = GraphItem()
        with g2.as_default():
            assert get_default_graph_item() is g2
        assert get_default_graph_item() is g
    assert get_default_graph_item() is None


def test_serialize_roundtrip():
    """Metadata round-trip (reference test_graph_item.py:100-120)."""
    g = GraphItem()
    net = TinyNet()
    g.extend_model(net)
    g.extend_optimizer_info(torch.optim.Adam(net.parameters(), lr=2e-3))
    s = g.serialize_to_string()
    g2 = GraphItem.parse_from_string(s)
    assert list(g2.variables) == list(g.variables)
    for name in g.variables:
        assert g2.variables[name].shape == g.variables[name].shape
        assert g2.variables[name].is_sparse == g.variables[name].is_sparse
    assert g2.optimizers[0].cls_name == "Adam"
    assert g2.optimizers[0].defaults["lr"] == 2e-3


def test_grad_target_pairs():
    g = GraphItem()
    net = TinyNet()
    g.extend_model(net)
    pairs = g.grad_target_pairs()
    assert len(pairs) == len(list(net.parameters()))
    assert all(gname == f"grad/{v.name}" for gname, v in pairs)


def test_cost_model_orders_strategies(tmp_gpu_resource_spec):
    """Sanity: for one huge dense var, AllReduce should beat single PS."""
    from autodist_amd.resource_spec import ResourceSpec
    from autodist_amd.simulator.cost_model import CostModel
    from autodist_amd.strategy import AllReduce, PS

    g = GraphItem()

    class Big(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.w = torch.nn.Parameter(torch.zeros(4096, 4096))

    g.extend_model(Big())
    rs = ResourceSpec(tmp_gpu_resource_spec)
    cm = CostModel(rs)
    t_ar = cm.estimate(AllReduce().build(g, rs), g)
    t_ps = cm.estimate(PS().build(g, rs), g)
    assert 0 < t_ar < t_ps
