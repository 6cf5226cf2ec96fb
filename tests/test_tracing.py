"""Tracing / observability tests (reference §5.1: chrome-trace timelines per
step when trace options set, runner.py:66-75; graph-phase dumps)."""
import glob
import json
import os

import torch

from autodist_amd.const import DEFAULT_GRAPH_DUMP_DIR, DEFAULT_TRACE_DIR
from autodist_amd.graph_item import GraphItem
from autodist_amd.parallel.graph_transformer import GraphTransformer
from autodist_amd.remapper import Remapper
from autodist_amd.resource_spec import ResourceSpec
from autodist_amd.runner import WrappedSession
from autodist_amd.strategy import AllReduce


def _session():
    torch.manual_seed(0)
    model = torch.nn.Linear(4, 2)
    g = GraphItem()
    g.extend_model(model)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    g.extend_optimizer_info(opt)
    strategy = AllReduce().build(g, ResourceSpec())
    engine = GraphTransformer(strategy, g, rank=0, world_size=1,
                              device=torch.device("cpu"),
                              dump_graphs=True).transform()
    remapper = Remapper(0, 1, torch.device("cpu"))
    return model, opt, WrappedSession(engine, remapper, g)


def test_chrome_trace_written():
    model, opt, sess = _session()

    def step(x):
        opt.zero_grad()
        loss = model(x).square().mean()
        loss.backward()
        opt.step()
        return loss

    before = set(glob.glob(os.path.join(DEFAULT_TRACE_DIR, "*.json")))
    sess.run(step, feed_dict={"x": torch.randn(8, 4)},
             options={"trace": True})
    after = set(glob.glob(os.path.join(DEFAULT_TRACE_DIR, "*.json")))
    new = after - before
    assert len(new) == 1
    with open(new.pop(), encoding="utf-8") as f:
        trace = json.load(f)
    assert "traceEvents" in trace and len(trace["traceEvents"]) > 0
    sess.engine.teardown()


def test_graph_phase_dumps():
    before = set(glob.glob(os.path.join(DEFAULT_GRAPH_DUMP_DIR, "*.txt")))
    model, opt, sess = _session()
    after = set(glob.glob(os.path.join(DEFAULT_GRAPH_DUMP_DIR, "*.txt")))
    new = sorted(after - before)
    names = [os.path.basename(p) for p in new]
    assert any("0-original" in n for n in names)
    assert any("3-transformed" in n for n in names)
    transformed = [p for p in new if "3-transformed" in p][0]
    content = open(transformed, encoding="utf-8").read()
    assert "bucket" in content
    sess.engine.teardown()


def test_describe_plan_mentions_all_kinds():
    from autodist_amd.parallel.engine import DistributedEngine
    from autodist_amd.strategy import Parallax
    from autodist_amd.utils.visualization_util import describe_plan
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
    desc = describe_plan(engine)
    assert "bucket" in desc and "ps" in desc and "sparse" in desc
    engine.teardown()
