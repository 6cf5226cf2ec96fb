"""Remapper unit coverage (world=1 paths; multi-rank behavior in
test_engine_gloo.py::test_remapper_feed_fetch)."""
import numpy as np
import torch

from autodist_amd.remapper import Remapper


def _r(world=1, rank=0):
    return Remapper(rank, world, torch.device("cpu"))


def test_feed_passthrough_world1():
    r = _r()
    a = np.arange(6, dtype=np.float32).reshape(3, 2)
    t = r.remap_feed(a)
    assert isinstance(t, torch.Tensor) and t.shape == (3, 2)
    x = torch.randn(4)
    assert r.remap_feed(x).shape == (4,)
    assert r.remap_feed(3.5) == 3.5  # scalars duplicated


def test_feed_dict():
    r = _r()
    out = r.remap_feed_dict({"a": np.zeros((2, 2), np.float32), "b": 7})
    assert set(out) == {"a", "b"}
    assert out["b"] == 7


def test_fetch_structures():
    r = _r()
    t = torch.randn(3)
    s = torch.tensor(2.0)
    out = r.remap_fetches((t, s))
    assert isinstance(out, tuple) and len(out) == 2
    out = r.remap_fetches([t, s])
    assert isinstance(out, list)
    out = r.remap_fetches({"loss": s, "preds": t})
    assert set(out) == {"loss", "preds"}
    assert r.remap_fetches("not-a-tensor") == "not-a-tensor"


def test_fetch_detaches():
    r = _r()
    x = torch.randn(3, requires_grad=True)
    y = (x * 2).sum()
    out = r.remap_fetch(y)
    assert not out.requires_grad
