"""bench.py JSON contract (the driver's interface): one JSON line with the
BASELINE-named metric/config fields, whole-job aggregate value."""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

pytestmark = pytest.mark.integration

REQUIRED = ["metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"]


def _run_bench(args):
    r = subprocess.run(
        [sys.executable, "bench.py", "--steps", "1", "--warmup", "0"] + args,
        cwd=REPO, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    return json.loads(line)


def test_default_contract():
    out = _run_bench(["--batch-size", "2", "--image-size", "64"])
    for k in REQUIRED:
        assert k in out, f"missing field {k}"
    assert out["metric"].startswith("images/sec ResNet-50")
    assert out["n_gpus"] == 1 and out["steps"] == 1 and out["warmup"] == 0
    assert out["scaling"] == "weak" and out["higher_is_better"] is True
    assert out["data"] == "synthetic"
    assert out["config"]["model"] == "resnet50"
    assert out["config"]["parallelism"] == "dp1"
    assert out["value"] > 0 and out["ms_per_step"] > 0


def test_bert_contract():
    out = _run_bench(["--model", "bert", "--batch-size", "2",
                      "--seq-len", "32"])
    assert out["metric"].startswith("sequences/sec BERT-base")
    assert out["config"]["seq_len"] == 32
    assert out["config"]["strategy"] == "Parallax"
