"""Example scripts run end-to-end (reference integration cases c0-c8 cover
the example surface; these exercise the shipped examples as subprocesses)."""
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

pytestmark = pytest.mark.integration


def _run(args, timeout=300):
    env = dict(os.environ)
    env["AUTODIST_IS_TESTING"] = "True"
    r = subprocess.run([sys.executable] + args, cwd=REPO, env=env,
                       capture_output=True, text=True, timeout=timeout)
    assert r.returncode == 0, f"{args}:\n{r.stdout[-2000:]}\n{r.stderr[-2000:]}"
    return r.stdout


def test_linear_regression_example():
    out = _run(["examples/linear_regression.py", "--strategy", "AllReduce",
                "--epochs", "120"])
    assert "final loss" in out


def test_linear_regression_ps():
    out = _run(["examples/linear_regression.py", "--strategy", "PS",
                "--epochs", "120"])
    assert "final loss" in out


def test_image_classifier_example():
    out = _run(["examples/image_classifier.py", "--epochs", "1",
                "--steps-per-epoch", "5"])
    assert "avg loss" in out


def test_sentiment_classifier_example():
    out = _run(["examples/sentiment_classifier.py", "--steps", "10"])
    assert "final loss" in out


def test_imagenet_benchmark_example():
    out = _run(["examples/benchmark/imagenet.py", "--model", "resnet18",
                "--batch-size", "4", "--steps", "2", "--image-size", "64"])
    assert "images/sec" in out


def test_bert_benchmark_example():
    out = _run(["examples/benchmark/bert.py", "--model", "bert_tiny",
                "--batch-size", "2", "--seq-len", "16", "--steps", "2"])
    assert "seq/sec" in out


def test_ncf_benchmark_example():
    out = _run(["examples/benchmark/ncf.py", "--batch-size", "64",
                "--steps", "2"])
    assert "samples/sec" in out


def test_lm1b_example():
    out = _run(["examples/lm1b/lm1b_train.py", "--small", "--batch-size",
                "8", "--seq-len", "10", "--steps", "3"])
    assert "words/sec" in out


def test_serving_example():
    out = _run(["examples/serving.py", "--model", "bert_tiny",
                "--batch", "4", "--seq-len", "32", "--iters", "3"])
    assert "serving bert_tiny" in out
