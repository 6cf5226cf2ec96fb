"""MNIST-shaped CNN classifier under the AutoDist API.

Reference: examples/image_classifier.py (Keras Sequential +
create_distributed_session). Synthetic data (no dataset downloads in the
image); same API flow: build under scope(), then session.run steps.
"""
import argparse
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from autodist_amd import AutoDist
from autodist_amd import strategy as strat


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--strategy", default="PSLoadBalancing")
    parser.add_argument("--epochs", type=int, default=3)
    parser.add_argument("--steps-per-epoch", type=int, default=20)
    args = parser.parse_args()

    ad = AutoDist(strategy_builder=getattr(strat, args.strategy)())
    with ad.scope():
        torch.manual_seed(0)
        model = torch.nn.Sequential(
            torch.nn.Conv2d(1, 16, 3, padding=1), torch.nn.ReLU(),
            torch.nn.MaxPool2d(2),
            torch.nn.Conv2d(16, 32, 3, padding=1), torch.nn.ReLU(),
            torch.nn.MaxPool2d(2),
            torch.nn.Flatten(),
            torch.nn.Linear(32 * 7 * 7, 10))
        optimizer = torch.optim.Adam(model.parameters(), lr=1e-3)

    rng = np.random.RandomState(0)
    xs = rng.randn(512, 1, 28, 28).astype(np.float32)
    ys = rng.randint(0, 10, size=(512,)).astype(np.int64)

    def train_step(x, y):
        optimizer.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        optimizer.step()
        return loss

    sess = ad.create_distributed_session()
    for epoch in range(args.epochs):
        perm = rng.permutation(len(xs))
        total = 0.0
        for i in range(args.steps_per_epoch):
            idx = perm[i * 16:(i + 1) * 16]
            loss = sess.run(train_step, feed_dict={"x": xs[idx], "y": ys[idx]})
            total += float(loss)
        print(f"epoch {epoch}: avg loss {total / args.steps_per_epoch:.4f}")
    sess.close()


if __name__ == "__main__":
    main()
