"""Linear regression — the plumbing smoke example.

Reference: examples/linear_regression.py (77 LoC). Same structure: build a
1-feature linear model under autodist scope, train with a distributed
session, print the fitted slope/intercept (true values 3.0 / 0.5).

Run: python examples/linear_regression.py [--strategy PS|AllReduce|...]
"""
import argparse
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from autodist_amd import AutoDist
from autodist_amd import strategy as strat

TRUE_W, TRUE_B = 3.0, 0.5
NUM_SAMPLES = 1024


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--strategy", default="PS")
    parser.add_argument("--epochs", type=int, default=200)
    parser.add_argument("--world-size", type=int, default=None)
    args = parser.parse_args()

    rng = np.random.RandomState(0)
    xs = rng.randn(NUM_SAMPLES, 1).astype(np.float32)
    ys = (TRUE_W * xs + TRUE_B +
          0.01 * rng.randn(NUM_SAMPLES, 1)).astype(np.float32)

    builder = getattr(strat, args.strategy)()
    ad = AutoDist(strategy_builder=builder, world_size=args.world_size)

    with ad.scope():
        torch.manual_seed(0)
        model = torch.nn.Linear(1, 1)
        optimizer = torch.optim.SGD(model.parameters(), lr=0.1)

    def train_step(x, y):
        optimizer.zero_grad()
        loss = torch.nn.functional.mse_loss(model(x), y)
        loss.backward()
        optimizer.step()
        return loss

    sess = ad.create_distributed_session()
    loss = None
    for epoch in range(args.epochs):
        loss = sess.run(train_step, feed_dict={"x": xs, "y": ys})
    w = model.weight.item()
    b = model.bias.item()
    print(f"final loss={float(loss):.6f} w={w:.4f} b={b:.4f} "
          f"(true {TRUE_W}/{TRUE_B})")
    assert abs(w - TRUE_W) < 0.1 and abs(b - TRUE_B) < 0.1, "did not converge"
    sess.close()


if __name__ == "__main__":
    main()
