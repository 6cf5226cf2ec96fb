"""Sentiment classifier: embedding (sparse) + dense mix under Parallax.

Reference: examples/sentiment_classifier.py — exercises the hybrid path
(sparse embedding grads -> PS, dense grads -> AllReduce).
"""
import argparse
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from autodist_amd import AutoDist
from autodist_amd.strategy import Parallax


class SentimentNet(torch.nn.Module):
    def __init__(self, vocab=5000, dim=64):
        super().__init__()
        self.emb = torch.nn.Embedding(vocab, dim, sparse=True)
        self.fc1 = torch.nn.Linear(dim, 64)
        self.fc2 = torch.nn.Linear(64, 2)

    def forward(self, tokens):
        x = self.emb(tokens).mean(dim=1)
        return self.fc2(torch.relu(self.fc1(x)))


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--steps", type=int, default=50)
    args = parser.parse_args()

    ad = AutoDist(strategy_builder=Parallax())
    with ad.scope():
        torch.manual_seed(0)
        model = SentimentNet()
        optimizer = torch.optim.SGD(model.parameters(), lr=0.1)

    rng = np.random.RandomState(0)
    tokens = rng.randint(0, 5000, size=(2048, 24)).astype(np.int64)
    labels = (tokens.mean(axis=1) > 2500).astype(np.int64)  # learnable rule

    def train_step(x, y):
        optimizer.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        optimizer.step()
        return loss

    sess = ad.create_distributed_session()
    loss = None
    for step in range(args.steps):
        lo = step * 64 % (len(tokens) - 64)
        loss = sess.run(train_step, feed_dict={
            "x": tokens[lo:lo + 64], "y": labels[lo:lo + 64]})
    print(f"final loss {float(loss):.4f}")
    sess.close()


if __name__ == "__main__":
    main()
