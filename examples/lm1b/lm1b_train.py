"""LM1B LSTM LM training via the `autodist.function` API.

Reference: examples/lm1b/lm1b_train.py (uses autodist.function,
lm1b_train.py:62) + language_model.py. Strategy default: simulator-selected
(AutoStrategy picks PartitionedAR-family for the big embedding/softmax
matrices on xGMI — BASELINE config #5).
"""
import argparse
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))

from autodist_amd import AutoDist
from autodist_amd import strategy as strat
from autodist_amd.models.lm1b import lm1b_full, lm1b_small


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--autodist_strategy", default="AutoStrategy")
    parser.add_argument("--batch-size", type=int, default=128)
    parser.add_argument("--seq-len", type=int, default=20)
    parser.add_argument("--steps", type=int, default=20)
    parser.add_argument("--sharded-softmax", action="store_true",
                        help="vocab-parallel tied projection + sharded CE "
                             "(parallel/vocab_parallel.py)")
    parser.add_argument("--small", action="store_true",
                        help="10k-vocab model (CPU-sized)")
    args = parser.parse_args()

    use_cuda = torch.cuda.is_available()
    ad = AutoDist(strategy_builder=getattr(strat, args.autodist_strategy)())
    with ad.scope():
        torch.manual_seed(0)
        kw = {"sharded_softmax": args.sharded_softmax}
        model = lm1b_small(**kw) if args.small else lm1b_full(**kw)
        optimizer = torch.optim.Adagrad(model.parameters(), lr=0.01)
    vocab = model.emb.num_embeddings

    @ad.function
    def train_step(tokens, targets):
        optimizer.zero_grad()
        with torch.autocast("cuda", torch.bfloat16, enabled=use_cuda):
            loss = model.loss(tokens, targets)
        loss.backward()
        optimizer.step()
        return loss

    rng = np.random.RandomState(0)
    B, S = args.batch_size, args.seq_len
    data = rng.randint(0, vocab, size=(B, S + 1)).astype(np.int64)
    for _ in range(3):
        train_step(data[:, :-1], data[:, 1:])
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    loss = None
    for _ in range(args.steps):
        loss = train_step(data[:, :-1], data[:, 1:])
    if use_cuda:
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    wps = ad.engine.world_size * B * S * args.steps / dt
    if ad.engine.rank == 0:
        print(f"lm1b {args.autodist_strategy}: {wps:.0f} words/sec, "
              f"loss {float(loss):.3f}")


if __name__ == "__main__":
    main()
