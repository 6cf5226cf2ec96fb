"""BERT pretraining benchmark (reference examples/benchmark/bert.py) —
BERT-base with the Parallax hybrid strategy (BASELINE config #3), synthetic
token data, sequences/sec."""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))

from autodist_amd import AutoDist
from autodist_amd import strategy as strat
from autodist_amd.models import bert


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--model", default="bert_base",
                        choices=["bert_tiny", "bert_base", "bert_large"])
    parser.add_argument("--autodist_strategy", default="Parallax")
    parser.add_argument("--batch-size", type=int, default=32)
    parser.add_argument("--seq-len", type=int, default=128)
    parser.add_argument("--steps", type=int, default=10)
    args = parser.parse_args()

    use_cuda = torch.cuda.is_available()
    ad = AutoDist(strategy_builder=getattr(strat, args.autodist_strategy)())
    with ad.scope():
        torch.manual_seed(0)
        model = getattr(bert, args.model)()
        optimizer = torch.optim.AdamW(model.parameters(), lr=1e-4,
                                      weight_decay=0.01)

    sess = ad.create_distributed_session()
    device = ad.engine.device
    vocab = model.bert.cfg.vocab_size
    B, S = args.batch_size, args.seq_len
    ids = torch.randint(0, vocab, (B, S), device=device)
    labels = ids.clone()
    labels[:, ::2] = -100  # predict every other position
    nsp = torch.randint(0, 2, (B,), device=device)

    def train_step():
        optimizer.zero_grad()
        with torch.autocast("cuda", torch.bfloat16, enabled=use_cuda):
            loss = model.loss(ids, labels, nsp)
        loss.backward()
        optimizer.step()
        return loss

    for _ in range(3):
        sess.run(train_step)
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        sess.run(train_step)
    if use_cuda:
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    sps = ad.engine.world_size * B * args.steps / dt
    if ad.engine.rank == 0:
        print(f"{args.model} {args.autodist_strategy}: {sps:.1f} seq/sec "
              f"({dt / args.steps * 1e3:.2f} ms/step)")
    sess.close()


if __name__ == "__main__":
    main()
