"""NCF benchmark (reference examples/benchmark/ncf.py) — MovieLens-sized
NeuMF with sparse embeddings under PartitionedPS (BASELINE config #4),
synthetic interactions, samples/sec."""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))

from autodist_amd import AutoDist
from autodist_amd import strategy as strat
from autodist_amd.models.ncf import ncf_movielens


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--autodist_strategy", default="PartitionedPS")
    parser.add_argument("--batch-size", type=int, default=4096)
    parser.add_argument("--steps", type=int, default=20)
    parser.add_argument("--sharded-embeddings", action="store_true",
                        help="row-shard the tables across ranks (xGMI "
                             "all-to-all) instead of PS routing")
    args = parser.parse_args()

    use_cuda = torch.cuda.is_available()
    ad = AutoDist(strategy_builder=getattr(strat, args.autodist_strategy)())
    with ad.scope():
        torch.manual_seed(0)
        model = ncf_movielens(sparse=not args.sharded_embeddings,
                              sharded=args.sharded_embeddings)
        optimizer = torch.optim.SGD(model.parameters(), lr=0.05)

    sess = ad.create_distributed_session()
    device = ad.engine.device
    B = args.batch_size
    users = torch.randint(0, 138493, (B,), device=device)
    items = torch.randint(0, 26744, (B,), device=device)
    labels = torch.randint(0, 2, (B,), device=device)

    def train_step():
        optimizer.zero_grad()
        loss = model.loss(users, items, labels)
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
        print(f"ncf {args.autodist_strategy}: {sps:.0f} samples/sec "
              f"({dt / args.steps * 1e3:.2f} ms/step)")
    sess.close()


if __name__ == "__main__":
    main()
