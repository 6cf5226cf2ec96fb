"""RCCL collective microbenchmark -> cost-model calibration.

Measures all-reduce time across a size sweep and fits the cost model's
latency + bandwidth terms (simulator/cost_model.py). Run:

  1 GPU (latency calibration, real world-1 RCCL collectives):
      python tools/comm_microbench.py --latency-only
  N GPUs (full bandwidth fit):
      python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
          --master-addr 127.0.0.1 tools/comm_microbench.py

With --save, rank 0 writes autodist_amd/simulator/calibration.json (the
committed constants AutoStrategy loads).
"""
import argparse
import json
import os
import sys
import time

import torch
import torch.distributed as dist

sys.path.insert(0, __file__.rsplit("/", 2)[0])

SIZES = [16 * 1024, 256 * 1024, 1 << 21, 1 << 23, 1 << 25, 1 << 27]


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=30)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--latency-only", action="store_true",
                   help="world-1 forced-collective run: fit only the "
                        "per-collective latency")
    p.add_argument("--save", action="store_true")
    args = p.parse_args()

    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    use_cuda = torch.cuda.is_available()
    dev = torch.device("cuda", int(os.environ.get("LOCAL_RANK", 0))
                       % max(torch.cuda.device_count(), 1)) \
        if use_cuda else torch.device("cpu")
    if use_cuda:
        torch.cuda.set_device(dev)
    backend = "nccl" if use_cuda else "gloo"
    if not dist.is_initialized():
        if "MASTER_ADDR" not in os.environ:
            os.environ["MASTER_ADDR"] = "127.0.0.1"
            os.environ["MASTER_PORT"] = "29531"
        dist.init_process_group(backend, rank=rank, world_size=world)

    samples = []  # (nbytes, world, seconds)
    for nbytes in SIZES:
        t = torch.ones(nbytes // 4, dtype=torch.float32, device=dev)
        for _ in range(args.warmup):
            dist.all_reduce(t)
        if use_cuda:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            dist.all_reduce(t)
        if use_cuda:
            torch.cuda.synchronize()
        secs = (time.perf_counter() - t0) / args.iters
        samples.append((nbytes, world, secs))
        if rank == 0:
            wire = 2.0 * max(world - 1, 1) / max(world, 1) * nbytes
            print(f"allreduce {nbytes / 1024:10.0f} KiB  world={world}  "
                  f"{secs * 1e6:9.1f} us  ({wire / max(secs, 1e-12) / 1e9:.1f}"
                  f" GB/s wire)", flush=True)

    if rank == 0:
        from autodist_amd.simulator.cost_model import CostModel
        cm = CostModel(calibration=None)
        if args.latency_only or world <= 1:
            cm.fit_latency([s for s in samples if s[0] <= 256 * 1024])
            measured = f"world={world} latency-only (1-GPU lease, " \
                       f"bandwidth terms are datasheet-derived)"
        else:
            cm.fit(samples)
            measured = f"world={world} RCCL {backend} full fit"
        out = {"samples": samples, "fit": {
            "allreduce_eff": cm.allreduce_eff,
            "coll_latency": cm.coll_latency}}
        print("JSON:" + json.dumps(out))
        if args.save:
            path = cm.save_calibration(measured_on=measured, samples=samples)
            print(f"saved {path}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
