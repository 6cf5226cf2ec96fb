"""Multi-process test harness: gloo world_size>1 on localhost.

Mirrors the reference's per-test process isolation (tests/integration/
test_all.py:55-70 runs each case in a forked Process); here each case runs
in spawned workers joined by a gloo file-store rendezvous.
"""
import os
import tempfile
import traceback

import torch.distributed as dist
import torch.multiprocessing as mp


def _worker(rank, world_size, init_file, fn, args, err_queue):
    try:
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world_size)
        dist.init_process_group(
            "gloo", init_method=f"file://{init_file}",
            rank=rank, world_size=world_size)
        fn(rank, world_size, *args)
        dist.barrier()
        dist.destroy_process_group()
    except Exception:  # noqa: BLE001
        err_queue.put((rank, traceback.format_exc()))
        raise


def run_distributed(fn, world_size=2, args=(), timeout=300):
    """Run fn(rank, world_size, *args) in `world_size` spawned processes."""
    with tempfile.TemporaryDirectory() as td:
        init_file = os.path.join(td, "rendezvous")
        ctx = mp.get_context("spawn")
        err_queue = ctx.SimpleQueue()
        procs = []
        for rank in range(world_size):
            p = ctx.Process(target=_worker,
                            args=(rank, world_size, init_file, fn, args,
                                  err_queue))
            p.start()
            procs.append(p)
        failed = []
        for rank, p in enumerate(procs):
            p.join(timeout)
            if p.is_alive():
                p.terminate()
                failed.append(f"rank {rank} timed out")
            elif p.exitcode != 0:
                failed.append(f"rank {rank} exit {p.exitcode}")
        errs = []
        while not err_queue.empty():
            errs.append(err_queue.get())
        if failed or errs:
            detail = "\n".join(f"--- rank {r} ---\n{tb}" for r, tb in errs)
            raise AssertionError(f"distributed test failed: {failed}\n{detail}")
