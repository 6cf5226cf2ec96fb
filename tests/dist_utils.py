"""Multi-process test harness: world_size>1 on localhost.

Mirrors the reference's per-test process isolation (tests/integration/
test_all.py:55-70 runs each case in a forked Process); here each case runs
in spawned workers joined by a file-store rendezvous. Backend "gloo" for
CPU-only CI; backend "nccl" (= RCCL on ROCm) for GPU validation — with
fewer GPUs than ranks, ranks share devices modulo the device count (the
2-ranks-on-1-MI355X RCCL validation mode).
"""
import os
import tempfile
import traceback

import torch.distributed as dist
import torch.multiprocessing as mp


def _worker(rank, world_size, init_file, fn, args, err_queue, backend):
    try:
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world_size)
        if backend == "nccl":
            import torch
            os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
            torch.cuda.set_device(rank % torch.cuda.device_count())
        dist.init_process_group(
            backend, init_method=f"file://{init_file}",
            rank=rank, world_size=world_size)
        fn(rank, world_size, *args)
        dist.barrier()
        dist.destroy_process_group()
    except Exception:  # noqa: BLE001
        err_queue.put((rank, traceback.format_exc()))
        raise


def run_distributed(fn, world_size=2, args=(), timeout=300, backend="gloo"):
    """Run fn(rank, world_size, *args) in `world_size` spawned processes."""
    with tempfile.TemporaryDirectory() as td:
        init_file = os.path.join(td, "rendezvous")
        ctx = mp.get_context("spawn")
        err_queue = ctx.SimpleQueue()
        procs = []
        for rank in range(world_size):
            p = ctx.Process(target=_worker,
                            args=(rank, world_size, init_file, fn, args,
                                  err_queue, backend))
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
