"""Flagship benchmark: ResNet-50 data-parallel training, images/sec.

BASELINE.json metric: "images/sec ResNet-50 AllReduce at 1/2/4/8 MI355X;
scaling efficiency". bf16 autocast compute, fp32 params, synthetic
ImageNet-shape data, random-init weights, AllReduce strategy over RCCL/xGMI.

Usage:  python bench.py --gpus N --steps K --warmup W
The driver launches N>1 via torch.distributed.run (one rank per GPU); run
standalone and it re-execs itself under the launcher.
"""
import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--batch-size", type=int, default=512,
                   help="per-GPU batch (weak scaling; 288 GB HBM3E/GPU)")
    p.add_argument("--model", default="resnet50")
    p.add_argument("--strategy", default="AllReduce")
    p.add_argument("--bucket-mb", type=int, default=25)
    p.add_argument("--image-size", type=int, default=224)
    p.add_argument("--no-channels-last", action="store_true")
    p.add_argument("--no-fused-bn", action="store_true",
                   help="disable the hand-written gfx950 fused BN kernels")
    p.add_argument("--hipgraph", choices=["auto", "on", "off"], default="auto",
                   help="capture the train step in a hipGraph (auto: on for "
                        "world_size==1)")
    return p.parse_args()


def relaunch_under_torchrun(args):
    """Self-relaunch with one rank per GPU when run standalone."""
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--nnodes=1", f"--nproc-per-node={args.gpus}",
           "--master-addr", "127.0.0.1", "--master-port", str(port),
           os.path.abspath(__file__)] + sys.argv[1:]
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    os.execvpe(sys.executable, cmd, os.environ)


def main():
    args = parse_args()
    if args.gpus > 1 and "RANK" not in os.environ:
        relaunch_under_torchrun(args)

    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    use_cuda = torch.cuda.is_available()
    device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", 0))) \
        if use_cuda else torch.device("cpu")
    if use_cuda:
        torch.cuda.set_device(device)
        # Exhaustive MIOpen find (~6 min once per conv config, amortized in
        # warmup; the find cache under $HOME makes later runs fast). Without
        # it MIOpen's immediate mode picks naive wrw kernels: 214 img/s vs
        # 8458 measured.
        torch.backends.cudnn.benchmark = True
        torch.backends.cuda.matmul.allow_tf32 = False

    from autodist_amd import strategy as strat
    from autodist_amd.graph_item import GraphItem
    from autodist_amd.models import resnet
    from autodist_amd.parallel.engine import DistributedEngine
    from autodist_amd.resource_spec import ResourceSpec

    torch.manual_seed(1234)
    fused = use_cuda and not args.no_fused_bn
    model = getattr(resnet, args.model)(num_classes=1000, fused=fused)
    model = model.to(device)
    channels_last = use_cuda and not args.no_channels_last
    if channels_last:
        model = model.to(memory_format=torch.channels_last)

    g = GraphItem()
    g.extend_model(model)
    opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9,
                          weight_decay=1e-4)
    g.extend_optimizer_info(opt)
    builder = getattr(strat, args.strategy)()
    rs = ResourceSpec()
    strategy = builder.build(g, rs)
    if len(strategy.graph_config.replicas) != world:
        strategy.graph_config.replicas = [
            f"127.0.0.1:GPU:{r}" for r in range(world)]
    engine = DistributedEngine(g, strategy, rank=rank, world_size=world,
                               device=device,
                               bucket_bytes=args.bucket_mb * 1024 * 1024)
    engine.setup()

    B = args.batch_size
    x = torch.randn(B, 3, args.image_size, args.image_size, device=device)
    if channels_last:
        x = x.contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (B,), device=device)
    loss_fn = torch.nn.CrossEntropyLoss()

    amp_dtype = torch.bfloat16
    amp_enabled = use_cuda

    def step():
        opt.zero_grad()
        with torch.autocast(device_type="cuda", dtype=amp_dtype,
                            enabled=amp_enabled):
            out = model(x)
            loss = loss_fn(out, y)
        loss.backward()
        opt.step()
        return loss

    import torch.distributed as dist

    def barrier_sync():
        if world > 1:
            dist.barrier()
        if use_cuda:
            torch.cuda.synchronize(device)

    # hipGraph capture: the per-step kernel chain (53 fused-BN trios + convs +
    # the fused optimizer launch) is launch-bound in eager mode; replaying a
    # captured graph removes host launch + python overhead entirely.
    use_graph = use_cuda and (args.hipgraph == "on" or
                              (args.hipgraph == "auto" and world == 1))
    run_step = step
    for _ in range(args.warmup):
        step()
    if use_graph:
        try:
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(2):
                    step()
            torch.cuda.current_stream().wait_stream(s)
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                static_loss = step()
            graph.replay()
            torch.cuda.synchronize(device)
            assert bool(torch.isfinite(static_loss).item()), "graph NaN"
            run_step = graph.replay
            if rank == 0:
                print(f"# hipGraph capture OK (loss {static_loss.item():.3f})",
                      file=sys.stderr)
        except Exception as exc:  # noqa: BLE001 - fall back to eager
            print(f"# hipGraph capture failed, eager fallback: {exc}",
                  file=sys.stderr)
            run_step = step
    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        run_step()
    barrier_sync()
    dt = time.perf_counter() - t0
    # MAX elapsed over ranks
    if world > 1:
        t = torch.tensor([dt], dtype=torch.float64,
                         device=device if use_cuda else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        dt = float(t.item())
    engine.drain()

    if rank == 0:
        ips = world * B * args.steps / dt
        result = {
            "metric": "images/sec ResNet-50 AllReduce",
            "value": round(ips, 2),
            "unit": "images/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(dt / args.steps * 1e3, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {"model": args.model, "global_batch": world * B,
                       "seq_len": None, "parallelism": f"dp{world}",
                       "strategy": args.strategy,
                       "image_size": args.image_size},
        }
        print(json.dumps(result))


if __name__ == "__main__":
    main()
