"""Microbenchmark: fused BN kernels vs MIOpen BN + add + relu at ResNet-50
shapes. Run on a GPU box: python tools/bn_microbench.py"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

SHAPES = [  # (N, C, H, W) conv output shapes in ResNet-50 @ bs256
    (256, 64, 112, 112),
    (256, 64, 56, 56),
    (256, 256, 56, 56),
    (256, 128, 28, 28),
    (256, 512, 28, 28),
    (256, 256, 14, 14),
    (256, 1024, 14, 14),
    (256, 512, 7, 7),
    (256, 2048, 7, 7),
]


def time_fn(fn, iters=20, warm=5):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    from autodist_amd.ops.fused_bn import fused_bn_train
    dev = torch.device("cuda")
    print(f"{'shape':>22} {'fusedF':>8} {'fusedFB':>9} {'miopenF':>9} "
          f"{'miopenFB':>9}  ms")
    for (N, C, H, W) in SHAPES:
        x = torch.randn(N, C, H, W, device=dev, dtype=torch.bfloat16
                        ).contiguous(memory_format=torch.channels_last)
        res = torch.randn_like(x)
        w = torch.nn.Parameter(torch.rand(C, device=dev) + 0.5)
        b = torch.nn.Parameter(torch.randn(C, device=dev))
        rm, rv = torch.zeros(C, device=dev), torch.ones(C, device=dev)
        gy = torch.randn_like(x)

        ws = (torch.empty(2 * 512 + 4, C, device=dev),
              torch.empty(2 * 512 + 3, C, device=dev))

        def fused_fwd():
            with torch.no_grad():
                return fused_bn_train(x, w, b, rm, rv, relu=True,
                                      residual=res, ws=ws)

        def fused_fwd_bwd():
            xg = x.detach().requires_grad_(True)
            rg = res.detach().requires_grad_(True)
            y = fused_bn_train(xg, w, b, rm, rv, relu=True, residual=rg, ws=ws)
            y.backward(gy)

        bn = torch.nn.BatchNorm2d(C).to(dev)

        def miopen_fwd():
            with torch.no_grad(), torch.autocast("cuda", torch.bfloat16):
                return torch.relu(bn(x) + res)

        def miopen_fwd_bwd():
            xg = x.detach().requires_grad_(True)
            rg = res.detach().requires_grad_(True)
            with torch.autocast("cuda", torch.bfloat16):
                y = torch.relu(bn(xg) + rg)
            y.backward(gy)

        tf = time_fn(fused_fwd)
        tfb = time_fn(fused_fwd_bwd)
        mf = time_fn(miopen_fwd)
        mfb = time_fn(miopen_fwd_bwd)
        gb = N * C * H * W * 2 / 1e9
        print(f"{str((N,C,H,W)):>22} {tf:8.3f} {tfb:9.3f} {mf:9.3f} "
              f"{mfb:9.3f}  ({gb:.2f} GB/tensor)")


if __name__ == "__main__":
    main()
