"""Microbenchmark: fused MFMA attention vs torch SDPA at BERT shapes.

Measures forward-only and forward+backward time for the hand-written
gfx950 kernels (attention.hip / attention_bwd.hip) against torch SDPA
(AOTriton flash attention on ROCm). Run on a GPU box:

    python tools/attn_microbench.py [--shapes bert_base,bert_large]
"""
import argparse
import math
import sys
import time

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])

SHAPES = {
    # (B, H, S, D)
    "bert_base_s128": (32, 12, 128, 64),
    "bert_base_s512": (8, 12, 512, 64),
    "bert_large_s128": (16, 16, 128, 64),
    "gpt_s1024": (4, 16, 1024, 64),
}


def bench(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3  # ms


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=50)
    args = p.parse_args()
    from autodist_amd.ops.fused_attention import FusedAttentionFn
    results = []
    for name, (B, H, S, D) in SHAPES.items():
        scale = 1.0 / math.sqrt(D)
        q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
        k = torch.randn_like(q)
        v = torch.randn_like(q)
        dout = torch.randn_like(q)

        def sdpa_fwd():
            with torch.no_grad():
                torch.nn.functional.scaled_dot_product_attention(
                    q, k, v, scale=scale)

        def fused_fwd():
            with torch.no_grad():
                from autodist_amd.ops import api
                api.ext().attn_fwd(q, k, v, scale)

        qg = q.clone().requires_grad_(True)
        kg = k.clone().requires_grad_(True)
        vg = v.clone().requires_grad_(True)

        def sdpa_fb():
            o = torch.nn.functional.scaled_dot_product_attention(
                qg, kg, vg, scale=scale)
            torch.autograd.grad(o, (qg, kg, vg), dout)

        def fused_fb():
            o = FusedAttentionFn.apply(qg, kg, vg, scale, None, 0.0, 0)
            torch.autograd.grad(o, (qg, kg, vg), dout)

        def sdpa_fb_drop():
            o = torch.nn.functional.scaled_dot_product_attention(
                qg, kg, vg, scale=scale, dropout_p=0.1)
            torch.autograd.grad(o, (qg, kg, vg), dout)

        def fused_fb_drop():
            o = FusedAttentionFn.apply(qg, kg, vg, scale, None, 0.1, 777)
            torch.autograd.grad(o, (qg, kg, vg), dout)

        row = {
            "shape": name, "B": B, "H": H, "S": S, "D": D,
            "sdpa_fwd_ms": round(bench(sdpa_fwd, args.iters), 4),
            "fused_fwd_ms": round(bench(fused_fwd, args.iters), 4),
            "sdpa_fb_ms": round(bench(sdpa_fb, args.iters), 4),
            "fused_fb_ms": round(bench(fused_fb, args.iters), 4),
            "sdpa_fb_drop_ms": round(bench(sdpa_fb_drop, args.iters), 4),
            "fused_fb_drop_ms": round(bench(fused_fb_drop, args.iters), 4),
        }
        row["fwd_speedup"] = round(row["sdpa_fwd_ms"] / row["fused_fwd_ms"], 3)
        row["fb_speedup"] = round(row["sdpa_fb_ms"] / row["fused_fb_ms"], 3)
        row["fb_drop_speedup"] = round(
            row["sdpa_fb_drop_ms"] / row["fused_fb_drop_ms"], 3)
        results.append(row)
        print(row, flush=True)
    import json
    print("JSON:" + json.dumps(results))


if __name__ == "__main__":
    main()
