"""Serving example: train briefly, checkpoint, reload, batched inference.

Mirrors the reference's deployment story (train under autodist.scope(),
save a single-node-compatible checkpoint, serve it without the
distributed runtime — autodist/checkpoint/saver.py:93-133 +
saved_model_builder.py). The MI355X serving path runs the hand-written
fused kernels in eval mode: MFMA attention (no dropout), fused bf16
LayerNorm, fused-BN running-stat normalize.

    python examples/serving.py [--model bert_tiny] [--batch 8]
"""
import argparse
import os
import sys
import tempfile
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from autodist_amd import AutoDist
from autodist_amd.checkpoint.saver import Saver
from autodist_amd.models import bert
from autodist_amd.strategy import AllReduce


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="bert_tiny",
                   choices=["bert_tiny", "bert_base"])
    p.add_argument("--batch", type=int, default=8)
    p.add_argument("--seq-len", type=int, default=64)
    p.add_argument("--iters", type=int, default=20)
    args = p.parse_args()
    use_cuda = torch.cuda.is_available()

    # ---- 1. train a few steps under the engine and checkpoint -----------
    ad = AutoDist(strategy_builder=AllReduce())
    with ad.scope():
        torch.manual_seed(0)
        model = getattr(bert, args.model)()
        opt = torch.optim.AdamW(model.parameters(), lr=1e-4)
    saver = Saver(ad.graph_item)
    sess = ad.create_distributed_session()
    device = ad.engine.device
    vocab = model.bert.cfg.vocab_size
    ids = torch.randint(0, vocab, (args.batch, args.seq_len), device=device)
    labels = ids.clone()
    labels[:, ::2] = -100

    def train_step():
        opt.zero_grad()
        with torch.autocast("cuda", torch.bfloat16, enabled=use_cuda):
            loss = model.loss(ids, labels)
        loss.backward()
        opt.step()
        return loss

    for _ in range(3):
        sess.run(train_step)
    ckpt_dir = tempfile.mkdtemp(prefix="autodist_serving_")
    path = saver.save(os.path.join(ckpt_dir, "model.pt"))
    sess.close()
    print(f"trained 3 steps, checkpoint at {path}")

    # ---- 2. serve: fresh process-style reload, no distributed runtime ---
    torch.manual_seed(1)
    served = getattr(bert, args.model)()
    state = torch.load(path, weights_only=False)
    served.load_state_dict(state["model"])
    dev = torch.device("cuda") if use_cuda else torch.device("cpu")
    served = served.to(dev).eval()
    req = torch.randint(0, vocab, (args.batch, args.seq_len), device=dev)
    mask = torch.ones(args.batch, args.seq_len, dtype=torch.long, device=dev)
    mask[:, -7:] = 0  # ragged padding
    with torch.no_grad(), torch.autocast("cuda", torch.bfloat16,
                                         enabled=use_cuda):
        served(req, attention_mask=mask)  # warmup
        if use_cuda:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            logits, nsp = served(req, attention_mask=mask)
        if use_cuda:
            torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    sps = args.batch * args.iters / dt
    assert torch.isfinite(logits).all()
    print(f"serving {args.model}: {sps:.1f} seq/s "
          f"({dt / args.iters * 1e3:.2f} ms/batch of {args.batch}, "
          f"masked fused-MFMA attention eval path)")


if __name__ == "__main__":
    main()
