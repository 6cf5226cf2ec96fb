"""ImageNet CNN benchmark driver (reference examples/benchmark/imagenet.py:
ResNet/VGG/DenseNet with --autodist_strategy flag, imagenet.py:52-63).

Synthetic ImageNet-shape data; strategy selectable; reports images/sec.
This is the example-level twin of the repo-root bench.py contract.
"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))

from autodist_amd import AutoDist
from autodist_amd import strategy as strat
from autodist_amd.models import densenet, inception, resnet, vgg

MODELS = {
    "resnet18": resnet.resnet18,
    "resnet50": resnet.resnet50,
    "resnet101": resnet.resnet101,
    "vgg16": lambda **kw: vgg.vgg16(batch_norm=True, **kw),
    "densenet121": densenet.densenet121,
    "inception_v3": inception.inception_v3,
}


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--model", default="resnet50",
                        choices=sorted(MODELS))
    parser.add_argument("--autodist_strategy", default="AllReduce")
    parser.add_argument("--batch-size", type=int, default=64)
    parser.add_argument("--steps", type=int, default=10)
    parser.add_argument("--image-size", type=int, default=224)
    args = parser.parse_args()

    use_cuda = torch.cuda.is_available()
    ad = AutoDist(strategy_builder=getattr(strat, args.autodist_strategy)())
    with ad.scope():
        torch.manual_seed(0)
        model = MODELS[args.model](fused=use_cuda)
        optimizer = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9,
                                    weight_decay=1e-4)

    sess = ad.create_distributed_session()
    device = ad.engine.device
    B = args.batch_size
    x = torch.randn(B, 3, args.image_size, args.image_size, device=device)
    if use_cuda:
        x = x.contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (B,), device=device)

    def train_step():
        optimizer.zero_grad()
        with torch.autocast("cuda", torch.bfloat16, enabled=use_cuda):
            loss = torch.nn.functional.cross_entropy(model(x), y)
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
    ips = ad.engine.world_size * B * args.steps / dt
    if ad.engine.rank == 0:
        print(f"{args.model} {args.autodist_strategy}: {ips:.1f} images/sec "
              f"({dt / args.steps * 1e3:.2f} ms/step)")
    sess.close()


if __name__ == "__main__":
    main()
