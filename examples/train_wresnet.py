"""Train a Wide-ResNet with data parallelism on synthetic CIFAR-shaped
data (the reference's `benchmark/alpa/suite_wresnet.py` / `examples/
imagenet` workflow).

  torchrun --standalone --nproc-per-node N examples/train_wresnet.py \
      --depth 28 --width 10 --steps 20
"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))
import alpa_amd as aa
from alpa_amd.mesh import memory_stats
from alpa_amd.models.wide_resnet import WideResNet


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--depth", type=int, default=16)
    p.add_argument("--width", type=int, default=2)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--batch", type=int, default=64)
    p.add_argument("--nmb", type=int, default=1)
    args = p.parse_args()

    aa.init()
    method = aa.DataParallel(num_micro_batches=args.nmb)
    dtype = torch.bfloat16 if torch.cuda.is_available() else torch.float32
    dev = "cuda" if torch.cuda.is_available() else None

    def build(mesh=None, axis=1, dtype=dtype, device=dev):
        torch.manual_seed(7)  # identical replicas across dp ranks
        return WideResNet(args.depth, args.width, dtype=dtype,
                          device=device)

    state = aa.TrainState.create(build, method, lr=1e-3)
    step = aa.parallelize(lambda m, b: m.loss(*b), method=method)

    g = torch.Generator().manual_seed(1234 + aa.rank())
    per_rank = args.batch // max(aa.world_size(), 1)
    for i in range(args.steps):
        x = torch.randn(per_rank, 3, 32, 32, generator=g).to(
            dtype=dtype, device=dev or "cpu")
        y = torch.randint(0, 10, (per_rank,), generator=g).to(dev or "cpu")
        t0 = time.perf_counter()
        loss = step(state, (x, y))
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        if aa.rank() == 0:
            print(f"step {i}: loss {float(loss):.4f} "
                  f"({(time.perf_counter() - t0) * 1e3:.1f} ms)")
    if aa.rank() == 0:
        print("memory:", memory_stats())
    aa.shutdown()


if __name__ == "__main__":
    main()
