"""Train ViT on synthetic images (reference examples/ViT): dp x tp over
the world, bf16 on GPU.

  torchrun --standalone --nproc-per-node N examples/train_vit.py \
      --dp -1 --tp 1
"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))
import alpa_amd as aa
from alpa_amd.models.vit import ViTConfig, ViTModel


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--image", type=int, default=224)
    p.add_argument("--hidden", type=int, default=768)
    p.add_argument("--layers", type=int, default=12)
    p.add_argument("--heads", type=int, default=12)
    p.add_argument("--classes", type=int, default=1000)
    p.add_argument("--batch", type=int, default=64)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--tp", type=int, default=1)
    args = p.parse_args()

    aa.init()
    n = aa.world_size()
    dp = n // args.tp
    method = aa.ShardParallel(logical_mesh_shape=(dp, args.tp))
    dtype = torch.bfloat16 if torch.cuda.is_available() else torch.float32
    cfg = ViTConfig(image_size=args.image, hidden_size=args.hidden,
                    num_layers=args.layers, num_heads=args.heads,
                    num_classes=args.classes)
    state = aa.TrainState.create(
        lambda mesh, axis, dtype=dtype, device=None: ViTModel(
            cfg, mesh, axis, dtype, aa.device(), init_seed=1),
        method, lr=3e-4)
    step = aa.parallelize(lambda m, b: m.loss(b["x"], b["y"]),
                          method=method)

    dp_idx = max(state.mesh.axis_index(0), 0) if state.mesh is not None \
        else 0
    g = torch.Generator().manual_seed(11 + dp_idx)

    def batch():
        x = torch.randn(args.batch, 3, args.image, args.image,
                        generator=g).to(aa.device(), dtype)
        y = torch.randint(0, args.classes, (args.batch,),
                          generator=g).to(aa.device())
        return {"x": x, "y": y}

    for _ in range(args.warmup):
        step(state, batch())
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        loss = step(state, batch())
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.steps
    if aa.rank() == 0:
        imgs = args.batch * dp
        print(f"loss {float(loss):.4f}  {dt*1e3:.1f} ms/step  "
              f"{imgs/dt:.0f} img/s  (ViT {args.hidden}x{args.layers}, "
              f"dp{dp} tp{args.tp})")
    aa.shutdown()


if __name__ == "__main__":
    main()
