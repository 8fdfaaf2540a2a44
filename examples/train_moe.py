"""Train a GShard-style MoE GPT (the reference's suite_auto_moe workflow
on synthetic data): top-2 gated experts, expert-parallel all-to-all when
run over multiple GPUs (BASELINE config 4).

  torchrun --standalone --nproc-per-node N examples/train_moe.py \
      --hidden 1024 --layers 8 --experts 8 --steps 10
"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))
import alpa_amd as aa
from alpa_amd.models.moe import MoEConfig, MoEGPTModel


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--hidden", type=int, default=1024)
    p.add_argument("--layers", type=int, default=8)
    p.add_argument("--heads", type=int, default=16)
    p.add_argument("--experts", type=int, default=8)
    p.add_argument("--seq", type=int, default=1024)
    p.add_argument("--batch", type=int, default=16)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=2)
    args = p.parse_args()

    aa.init()
    n = aa.world_size()
    # (n, 1) mesh: data-parallel over axis 0 with the experts sharded over
    # the SAME axis (ep_axis=0, DeepSpeed-MoE style EP-within-DP) — each
    # rank feeds its own batch and the top-2 dispatch all-to-alls tokens
    # to the expert owners; dense block grads all-reduce over dp while
    # expert grads stay local (marked _expert_parallel).
    method = aa.ShardParallel(logical_mesh_shape=(n, 1))
    dtype = torch.bfloat16 if torch.cuda.is_available() else torch.float32
    cfg = MoEConfig(hidden_size=args.hidden, num_layers=args.layers,
                    num_heads=args.heads, seq_len=args.seq,
                    vocab_size=51200, num_experts=args.experts,
                    moe_every=2, capacity_factor=2.0)
    state = aa.TrainState.create(
        lambda mesh, axis, dtype=dtype, device=None: MoEGPTModel(
            cfg, mesh, axis, dtype, aa.device(), init_seed=1),
        method, lr=1e-4)
    step = aa.parallelize(lambda m, b: m.loss(b["ids"], b["labels"]),
                          method=method)

    per_rank = args.batch
    # distinct batch per dp rank — the printed tokens/s below counts the
    # DISTINCT global batch (per_rank * n), which is only honest when the
    # ranks' data actually differs
    g = torch.Generator().manual_seed(7 + aa.rank())

    def make_batch():
        ids = torch.randint(0, cfg.vocab_size, (per_rank, cfg.seq_len),
                            generator=g).to(aa.device())
        return {"ids": ids, "labels": ids}

    for _ in range(args.warmup):
        step(state, make_batch())
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        loss = step(state, make_batch())
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.steps
    if aa.rank() == 0:
        tokens = per_rank * cfg.seq_len * n
        ep = min(n, cfg.num_experts)
        print(f"loss {float(loss):.4f}  {dt * 1e3:.1f} ms/step  "
              f"{tokens / dt / 1e6:.2f} Mtok/s  "
              f"({cfg.num_experts} experts, top-2, "
              f"{f'dp{n} EP{ep}' if n > 1 else 'serial'})")
    aa.shutdown()


if __name__ == "__main__":
    main()
