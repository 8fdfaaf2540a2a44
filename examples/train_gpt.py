"""Train a GPT with auto-parallelization (the reference's examples/gpt2
workflow on synthetic data).

  torchrun --standalone --nproc-per-node N examples/train_gpt.py \
      --model 1.3B --steps 20
"""
import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))
import alpa_amd as aa
from alpa_amd.data_loader import DataLoader, synthetic_lm_batches
from alpa_amd.models.gpt import GPTModel, gpt_config
from alpa_amd.serialization import save_train_state


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="125M")
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--batch", type=int, default=8)
    p.add_argument("--nmb", type=int, default=2)
    p.add_argument("--ckpt", default=None)
    args = p.parse_args()

    aa.init()
    cfg = gpt_config(args.model, seq_len=512)
    n = aa.world_size()
    method = aa.ShardParallel(
        num_micro_batches=args.nmb,
        model_hint={"family": "gpt", "hidden": cfg.hidden_size,
                    "layers": cfg.num_layers, "vocab": cfg.vocab_size,
                    "tokens": args.batch * cfg.seq_len // args.nmb})

    def build(mesh=None, axis=1, dtype=torch.float32, device=None):
        if torch.cuda.is_available():
            dtype = torch.bfloat16
        return GPTModel(cfg, mesh, axis, dtype, device, init_seed=0)

    state = aa.TrainState.create(build, method, lr=3e-4, weight_decay=0.01)
    step = aa.parallelize(lambda m, b: m.loss(b["ids"], b["labels"]),
                          method=method)

    batches = synthetic_lm_batches(args.steps, args.batch * n, cfg.seq_len,
                                   cfg.vocab_size)
    loader = DataLoader(batches, mesh=state.mesh, device=aa.device())
    for i, batch in enumerate(loader):
        loss = step(state, batch)
        if aa.rank() == 0:
            print(f"step {i}: loss {float(loss):.4f}")
    if args.ckpt:
        save_train_state(args.ckpt, state)
        if aa.rank() == 0:
            print(f"checkpoint saved to {args.ckpt}")
    aa.shutdown()


if __name__ == "__main__":
    main()
