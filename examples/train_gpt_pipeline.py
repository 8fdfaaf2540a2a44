"""Pipeline-parallel GPT training (PipeshardParallel, 1F1B).

  torchrun --standalone --nproc-per-node 4 examples/train_gpt_pipeline.py
"""
import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))
import alpa_amd as aa
from alpa_amd.models.gpt import gpt_config, gpt_pipeline_spec


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="125M")
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--batch", type=int, default=8)
    p.add_argument("--nmb", type=int, default=4)
    p.add_argument("--stages", type=int, default=None)
    args = p.parse_args()

    aa.init()
    cfg = gpt_config(args.model, seq_len=512)
    # --stages N pins the layout; without it the profile-guided search
    # (measured cost DB) picks (stages x submesh) itself
    method = aa.PipeshardParallel(num_micro_batches=args.nmb,
                                  num_stages=args.stages,
                                  stage_option="auto",
                                  schedule="1f1b")
    spec = gpt_pipeline_spec(
        cfg, microbatch_tokens=args.batch * cfg.seq_len // args.nmb)
    state = aa.TrainState.create(spec, method, lr=3e-4)
    step = aa.parallelize(lambda m, b: None, method=method)

    for i in range(args.steps):
        g = torch.Generator().manual_seed(i)
        batch = {
            "ids": torch.randint(0, cfg.vocab_size,
                                 (args.batch, cfg.seq_len), generator=g
                                 ).to(aa.device()),
            "labels": torch.randint(0, cfg.vocab_size,
                                    (args.batch, cfg.seq_len), generator=g
                                    ).to(aa.device()),
        }
        loss = step(state, batch)
        if aa.rank() == 0:
            print(f"step {i}: loss {float(loss):.4f}")
    aa.shutdown()


if __name__ == "__main__":
    main()
