"""Finetune an OPT checkpoint (reference examples/opt_finetune/
run_clm_flax.py): load HuggingFace weights into the TP-sharded model and
train with @parallelize on synthetic causal-LM batches.

  torchrun --standalone --nproc-per-node N examples/finetune_opt.py \
      --hf /path/to/hf_opt_dir        # or --random 125M (no network)
"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))
import alpa_amd as aa
from alpa_amd.serve.weights import load_opt_hf


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--hf", default=None,
                   help="HF OPTForCausalLM directory (save_pretrained)")
    p.add_argument("--random", default="125M",
                   help="no checkpoint: random-init this OPT size")
    p.add_argument("--batch", type=int, default=8)
    p.add_argument("--seq", type=int, default=512)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--lr", type=float, default=2e-5)
    p.add_argument("--tp", type=int, default=0,
                   help="tensor-parallel degree (default: world size)")
    args = p.parse_args()

    aa.init()
    n = aa.world_size()
    tp = args.tp or n
    method = aa.ShardParallel(logical_mesh_shape=(n // tp, tp))
    mesh = method.resolve_mesh() if n > 1 else None
    dtype = torch.bfloat16 if torch.cuda.is_available() else torch.float32

    if args.hf:
        model = load_opt_hf(args.hf, mesh=mesh, axis=1, dtype=dtype,
                            device=aa.device())
    else:
        from alpa_amd.models.opt import OPTModel, opt_config
        cfg = opt_config(args.random, max_seq_len=args.seq)
        model = OPTModel(cfg, mesh, 1, dtype, aa.device(), init_seed=0)
    cfg = model.cfg

    state = aa.TrainState.create(
        lambda **kw: model, method, lr=args.lr, weight_decay=0.01)

    def loss_fn(m, batch):
        ids, labels = batch["ids"], batch["labels"]
        # differentiable no-cache path (the serving cache would detach
        # the k/v projections)
        logits = m.forward_train(ids)
        from alpa_amd.parallel.layers import vocab_parallel_cross_entropy
        vs = m.lm_head.mesh.axis_index(1) * m.lm_head.out_per_rank \
            if m.lm_head.mesh is not None and m.lm_head.mesh.is_member \
            else 0
        per_tok = vocab_parallel_cross_entropy(
            logits[:, :-1].reshape(-1, logits.shape[-1]),
            labels[:, 1:].reshape(-1), m.lm_head.mesh, 1, vs)
        return per_tok.mean()

    step = aa.parallelize(loss_fn, method=method)
    g = torch.Generator().manual_seed(0)
    t0 = time.perf_counter()
    for i in range(args.steps):
        ids = torch.randint(0, cfg.vocab_size, (args.batch, args.seq),
                            generator=g).to(aa.device())
        loss = step(state, {"ids": ids, "labels": ids})
        if aa.rank() == 0 and i % 2 == 0:
            print(f"step {i}: loss {float(loss):.4f}")
    dt = (time.perf_counter() - t0) / args.steps
    if aa.rank() == 0:
        print(f"{dt*1e3:.1f} ms/step  (OPT "
              f"{args.hf or args.random}, tp{tp})")
    aa.shutdown()


if __name__ == "__main__":
    main()
