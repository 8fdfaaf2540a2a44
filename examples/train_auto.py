"""Automatic parallelization of a PLAIN torch model — no zoo membership,
no model_hint, no manual specs (the reference's headline capability:
@parallelize any program; here capture -> per-node ILP -> executed plan).

  torchrun --standalone --nproc-per-node N examples/train_auto.py

The model below is ordinary PyTorch.  TrainState.create_auto traces it
(torch.fx), the ILP picks a per-op sharding plan over the (dp, tp)
factorizations of the world under the memory budget, and the plan is
EXECUTED by converting the matched modules to parallel layers with
automatic resharding on mismatched edges.
"""
import argparse
import os
import sys
import time

import torch
import torch.nn as nn
import torch.nn.functional as F

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))
import alpa_amd as aa


class PlainTransformer(nn.Module):
    """Written with raw reshape/matmul attention — nothing the framework
    knows about."""

    def __init__(self, h=1024, heads=16, layers=4, vocab=32000, seq=512):
        super().__init__()
        torch.manual_seed(0)
        self.emb = nn.Embedding(vocab, h)
        self.blocks = nn.ModuleList()
        for _ in range(layers):
            blk = nn.ModuleDict({
                "ln1": nn.LayerNorm(h), "qkv": nn.Linear(h, 3 * h),
                "proj": nn.Linear(h, h), "ln2": nn.LayerNorm(h),
                "fc1": nn.Linear(h, 4 * h), "fc2": nn.Linear(4 * h, h),
            })
            self.blocks.append(blk)
        self.heads = heads
        self.ln_f = nn.LayerNorm(h)
        self.head = nn.Linear(h, vocab, bias=False)

    def forward(self, ids):
        x = self.emb(ids)
        B, S, H = x.shape
        for blk in self.blocks:
            y = blk["ln1"](x)
            qkv = blk["qkv"](y).reshape(B, S, 3, self.heads,
                                        H // self.heads)
            q = qkv[:, :, 0].transpose(1, 2)
            k = qkv[:, :, 1].transpose(1, 2)
            v = qkv[:, :, 2].transpose(1, 2)
            a = torch.matmul(q, k.transpose(-2, -1)) / (H // self.heads) ** .5
            a = F.softmax(a, dim=-1)
            o = torch.matmul(a, v).transpose(1, 2).reshape(B, S, H)
            x = x + blk["proj"](o)
            x = x + blk["fc2"](F.gelu(blk["fc1"](blk["ln2"](x))))
        return self.head(self.ln_f(x))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--hidden", type=int, default=1024)
    p.add_argument("--layers", type=int, default=4)
    p.add_argument("--vocab", type=int, default=32000)
    p.add_argument("--seq", type=int, default=512)
    p.add_argument("--batch", type=int, default=8)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--budget-gb", type=float, default=0.0,
                   help="per-device memory budget (forces sharding)")
    args = p.parse_args()

    aa.init()
    method = aa.ShardParallel(num_micro_batches=1)
    if args.budget_gb:
        method.auto_sharding_option.memory_budget_per_device = \
            args.budget_gb * 1e9
    x = torch.randint(0, args.vocab, (args.batch, args.seq))
    state = aa.TrainState.create_auto(
        lambda: PlainTransformer(args.hidden, 16, args.layers,
                                 args.vocab, args.seq),
        x, method, lr=1e-4)
    if aa.rank() == 0:
        def is_sharded(v):
            return (("_col" in v and not v.endswith("colNone"))
                    or "_row" in v or "_vocab" in v)
        picks = state.plan.choices
        print(f"plan: mesh {state.plan.mesh_shape}, "
              f"{sum(1 for v in picks.values() if is_sharded(v))}"
              f" sharded weight ops / {len(picks)} solved nodes")

    step = aa.parallelize(
        lambda m, b: F.cross_entropy(
            m(b["ids"]).reshape(-1, args.vocab), b["labels"].reshape(-1)),
        method=method)
    # data is sharded over the DP axis: ranks in the same dp row (tp
    # peers) must consume IDENTICAL batches
    dp_idx = state.mesh.axis_index(0) if state.mesh is not None else 0
    g = torch.Generator().manual_seed(3 + max(dp_idx, 0))
    t0 = time.perf_counter()
    for i in range(args.steps):
        ids = torch.randint(0, args.vocab, (args.batch, args.seq),
                            generator=g).to(aa.device())
        loss = step(state, {"ids": ids, "labels": ids})
    dt = (time.perf_counter() - t0) / args.steps
    if aa.rank() == 0:
        print(f"loss {float(loss):.4f}  {dt*1e3:.1f} ms/step")
    aa.shutdown()


if __name__ == "__main__":
    main()
