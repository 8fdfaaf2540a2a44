"""Measure per-layer memory facts on the GPU and record them in the
profiling database (ground the stage DP's memory
model in measurement instead of the "3x boundary" heuristic).

Measured quantities (written to MeshProfilingResult.scalars under
("mi355x", (1,1)) and dumped to gpurun_out/memory_coeffs.json):

- gpt_act_bytes_per_token_hidden        activation bytes ONE transformer
  block pins for backward, per token per hidden unit (bf16)
- gpt_act_bytes_per_token_hidden_remat  same with activation remat (only
  the block boundary survives)
- gpt_state_bytes_per_param             optimizer+grad+param bytes per
  parameter element actually allocated (nominal 12: bf16 p+g, fp32 m+v)

Reference analog: stage_profiling.py:1163 measures per-candidate
max_n_succ_stages from real compilation; here the measured coefficients
feed the same feasibility test in stage_construction.training_dp_search.

Run on the GPU box:  python tools/measure_memory.py [hidden] [seq]
"""
import json
import os
import sys

sys.path.insert(0, ".")
import torch

from alpa_amd.models.gpt import Block, GPTConfig, _PLAIN, _run_block
from alpa_amd.optim import AdamW


def _measure_block_act_bytes(cfg: GPTConfig, batch: int, remat: bool
                             ) -> float:
    """Bytes of autograd-saved activation a single block holds after its
    forward (input excluded)."""
    dev = torch.device("cuda")
    blk = Block(cfg, None, 1, torch.bfloat16, dev, layer_idx=0,
                init_seed=0)
    tokens = batch * cfg.seq_len
    x = torch.randn(batch, cfg.seq_len, cfg.hidden_size,
                    dtype=torch.bfloat16, device=dev, requires_grad=True)
    torch.cuda.synchronize()
    # warm up allocator/workspaces so the delta is activations only
    y = _run_block(blk, x, _PLAIN, remat)
    y.sum().backward()
    del y
    x.grad = None
    for p in blk.parameters():
        p.grad = None
    torch.cuda.synchronize()
    torch.cuda.empty_cache()
    base = torch.cuda.memory_allocated()
    y = _run_block(blk, x, _PLAIN, remat)
    torch.cuda.synchronize()
    held = torch.cuda.memory_allocated() - base
    # subtract the block OUTPUT (it belongs to the next layer's input)
    held -= y.numel() * y.element_size()
    y.sum().backward()
    return max(held, 0.0) / tokens / cfg.hidden_size


def _measure_state_bytes_per_param(cfg: GPTConfig) -> float:
    dev = torch.device("cuda")
    blk = Block(cfg, None, 1, torch.bfloat16, dev, layer_idx=0,
                init_seed=0)
    n_params = sum(p.numel() for p in blk.parameters())
    torch.cuda.synchronize()
    base = torch.cuda.memory_allocated()
    opt = AdamW(blk.parameters(), lr=1e-4)
    for p in blk.parameters():
        p.grad = torch.zeros_like(p)
    torch.cuda.synchronize()
    state = torch.cuda.memory_allocated() - base
    # params themselves (already allocated before `base`)
    state += sum(p.numel() * p.element_size() for p in blk.parameters())
    del opt
    return state / n_params


def main():
    hidden = int(sys.argv[1]) if len(sys.argv) > 1 else 2560
    seq = int(sys.argv[2]) if len(sys.argv) > 2 else 1024
    heads = max(1, hidden // 80)
    cfg = GPTConfig(hidden_size=hidden, num_layers=1, num_heads=heads,
                    seq_len=seq, vocab_size=51200)
    batch = 4
    coeff = _measure_block_act_bytes(cfg, batch, remat=False)
    coeff_remat = _measure_block_act_bytes(cfg, batch, remat=True)
    state_pp = _measure_state_bytes_per_param(cfg)
    out = {
        "gpt_act_bytes_per_token_hidden": coeff,
        "gpt_act_bytes_per_token_hidden_remat": coeff_remat,
        "gpt_state_bytes_per_param": state_pp,
        "config": {"hidden": hidden, "seq": seq, "batch": batch},
    }
    print(json.dumps(out, indent=2))

    # fold into the profiling DB (written under gpurun_out/ so the
    # harness merges it back; copy over prof_database.pkl afterwards)
    from alpa_amd.mesh_profiling import (MeshProfilingResult,
                                         ProfilingResultDatabase)
    db = ProfilingResultDatabase()
    if os.path.exists("prof_database.pkl"):
        db.load("prof_database.pkl")
    key = ("mi355x", (1, 1))
    r = db.data.get(key) or MeshProfilingResult((1, 1))
    if not hasattr(r, "scalars"):
        r.scalars = {}
    r.scalars.update({k: v for k, v in out.items() if k != "config"})
    db.update_one_mesh("mi355x", (1, 1), r)
    os.makedirs("gpurun_out", exist_ok=True)
    db.save("gpurun_out/prof_database.pkl")
    with open("gpurun_out/memory_coeffs.json", "w") as f:
        json.dump(out, f, indent=2)


if __name__ == "__main__":
    main()
