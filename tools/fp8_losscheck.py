"""fp8-vs-bf16 loss-curve check (the numerics evidence VERDICT r1 item 4
asks for): train the same random-init GPT twice on the same synthetic
stream — bf16 GEMMs vs fp8 GEMMs — and compare the loss trajectories.

Pass criterion: mean relative loss gap over the last 50 steps < 1%, and
no divergence/NaN.  Results are committed to profiles/fp8_losscheck.json
and quoted in docs/BENCHMARK.md.

Run on the GPU box:  python tools/fp8_losscheck.py [steps]
"""
import json
import os
import sys

sys.path.insert(0, ".")
import torch

from alpa_amd.global_env import global_config
from alpa_amd.models.gpt import GPTConfig, GPTModel
from alpa_amd.optim import AdamW
from alpa_amd.ops import fp8 as _f8


def run(steps: int, use_fp8: bool):
    global_config.fp8_gemm = use_fp8
    torch.manual_seed(7)
    cfg = GPTConfig(hidden_size=512, num_layers=4, num_heads=8,
                    seq_len=512, vocab_size=8192)
    m = GPTModel(cfg, None, 1, torch.bfloat16, torch.device("cuda"),
                 init_seed=11)
    opt = AdamW(m.parameters(), lr=3e-4, weight_decay=0.01)
    g = torch.Generator().manual_seed(123)
    losses = []
    for i in range(steps):
        ids = torch.randint(0, cfg.vocab_size, (8, cfg.seq_len),
                            generator=g).cuda()
        loss = m.loss(ids, ids)
        for p in m.parameters():
            p.grad = None
        loss.backward()
        opt.step()
        _f8.bump_epoch()
        losses.append(float(loss))
    return losses


def main():
    steps = int(sys.argv[1]) if len(sys.argv) > 1 else 200
    bf16 = run(steps, False)
    fp8 = run(steps, True)
    tail = slice(-50, None)
    import statistics
    gap = [abs(a - b) / max(abs(b), 1e-9)
           for a, b in zip(fp8[tail], bf16[tail])]
    out = {
        "steps": steps,
        "bf16_first_last": [bf16[0], bf16[-1]],
        "fp8_first_last": [fp8[0], fp8[-1]],
        "mean_rel_gap_last50": statistics.mean(gap),
        "max_rel_gap_last50": max(gap),
        "bf16_curve_every10": bf16[::10],
        "fp8_curve_every10": fp8[::10],
        "pass": statistics.mean(gap) < 0.01 and
                all(map(lambda x: x == x, fp8)),
    }
    print(json.dumps({k: v for k, v in out.items()
                      if not k.endswith("every10")}, indent=2))
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/fp8_losscheck.json", "w") as f:
        json.dump(out, f, indent=2)


if __name__ == "__main__":
    main()
