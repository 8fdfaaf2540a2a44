"""Continuous-batching throughput: batched varlen prefill admission vs
the round-1 per-request path (VERDICT r1 item 9 measured close).

A burst of requests with mixed prompt lengths arrives at once; measure
wall time and aggregate generated tokens/s for both admission modes.

Run on the GPU box:  python tools/batching_bench.py [--model 13B]
"""
import argparse
import sys
import time

sys.path.insert(0, ".")
import torch

from alpa_amd.models.opt import OPTModel, opt_config
from alpa_amd.serve.batching import ContinuousBatcher, GenRequest


def run(model, reqs_spec, max_batch, one_by_one: bool):
    b = ContinuousBatcher(model, max_batch)
    if one_by_one:
        b._admit = lambda: ContinuousBatcher._admit_one_by_one(
            b, [s for s in range(b.B) if b.slots[s] is None])
    g = torch.Generator().manual_seed(0)
    for (plen, gen) in reqs_spec:
        ids = torch.randint(0, model.cfg.vocab_size, (plen,), generator=g)
        b.submit(GenRequest(prompt_ids=ids, max_new_tokens=gen))
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    b.run_all()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    toks = sum(gen for (_, gen) in reqs_spec)
    return dt, toks / dt


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="13B")
    p.add_argument("--batch", type=int, default=16)
    p.add_argument("--requests", type=int, default=48)
    args = p.parse_args()
    dev = torch.device("cuda")
    cfg = opt_config(args.model, max_seq_len=1024)
    model = OPTModel(cfg, None, 1, torch.bfloat16, dev, init_seed=0)
    g = torch.Generator().manual_seed(7)
    reqs = [(int(torch.randint(32, 512, (1,), generator=g)), 32)
            for _ in range(args.requests)]
    for mode, name in ((False, "batched-prefill"),
                       (True, "per-request-prefill(r1)")):
        dt, tps = run(model, reqs, args.batch, mode)
        print(f"{name:24s} {dt*1e3:8.1f} ms  {tps:8.1f} tok/s "
              f"({args.requests} reqs, prompts 32-512, gen 32)")


if __name__ == "__main__":
    main()
