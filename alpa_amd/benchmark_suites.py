"""Benchmark suites: named (model, parallelism) cases mirroring the
reference's ``benchmark/alpa/suite_manual_gpt.py:16-28`` /
``suite_auto_gpt.py:53-85`` / ``suite_manual_moe.py`` / ``suite_wresnet.py``
— the same spec ladder re-targeted at MI355X node shapes (288 GB HBM3E:
2.6B needs no remat and no pipeline on ONE GPU; the multi-GPU cases shard
for throughput, not to fit).

Each case maps onto a `bench.py` invocation (the driver's contract) or an
examples/ script; `case_command` renders the command line.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, Optional, Tuple


@dataclass(frozen=True)
class BenchmarkCase:
    """(reference BenchmarkCase, benchmark_parallel_utils.py:20)"""
    name: str
    model: str              # bench.py --model name (GPT_SPECS key) or family
    n_gpus: int
    batch_per_gpu: int
    num_micro_batches: int
    seq_len: int = 1024
    parallel: str = "auto"  # "auto" (ILP) | "manual"
    dp: int = 0             # manual only; 0 = n_gpus/tp
    tp: int = 1


# ---- manual GPT ladder (reference suite_manual_gpt.py:16-28) ----
GPT_MANUAL = {
    "gpt-125m-1gpu": BenchmarkCase("gpt-125m-1gpu", "125M", 1, 32, 1),
    "gpt-1.3b-1gpu": BenchmarkCase("gpt-1.3b-1gpu", "1.3B", 1, 32, 2),
    "gpt-2.6b-1gpu": BenchmarkCase("gpt-2.6b-1gpu", "2.6B", 1, 32, 4),
    # the BASELINE.md headline config at 8 GPUs: the reference needed
    # dp2 x op2 x pp2 + remat on 16 GB V100s; on MI355X pure DP fits
    # comfortably and avoids every TP activation all-reduce
    "gpt-2.6b-8gpu": BenchmarkCase("gpt-2.6b-8gpu", "2.6B", 8, 32, 4,
                                   parallel="manual", dp=8, tp=1),
    "gpt-6.7b-1gpu": BenchmarkCase("gpt-6.7b-1gpu", "6.7B", 1, 16, 4),
    "gpt-15b-1gpu": BenchmarkCase("gpt-15b-1gpu", "15B", 1, 8, 4),
    "gpt-15b-8gpu": BenchmarkCase("gpt-15b-8gpu", "15B", 8, 8, 4,
                                  parallel="manual", dp=8, tp=1),
    "gpt-39b-8gpu": BenchmarkCase("gpt-39b-8gpu", "39B", 8, 4, 4,
                                  parallel="manual", dp=4, tp=2),
}

# ---- auto (ILP-searched) GPT (reference suite_auto_gpt.py:53-85) ----
GPT_AUTO = {
    "gpt-2.6b-auto": BenchmarkCase("gpt-2.6b-auto", "2.6B", 8, 32, 4,
                                   parallel="auto"),
    "gpt-1.3b-auto": BenchmarkCase("gpt-1.3b-auto", "1.3B", 4, 32, 4,
                                   parallel="auto"),
}

ALL_SUITES: Dict[str, Dict[str, BenchmarkCase]] = {
    "gpt_manual": GPT_MANUAL,
    "gpt_auto": GPT_AUTO,
}


def case_command(c: BenchmarkCase, steps: int = 8, warmup: int = 3) -> str:
    """Render the torchrun/bench.py command the driver would use."""
    base = (f"python bench.py --gpus {c.n_gpus} --steps {steps} "
            f"--warmup {warmup} --model {c.model} "
            f"--batch-per-gpu {c.batch_per_gpu} --nmb "
            f"{c.num_micro_batches} --seq {c.seq_len} "
            f"--parallel {c.parallel}")
    if c.parallel == "manual":
        base += f" --dp {c.dp} --tp {c.tp}"
    if c.n_gpus > 1:
        base = (f"python -m torch.distributed.run --nnodes=1 "
                f"--nproc-per-node {c.n_gpus} --master-addr 127.0.0.1 "
                + base[len("python "):])
    return base
