"""Graph FLOP counting for TFLOPS reporting.

Capability analog of the reference's HLO FLOP counter (N9 in SURVEY.md
§2.3: ``xe.hlo_module_count_flop_dot_conv_only`` called from
``shard_parallel/compile_executable.py:136,183``) — a dot/conv-only count
of the compiled step, used to report model TFLOPS.  Here the count comes
from torch's dispatch-level FlopCounterMode over the actual fwd+bwd,
which sees exactly the GEMMs the step launches; it cross-checks the
closed-form accounting bench.py uses (alpa's formula,
``benchmark/alpa/util.py:65-89``).
"""
from __future__ import annotations

from typing import Callable

import torch


def count_step_flops(step_fn: Callable[[], torch.Tensor]) -> int:
    """Measured dot/conv FLOPs of one fwd+bwd step: runs
    ``loss = step_fn(); loss.backward()`` under FlopCounterMode and
    returns the total (matmul/bmm/conv/attention ops only, like the
    reference's dot_conv_only counter)."""
    from torch.utils.flop_counter import FlopCounterMode
    with FlopCounterMode(display=False) as fc:
        loss = step_fn()
        loss.backward()
    return int(fc.get_total_flops())


def gpt_analytic_flops(hidden: int, layers: int, vocab: int, batch: int,
                       seq_len: int, remat: bool = False) -> float:
    """Alpa's closed-form GPT fwd+bwd FLOPs (benchmark/alpa/util.py:65):
    ``factor*B*S*H^2*L*(1 + S/(6H)) + 6*B*S*H*V``, factor 72 (96 with
    remat).  This is what bench.py divides wall time into."""
    B, S, H, L, V = batch, seq_len, hidden, layers, vocab
    factor = 96 if remat else 72
    return factor * B * S * H * H * L * (1 + S / (6 * H)) + 6 * B * S * H * V
