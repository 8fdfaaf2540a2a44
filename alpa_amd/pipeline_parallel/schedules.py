"""Pipeline microbatch schedules: GPipe, 1F1B, inference.

Semantics mirror the reference's ``alpa/pipeline_parallel/schedules.py``
(GpipeSchedule:192, PipeDreamFlush:271, InferenceSchedule:393): a schedule
is, per stage, an ordered list of instructions executed by that stage's
submesh.  Instructions here are (op, microbatch) pairs; the runtime maps
them onto compute + p2p RCCL traffic.

Device-free and unit-tested on invariants (like the reference's
tests/pipeline_parallel/test_schedules.py pattern).
"""
from __future__ import annotations

from typing import List, Tuple

# instruction ops
FWD = "F"   # forward microbatch
BWD = "B"   # backward microbatch


def gpipe_schedule(num_stages: int, num_microbatches: int
                   ) -> List[List[Tuple[str, int]]]:
    """All forwards, then all backwards (reference GpipeSchedule:192)."""
    out = []
    for s in range(num_stages):
        instrs = [(FWD, i) for i in range(num_microbatches)]
        instrs += [(BWD, i) for i in reversed(range(num_microbatches))]
        out.append(instrs)
    return out


def one_f_one_b_schedule(num_stages: int, num_microbatches: int
                         ) -> List[List[Tuple[str, int]]]:
    """PipeDream-Flush / 1F1B (reference PipeDreamFlush:271).

    Stage s runs min(num_stages - s - 1, M) warmup forwards, then steady
    alternating F/B, then cooldown backwards.  Peak live activations on
    stage s is min(num_stages - s, M) — the memory advantage over GPipe.
    """
    P, M = num_stages, num_microbatches
    out = []
    for s in range(P):
        warmup = min(P - s - 1, M)
        steady = M - warmup
        instrs: List[Tuple[str, int]] = []
        for i in range(warmup):
            instrs.append((FWD, i))
        for i in range(steady):
            instrs.append((FWD, warmup + i))
            instrs.append((BWD, i))
        for i in range(steady, M):
            instrs.append((BWD, i))
        out.append(instrs)
    return out


def inference_schedule(num_stages: int, num_microbatches: int
                       ) -> List[List[Tuple[str, int]]]:
    """Forward-only pipeline (reference InferenceSchedule:393)."""
    return [[(FWD, i) for i in range(num_microbatches)]
            for s in range(num_stages)]


def overlap_friendly_1f1b_schedule(num_stages: int, num_microbatches: int,
                                   extra_warmup: int = 1
                                   ) -> List[List[Tuple[str, int]]]:
    """1F1B with extra forward warmup so cross-stage sends can overlap
    compute (reference OverlapFriendlyPipeDreamSchedule:452): stage s runs
    min(P - s - 1 + extra_warmup, M) warmup forwards — deeper in-flight
    buffering trades a little activation memory for comm/compute overlap
    headroom on the xGMI links."""
    P, M = num_stages, num_microbatches
    out = []
    for s in range(P):
        warmup = min(P - s - 1 + extra_warmup, M)
        steady = M - warmup
        instrs: List[Tuple[str, int]] = []
        for i in range(warmup):
            instrs.append((FWD, i))
        for i in range(steady):
            instrs.append((FWD, warmup + i))
            instrs.append((BWD, i))
        for i in range(steady, M):
            instrs.append((BWD, i))
        out.append(instrs)
    return out


def make_schedule(name: str, num_stages: int, num_microbatches: int):
    if name == "gpipe":
        return gpipe_schedule(num_stages, num_microbatches)
    if name == "1f1b":
        return one_f_one_b_schedule(num_stages, num_microbatches)
    if name == "inference":
        return inference_schedule(num_stages, num_microbatches)
    if name == "1f1b_overlap_friendly":
        return overlap_friendly_1f1b_schedule(num_stages, num_microbatches)
    raise ValueError(f"unknown schedule {name!r}")


def peak_live_activations(instrs: List[Tuple[str, int]]) -> int:
    """Max number of microbatches with a live forward (schedule analysis)."""
    live = 0
    peak = 0
    for op, _ in instrs:
        if op == FWD:
            live += 1
            peak = max(peak, live)
        elif op == BWD:
            live -= 1
    return peak
