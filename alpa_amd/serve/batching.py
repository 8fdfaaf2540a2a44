"""Continuous batching for decoder serving.

Capability analog of the reference's 1-D batching serving variant
(``examples/llm_serving/model/opt_model_1d.py`` + ``wrapper_1d.py``):
requests of different lengths share one decode batch.  Slot-based design
for the gfx950 varlen kernel: a fixed [max_batch, heads, max_len, d] KV
cache; each slot carries its own length; every decode step runs ONE
fused varlen attention over all slots (per-slot masking happens inside
the kernel via kv_lens); finished slots are retired and refilled from
the pending queue without stalling the others — new requests prefill
into their slot while the rest keep decoding.
"""
from __future__ import annotations

from collections import deque
from dataclasses import dataclass, field
from typing import List, Optional

import torch


@dataclass
class GenRequest:
    prompt_ids: torch.Tensor          # [S0] long
    max_new_tokens: int
    eos_token: Optional[int] = None
    output: List[int] = field(default_factory=list)
    done: bool = False


class _SlotCacheView:
    """cache.k/v views of one slot, so the model's ordinary prefill path
    writes straight into the shared buffer."""

    def __init__(self, cache, slot: int):
        self.k = [t[slot:slot + 1] for t in cache.k]
        self.v = [t[slot:slot + 1] for t in cache.v]
        self.length = 0


class ContinuousBatcher:
    """submit() requests, call step() in a loop (or run_all()); each step
    decodes one token for EVERY active slot and admits pending requests
    into freed slots."""

    def __init__(self, model, max_batch: int):
        self.model = model
        self.B = max_batch
        self.cache = model.new_cache(max_batch)
        dev = self.cache.k[0].device
        self.lens = torch.zeros(max_batch, dtype=torch.long, device=dev)
        self.cur = torch.zeros(max_batch, 1, dtype=torch.long, device=dev)
        self.slots: List[Optional[GenRequest]] = [None] * max_batch
        self.pending: deque = deque()
        self.device = dev
        self.max_len = self.cache.k[0].shape[2]

    def submit(self, req: GenRequest):
        self.pending.append(req)

    @property
    def num_active(self) -> int:
        return sum(r is not None for r in self.slots)

    def _admit(self):
        free = [s for s in range(self.B) if self.slots[s] is None]
        if not hasattr(self.model, "forward_prefill") or \
                min(len(free), len(self.pending)) < 16:
            # small admissions: exact-length per-request prefill wins
            # (padding + grouping waste > launch savings; measured
            # crossover between 16- and 32-slot admission waves at
            # OPT-13B — tools/batching_bench.py: 570 vs 520 tok/s at 16,
            # 781 vs 836 at 32)
            return self._admit_one_by_one(free)
        # admit group after group until slots or pending run out (each
        # group = one batched prefill launch into a contiguous slot run)
        while True:
            if not self._admit_group():
                free = [s for s in range(self.B)
                        if self.slots[s] is None]
                if free and self.pending:
                    self._admit_one_by_one(free)
                return

    def _admit_group(self) -> bool:
        """One batched prefill launch into a CONTIGUOUS run of free
        slots: the stage writes K/V straight into the live cache views
        (a scatter via a temp cache measured ~0.8 s/wave of pure HBM
        copy at 13B — tools/batching_bench.py).  Requests are admitted
        longest-first with a 25% padded-work bound so right-padding
        cannot regress below the per-request path."""
        free = [s for s in range(self.B) if self.slots[s] is None]
        if not free or not self.pending:
            return False
        # longest contiguous free run
        runs, cur = [], [free[0]]
        for s_ in free[1:]:
            if s_ == cur[-1] + 1:
                cur.append(s_)
            else:
                runs.append(cur)
                cur = [s_]
        runs.append(cur)
        run = max(runs, key=len)
        n = min(len(run), len(self.pending))
        # length-grouped admission bounded by padded-work waste
        cand = sorted([self.pending.popleft() for _ in range(n)],
                      key=lambda r: -r.prompt_ids.numel())
        take = 1
        real = cand[0].prompt_ids.numel()
        smax0 = real
        for r in cand[1:]:
            L = r.prompt_ids.numel()
            if smax0 * (take + 1) > 1.25 * (real + L):
                break
            real += L
            take += 1
        for r in cand[take:][::-1]:
            self.pending.appendleft(r)  # preserve order for next admit
        reqs = cand[:take]
        n = take
        slots = run[:n]
        s0 = slots[0]
        lens = torch.tensor([r.prompt_ids.numel() for r in reqs])
        Smax = int(lens.max())
        ids = torch.zeros(n, Smax, dtype=torch.long, device=self.device)
        for j, r in enumerate(reqs):
            assert int(lens[j]) + r.max_new_tokens < self.max_len, \
                "prompt too long"
            ids[j, :int(lens[j])] = r.prompt_ids.to(self.device).view(-1)

        class _RunView:
            pass
        view = _RunView()
        view.k = [t[s0:s0 + n] for t in self.cache.k]
        view.v = [t[s0:s0 + n] for t in self.cache.v]
        view.length = 0
        with torch.no_grad():
            logits = self.model.forward_prefill(ids, lens, view)
        toks = self.model.greedy_token(logits)
        for j, (s_, r) in enumerate(zip(slots, reqs)):
            self.slots[s_] = r
            self.lens[s_] = int(lens[j])
            self.cur[s_, 0] = toks[j]
            r.output.append(int(toks[j]))
            self._maybe_finish(s_)
        return True

    def _admit_one_by_one(self, free):
        for s in free:
            if not self.pending:
                break
            req = self.pending.popleft()
            ids = req.prompt_ids.to(self.device).view(1, -1)
            S0 = ids.shape[1]
            assert S0 + req.max_new_tokens < self.max_len, "prompt too long"
            view = _SlotCacheView(self.cache, s)
            with torch.no_grad():
                logits = self.model.forward_step(ids, view)
            tok = self.model.greedy_token(logits)
            self.slots[s] = req
            self.lens[s] = S0
            self.cur[s, 0] = tok[0]
            req.output.append(int(tok[0]))
            self._maybe_finish(s)

    def _maybe_finish(self, s: int):
        req = self.slots[s]
        if req is None:
            return
        if len(req.output) >= req.max_new_tokens or (
                req.eos_token is not None and
                req.output[-1] == req.eos_token):
            req.done = True
            self.slots[s] = None
            self.lens[s] = 0

    def step(self) -> List[GenRequest]:
        """Admit + one decode tick.  Returns requests finished this tick."""
        self._admit()
        active = [s for s in range(self.B) if self.slots[s] is not None]
        if not active:
            return []
        # free slots ride along with lens=0 (their rows are fully masked
        # by the kernel and their outputs ignored)
        logits = self.model.forward_decode(self.cur, self.cache, self.lens)
        toks = self.model.greedy_token(logits)
        finished = []
        for s in active:
            self.lens[s] += 1
            self.cur[s, 0] = toks[s]
            self.slots[s].output.append(int(toks[s]))
            req = self.slots[s]
            self._maybe_finish(s)
            if req.done:
                finished.append(req)
        return finished

    def run_all(self) -> None:
        while self.pending or self.num_active:
            self.step()
