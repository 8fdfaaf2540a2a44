"""Shared autoregressive generation: greedy, sampling + beam search.

Capability analog of the reference's huggingface GenerationMixin adapter
(``examples/llm_serving/model/wrapper.py:501`` wraps alpa executables in
``GenerationMixin``; beam reorder via index-select executables
:115-182).  A decoder model family mixes this in by providing:

  - ``self.mesh`` / ``self.axis``       TP mesh + axis (or mesh=None)
  - ``self.lm_head.out_per_rank``       vocab shard size
  - ``self.new_cache(batch)``           preallocated KV cache
  - ``self.forward_step(ids, cache)``   prefill/decode -> last logits

Used by OPTModel and CodeGenModel.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist

from ..mesh import is_distributed


class GenerationMixin:

    @torch.no_grad()
    def greedy_token(self, logits: torch.Tensor) -> torch.Tensor:
        """Vocab-parallel argmax: local top-1 + cross-tp argmax (no logits
        gather)."""
        tp = self.mesh.axis_size(self.axis) if self.mesh is not None else 1
        local_max, local_idx = logits.float().max(dim=-1)
        if tp == 1 or not is_distributed():
            return local_idx
        vocab_start = self.mesh.axis_index(self.axis) * \
            self.lm_head.out_per_rank
        pair = torch.stack([local_max,
                            (local_idx + vocab_start).float()], dim=-1)
        gathered = torch.empty(tp * pair.shape[0], pair.shape[1],
                               dtype=pair.dtype, device=pair.device)
        dist.all_gather_into_tensor(gathered, pair.contiguous(),
                                    group=self.mesh.axis_group(self.axis))
        gathered = gathered.view(tp, pair.shape[0], pair.shape[1])
        best = gathered[:, :, 0].argmax(dim=0)
        idx = gathered[best, torch.arange(best.shape[0],
                                          device=best.device), 1]
        return idx.long()

    @torch.no_grad()
    def sample_token(self, logits: torch.Tensor, temperature: float = 1.0,
                     top_k: int = 0, top_p: float = 1.0,
                     generator: Optional[torch.Generator] = None
                     ) -> torch.Tensor:
        """Temperature / top-k / top-p sampling, vocab-parallel-safe: the
        candidate set is the global top-max(top_k, 64) (covering all mass
        that survives top-p in practice); every TP rank draws the SAME
        sample (shared generator seed), so SPMD decode stays in lockstep.
        """
        kk = max(top_k if top_k > 0 else 0, 64)
        lp, idx = self._log_probs_topk(logits.float() / max(temperature,
                                                            1e-6), kk)
        if top_k > 0:
            lp = lp[:, :top_k]
            idx = idx[:, :top_k]
        probs = torch.softmax(lp, dim=-1)
        if top_p < 1.0:
            cum = probs.cumsum(dim=-1)
            # keep the smallest prefix reaching top_p (always >= 1 token)
            cut = (cum - probs) >= top_p
            probs = probs.masked_fill(cut, 0.0)
            probs = probs / probs.sum(dim=-1, keepdim=True)
        choice = torch.multinomial(probs, 1, generator=generator)
        return idx.gather(1, choice)[:, 0]

    @torch.no_grad()
    def _log_probs_topk(self, logits: torch.Tensor, k: int):
        """Global top-k log-probs over the vocab-parallel logits without
        gathering the full vocab: the softmax normalizer comes from two
        scalar-per-row all-reduces (max, sumexp); candidates are local
        top-k + a tp-wide gather of (value, global index) pairs.  Every
        rank computes identical results (SPMD-safe)."""
        tp = self.mesh.axis_size(self.axis) if self.mesh is not None else 1
        lf = logits.float()
        m = lf.max(dim=-1, keepdim=True).values
        if tp > 1 and is_distributed():
            dist.all_reduce(m, op=dist.ReduceOp.MAX,
                            group=self.mesh.axis_group(self.axis))
        sumexp = (lf - m).exp().sum(dim=-1, keepdim=True)
        if tp > 1 and is_distributed():
            dist.all_reduce(sumexp, group=self.mesh.axis_group(self.axis))
        log_z = m + sumexp.log()
        kk = min(k, lf.shape[-1])
        vals, idx = lf.topk(kk, dim=-1)
        if tp > 1 and is_distributed():
            vocab_start = self.mesh.axis_index(self.axis) * \
                self.lm_head.out_per_rank
            pair = torch.cat([vals, (idx + vocab_start).float()],
                             dim=-1).contiguous()
            gathered = torch.empty(tp * pair.shape[0], pair.shape[1],
                                   dtype=pair.dtype, device=pair.device)
            dist.all_gather_into_tensor(
                gathered, pair, group=self.mesh.axis_group(self.axis))
            gathered = gathered.view(tp, -1, 2 * kk)
            vals = gathered[:, :, :kk].permute(1, 0, 2).reshape(-1, tp * kk)
            idx = gathered[:, :, kk:].permute(1, 0, 2).reshape(-1, tp * kk)
            vals, sel = vals.topk(min(k, vals.shape[-1]), dim=-1)
            idx = idx.gather(1, sel).long()
        return vals - log_z, idx

    @torch.no_grad()
    def beam_search(self, prompt_ids: torch.Tensor, max_new_tokens: int,
                    num_beams: int = 4, eos_token: Optional[int] = None
                    ) -> torch.Tensor:
        """Beam-search generation with KV-cache reorder between steps
        (reference: llm_serving wrapper's beam path reorders the
        DistributedArray cache via index-select executables,
        model/wrapper.py:115-182).  Returns [B, S0 + T] best sequences.
        """
        B, S0 = prompt_ids.shape
        K = num_beams
        dev = prompt_ids.device
        cache = self.new_cache(B * K)
        expanded = prompt_ids.repeat_interleave(K, dim=0)
        logits = self.forward_step(expanded, cache)       # [B*K, v/tp]
        lp, idx = self._log_probs_topk(logits, K)         # [B*K, K]
        # step 0: all beams are identical, pick from beam 0 only
        scores = lp.view(B, K, K)[:, 0]                   # [B, K]
        tok = idx.view(B, K, K)[:, 0]                     # [B, K]
        seqs = tok.unsqueeze(-1)                          # [B, K, 1]
        ended = (tok == eos_token) if eos_token is not None \
            else torch.zeros(B, K, dtype=torch.bool, device=dev)
        cur = tok.reshape(B * K, 1)
        neg_inf = torch.finfo(torch.float32).min
        for _ in range(max_new_tokens - 1):
            if bool(ended.all()):
                break
            logits = self.forward_step(cur, cache)
            lp, idx = self._log_probs_topk(logits, K)
            lp = lp.view(B, K, K).clone()
            idx = idx.view(B, K, K).clone()
            if eos_token is not None:
                # finished beams persist unchanged: one zero-cost eos
                # continuation, the rest impossible
                lp[ended] = neg_inf
                lp[:, :, 0][ended] = 0.0
                idx[:, :, 0][ended] = eos_token
            cand = scores.unsqueeze(-1) + lp              # [B, K, K]
            scores, top = cand.view(B, -1).topk(K, dim=-1)
            beam_src = top // K                           # [B, K]
            tok = idx.view(B, -1).gather(1, top)
            ended = ended.gather(1, beam_src) | \
                (tok == eos_token if eos_token is not None
                 else torch.zeros_like(tok, dtype=torch.bool))
            flat_src = (torch.arange(B, device=dev).unsqueeze(1) * K +
                        beam_src).reshape(-1)
            cache.reorder(flat_src)
            seqs = torch.cat([
                seqs.gather(1, beam_src.unsqueeze(-1)
                            .expand(-1, -1, seqs.shape[-1])),
                tok.unsqueeze(-1)], dim=-1)
            cur = tok.reshape(B * K, 1)
        best = scores.argmax(dim=1)
        out = seqs[torch.arange(B, device=dev), best]
        return torch.cat([prompt_ids, out], dim=1)

    @torch.no_grad()
    def generate(self, prompt_ids: torch.Tensor, max_new_tokens: int,
                 eos_token: Optional[int] = None,
                 do_sample: bool = False, temperature: float = 1.0,
                 top_k: int = 0, top_p: float = 1.0,
                 generator: Optional[torch.Generator] = None
                 ) -> torch.Tensor:
        """Greedy (default) or sampled generation: prefill + cached decode
        loop.  prompt_ids [B, S0]; returns [B, S0 + max_new_tokens].
        With do_sample, pass the SAME generator seed on every TP rank."""
        B, S0 = prompt_ids.shape
        cache = self.new_cache(B)

        def pick(lg):
            if do_sample:
                return self.sample_token(lg, temperature, top_k, top_p,
                                         generator)
            return self.greedy_token(lg)

        logits = self.forward_step(prompt_ids, cache)
        toks = [pick(logits)]
        for _ in range(max_new_tokens - 1):
            logits = self.forward_step(toks[-1].unsqueeze(1), cache)
            toks.append(pick(logits))
            if eos_token is not None and bool((toks[-1] == eos_token).all()):
                break
        return torch.cat([prompt_ids] + [t.unsqueeze(1) for t in toks],
                         dim=1)


