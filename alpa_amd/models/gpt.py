"""GPT model family (decoder-only transformer), MI355X-first.

Capability analog of the reference's Flax GPT (``alpa/model/gpt_model.py:19``
builds a BERT-style decoder + LM head) and the benchmark spec ladder
(``benchmark/alpa/suite_manual_gpt.py:16-28``): 125M…76B at seq 1024,
vocab 51200.

Design: plain nn.Modules over the hand-written op library (fused LayerNorm,
flash attention, bias+GeLU epilogue, fused vocab cross-entropy).  Tensor
parallelism is built into the layers via a DeviceMesh axis — the
auto-sharding planner picks (dp, tp) and constructs the model accordingly.
Every matmul is hipBLASLt via torch.matmul; everything fusable is a gfx950
HIP kernel.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn

from .. import ops
from ..mesh import DeviceMesh
from ..parallel.layers import (ColumnParallelLinear, RowParallelLinear,
                               VocabParallelEmbedding, tag_seed,
                               vocab_parallel_cross_entropy)


@dataclass
class GPTConfig:
    hidden_size: int = 1024
    num_layers: int = 12
    num_heads: int = 16
    seq_len: int = 1024
    vocab_size: int = 51200
    ffn_mult: int = 4
    layernorm_eps: float = 1e-5
    tie_embeddings: bool = False
    #: rematerialize each block's activations in backward (reference
    #: automatic_remat, layer_construction.py:571 — remat at layer
    #: boundaries).  288 GB HBM3E rarely needs it; for very long seq or
    #: 70B-scale single-node runs it trades ~33% recompute for O(L) less
    #: activation memory
    remat: bool = False

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_heads

    def num_params(self) -> int:
        h, l, v = self.hidden_size, self.num_layers, self.vocab_size
        per_layer = 4 * h * h + 2 * self.ffn_mult * h * h
        return l * per_layer + v * h + self.seq_len * h + v * h


# The reference's GPT spec ladder (suite_manual_gpt.py:16-28):
# name -> (hidden, layers, heads)
GPT_SPECS = {
    "125M": (768, 12, 12),
    "350M": (1024, 24, 16),
    "760M": (1536, 24, 16),
    "1.3B": (2048, 24, 32),
    "2.6B": (2560, 32, 32),
    "6.7B": (4096, 32, 32),
    "15B": (5120, 48, 40),
    "39B": (8192, 48, 64),
    "76B": (10240, 60, 80),
}


def gpt_config(name: str, seq_len: int = 1024,
               vocab_size: int = 51200) -> GPTConfig:
    h, l, heads = GPT_SPECS[name]
    return GPTConfig(hidden_size=h, num_layers=l, num_heads=heads,
                     seq_len=seq_len, vocab_size=vocab_size)


def _wpe_init(cfg, dtype, device, init_seed: int):
    gen_device = device if (device is not None and
                            torch.device(device).type == "cuda") else "cpu"
    g = torch.Generator(device=gen_device)
    g.manual_seed(tag_seed(init_seed, "wpe"))
    w = torch.empty(cfg.seq_len, cfg.hidden_size, dtype=torch.float32,
                    device=gen_device).normal_(0, 0.02, generator=g)
    return w.to(dtype=dtype, device=device)


class Attention(nn.Module):
    """Multi-head attention: fused qkv column-split by heads over the tp
    axis, flash-attention kernel, output row-split + all-reduce."""

    def __init__(self, cfg: GPTConfig, mesh: Optional[DeviceMesh], axis: int,
                 dtype, device, layer_idx: int = 0, init_seed: int = 0):
        super().__init__()
        self.cfg = cfg
        tp = mesh.axis_size(axis) if mesh is not None else 1
        assert cfg.num_heads % tp == 0
        self.heads_per_rank = cfg.num_heads // tp
        self.head_dim = cfg.head_dim
        self.qkv = ColumnParallelLinear(cfg.hidden_size, 3 * cfg.hidden_size,
                                        mesh, axis, dtype=dtype, device=device,
                                        init_seed=init_seed,
                                        init_tag=f"b{layer_idx}.qkv")
        self.out = RowParallelLinear(cfg.hidden_size, cfg.hidden_size, mesh,
                                     axis, dtype=dtype, device=device,
                                     init_seed=init_seed,
                                     init_tag=f"b{layer_idx}.out")

    def forward(self, x):
        # packed per-head [q|k|v] projection; the strided attention kernel
        # consumes/produces these layouts directly (no permute copies)
        qkv = self.qkv(x)  # [B, S, 3*H/tp]
        o = ops.flash_attention_qkv(qkv, self.heads_per_rank, causal=True)
        return self.out(o)


class MLP(nn.Module):

    def __init__(self, cfg: GPTConfig, mesh: Optional[DeviceMesh], axis: int,
                 dtype, device, layer_idx: int = 0, init_seed: int = 0):
        super().__init__()
        ffn = cfg.ffn_mult * cfg.hidden_size
        self.fc1 = ColumnParallelLinear(cfg.hidden_size, ffn, mesh, axis,
                                        gelu=True, dtype=dtype, device=device,
                                        init_seed=init_seed,
                                        init_tag=f"b{layer_idx}.fc1")
        self.fc2 = RowParallelLinear(ffn, cfg.hidden_size, mesh, axis,
                                     dtype=dtype, device=device,
                                     init_seed=init_seed,
                                     init_tag=f"b{layer_idx}.fc2")

    def forward(self, x):
        return self.fc2(self.fc1(x))


_PLAIN = object()  # forward() sentinel: single-arg plain-path call


class LayerNorm(nn.Module):

    def __init__(self, hidden: int, eps: float, dtype, device):
        super().__init__()
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(hidden, dtype=dtype,
                                              device=device))
        self.bias = nn.Parameter(torch.zeros(hidden, dtype=dtype,
                                             device=device))

    def forward(self, x, delta=_PLAIN):
        """Single-arg: plain LayerNorm.  Two-arg (res, delta): fused
        residual-add + LN, returning only the normalized stream (the
        model epilogue).  Dispatching through __call__ keeps nn.Module
        hooks (ZeRO-3 gather/release) working."""
        if delta is _PLAIN:
            return ops.layer_norm(x, self.weight, self.bias, self.eps)
        _, y = ops.add_layer_norm(x, delta, self.weight, self.bias,
                                  self.eps)
        return y


class Block(nn.Module):

    def __init__(self, cfg: GPTConfig, mesh, axis, dtype, device,
                 layer_idx: int = 0, init_seed: int = 0):
        super().__init__()
        self.ln1 = LayerNorm(cfg.hidden_size, cfg.layernorm_eps, dtype, device)
        self.attn = Attention(cfg, mesh, axis, dtype, device, layer_idx,
                              init_seed)
        self.ln2 = LayerNorm(cfg.hidden_size, cfg.layernorm_eps, dtype, device)
        self.mlp = MLP(cfg, mesh, axis, dtype, device, layer_idx, init_seed)

    def forward(self, x, delta=_PLAIN):
        """Single-arg: plain block on a materialized stream.  Two-arg
        (res, delta): the fused residual-threaded path.  Dispatching
        through forward/__call__ (instead of calling forward_fused
        directly) keeps nn.Module hooks working — ZeRO-3's gather/release
        hooks depend on it."""
        if delta is _PLAIN:
            x = x + self.attn(self.ln1(x))
            x = x + self.mlp(self.ln2(x))
            return x
        return self.forward_fused(x, delta)

    def forward_fused(self, res, delta):
        """Residual-threaded form: both residual adds fuse into the
        LayerNorm kernels (ops.add_layer_norm).  Returns the next
        (residual, delta) pair; the stream value is res + delta."""
        h, y1 = ops.add_layer_norm(res, delta, self.ln1.weight,
                                   self.ln1.bias, self.ln1.eps)
        a = self.attn(y1)
        h2, y2 = ops.add_layer_norm(h, a, self.ln2.weight, self.ln2.bias,
                                    self.ln2.eps)
        return h2, self.mlp(y2)


def _run_block(blk, res, delta, remat: bool):
    """One transformer block, optionally under activation rematerialization
    (torch.utils.checkpoint, non-reentrant): the block's internals are
    recomputed during backward — the layer-boundary remat the reference
    inserts at pipeline-layer slices (remat_sliced_eqns)."""
    if remat and torch.is_grad_enabled() and res.requires_grad:
        from torch.utils.checkpoint import checkpoint
        return checkpoint(blk, res, delta, use_reentrant=False)
    return blk(res, delta)


class GPTModel(nn.Module):
    """Full GPT LM. `mesh`/`axis` give the tensor-parallel axis (axis 1 of a
    (dp, tp) mesh); dp replication is handled by the trainer."""

    def __init__(self, cfg: GPTConfig, mesh: Optional[DeviceMesh] = None,
                 axis: int = 1, dtype=torch.float32, device=None,
                 init_seed: int = 0):
        super().__init__()
        self.cfg = cfg
        self.mesh, self.axis = mesh, axis
        self.wte = VocabParallelEmbedding(cfg.vocab_size, cfg.hidden_size,
                                          mesh, axis, dtype=dtype,
                                          device=device, init_seed=init_seed,
                                          init_tag="wte")
        self.wpe = nn.Parameter(_wpe_init(cfg, dtype, device, init_seed))
        self.blocks = nn.ModuleList([
            Block(cfg, mesh, axis, dtype, device, i, init_seed)
            for i in range(cfg.num_layers)
        ])
        self.ln_f = LayerNorm(cfg.hidden_size, cfg.layernorm_eps, dtype,
                              device)
        # LM head: column-parallel over vocab; loss is computed shard-local
        # via vocab-parallel cross-entropy (no logits gather).
        self.lm_head = ColumnParallelLinear(cfg.hidden_size, cfg.vocab_size,
                                            mesh, axis, bias=False,
                                            dtype=dtype, device=device,
                                            init_seed=init_seed,
                                            init_tag="lm_head")
        self.lm_head._fp8_exclude = True  # logits GEMM stays bf16
        if cfg.tie_embeddings:
            self.lm_head.weight = self.wte.weight

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        """ids [B, S] -> logits [B, S, V/tp] (vocab-sharded)."""
        B, S = ids.shape
        res = self.wte(ids) + self.wpe[:S]
        delta = None
        for blk in self.blocks:
            res, delta = _run_block(blk, res, delta, self.cfg.remat)
        x = self.ln_f(res, delta)
        return self.lm_head(x)

    def loss(self, ids: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
        """Mean per-token LM loss. labels [B, S]."""
        logits = self.forward(ids)
        N = logits.shape[0] * logits.shape[1]
        logits = logits.reshape(N, -1)
        vocab_start = self.lm_head.mesh.axis_index(self.axis) * \
            self.lm_head.out_per_rank if self.mesh is not None else 0
        per_tok = vocab_parallel_cross_entropy(
            logits, labels.reshape(N), self.mesh, self.axis, vocab_start)
        return per_tok.mean()


# --------------------------------------------------------------------------
# Pipeline-parallel stage (inter-op parallelism; reference slices the jaxpr
# at pipeline markers — here the model family builds its stage directly)
# --------------------------------------------------------------------------


class GPTStage(nn.Module):
    """One pipeline stage of the GPT LM: [layer_range) blocks, plus the
    embedding on the first stage and ln_f + LM head + loss on the last.

    forward(x, microbatch): x = activations [B, S, H] from the previous
    stage (None on the first stage); returns activations, or the scalar
    microbatch loss on the last stage.
    """

    def __init__(self, cfg: GPTConfig, layer_range, is_first: bool,
                 is_last: bool, mesh: Optional[DeviceMesh] = None,
                 axis: int = 1, dtype=torch.float32, device=None,
                 init_seed: int = 0):
        super().__init__()
        self.cfg = cfg
        self.mesh, self.axis = mesh, axis
        self.is_first, self.is_last = is_first, is_last
        self.layer_range = tuple(layer_range)
        if is_first:
            self.wte = VocabParallelEmbedding(cfg.vocab_size, cfg.hidden_size,
                                              mesh, axis, dtype=dtype,
                                              device=device,
                                              init_seed=init_seed,
                                              init_tag="wte")
            self.wpe = nn.Parameter(_wpe_init(cfg, dtype, device, init_seed))
        # blocks registered under their GLOBAL layer indices so stage
        # checkpoints interoperate with the serial layout (restore under a
        # different parallelization — the reference's headline
        # checkpoint feature)
        self.blocks = nn.Module()
        for i in range(layer_range[0], layer_range[1]):
            self.blocks.add_module(str(i), Block(cfg, mesh, axis, dtype,
                                                 device, i, init_seed))
        if is_last:
            self.ln_f = LayerNorm(cfg.hidden_size, cfg.layernorm_eps, dtype,
                                  device)
            # tied embeddings: the LM head draws the SAME init as wte
            # (tag + std); the cross-stage grad all-reduce keeps the two
            # copies synchronized (reference __builtin$CrossMeshAllReduce,
            # SURVEY.md §2.3 N15)
            self.lm_head = ColumnParallelLinear(
                cfg.hidden_size, cfg.vocab_size, mesh, axis, bias=False,
                dtype=dtype, device=device, init_seed=init_seed,
                init_tag="wte" if cfg.tie_embeddings else "lm_head",
                init_std=0.02 if cfg.tie_embeddings else None)
            self.lm_head._fp8_exclude = True  # logits GEMM stays bf16

    def forward(self, x, microbatch):
        ids, labels = microbatch["ids"], microbatch.get("labels")
        if self.is_first:
            S = ids.shape[1]
            x = self.wte(ids) + self.wpe[:S]
        res, delta = x, None
        for blk in self.blocks.children():
            res, delta = _run_block(blk, res, delta, self.cfg.remat)
        if not self.is_last:
            # materialize the stream value at the stage boundary
            return res + delta if delta is not None else res
        x = self.ln_f(res, delta)
        logits = self.lm_head(x)
        N = logits.shape[0] * logits.shape[1]
        logits = logits.reshape(N, -1)
        vocab_start = self.lm_head.mesh.axis_index(self.axis) * \
            self.lm_head.out_per_rank if self.mesh is not None else 0
        per_tok = vocab_parallel_cross_entropy(
            logits, labels.reshape(N), self.mesh, self.axis, vocab_start)
        return per_tok.mean()


def gpt_pipeline_spec(cfg: GPTConfig, microbatch_tokens: int = 0):
    """PipelineModelSpec for the GPT family (used by PipeshardParallel).

    Pass ``microbatch_tokens`` (= microbatch size x seq_len) to enable the
    profile-guided auto stage search (`stage_option="auto"` with
    num_stages=None): it fills per-layer flops / boundary bytes / param
    bytes for stage_construction.profiled_stage_search."""
    from ..pipeline_parallel.spec import PipelineModelSpec

    def build_stage(layer_range, is_first, is_last, mesh, axis, dtype,
                    device):
        return GPTStage(cfg, layer_range, is_first, is_last, mesh, axis,
                        dtype, device)

    def act_shape(microbatch):
        ids = microbatch["ids"]
        return (ids.shape[0], ids.shape[1], cfg.hidden_size)

    # per-block cost uniform; embedding/LM head are pinned to the
    # first/last stage by construction
    spec = PipelineModelSpec(num_layers=cfg.num_layers,
                             build_stage=build_stage, act_shape=act_shape)
    if cfg.tie_embeddings:
        # stage -1 = last; resolved by the pipeline compiler
        spec.tied_groups = [{0: "wte.weight", -1: "lm_head.weight"}]
    if microbatch_tokens:
        H, S = cfg.hidden_size, cfg.seq_len
        T = microbatch_tokens
        # fwd+bwd GEMM flops per block (attention QKV/proj + 2 FFN mats +
        # the S^2 score/context GEMMs), factor 3 for fwd + dX + dW
        per_layer = 3.0 * 2.0 * T * (12 * H * H + 2 * S * H)
        spec.layer_flops = [per_layer] * cfg.num_layers
        spec.boundary_act_bytes = 2.0 * T * H
        spec.layer_param_bytes = [2.0 * 12 * H * H] * cfg.num_layers
        # MEASURED stage costs (tools/profile_stages.py) anchor the
        # training DP when the profiling DB carries this curve
        spec.stage_cost_curve = f"gpt_stage_cost_h{H}"
        spec.microbatch_tokens = float(T)
    return spec
