"""HF checkpoint loading for the serving zoo: a
random-init HuggingFace OPT/BLOOM converts into our TP-sharded serving
model and produces the SAME logits — turning the round-1 "throughput
claim" into a serving-capability claim.  No network: the HF models are
built in memory from configs."""
import pytest
import torch

from dist_utils import run_distributed

import alpa_amd as aa
from alpa_amd.serve.weights import load_bloom_hf, load_opt_hf

transformers = pytest.importorskip("transformers")


def _tiny_hf_opt():
    from transformers import OPTConfig, OPTForCausalLM
    torch.manual_seed(0)
    cfg = OPTConfig(hidden_size=64, num_hidden_layers=2,
                    num_attention_heads=4, ffn_dim=256, vocab_size=128,
                    max_position_embeddings=64, word_embed_proj_dim=64,
                    do_layer_norm_before=True, dropout=0.0,
                    attention_dropout=0.0, activation_function="relu")
    return OPTForCausalLM(cfg)


def _tiny_hf_bloom():
    from transformers import BloomConfig, BloomForCausalLM
    torch.manual_seed(1)
    cfg = BloomConfig(hidden_size=64, n_layer=2, n_head=8,
                      vocab_size=128, hidden_dropout=0.0,
                      attention_dropout=0.0)
    return BloomForCausalLM(cfg)


def test_opt_hf_logits_match():
    hf = _tiny_hf_opt().eval()
    model = load_opt_hf(hf)
    ids = torch.randint(0, 128, (2, 10),
                        generator=torch.Generator().manual_seed(3))
    with torch.no_grad():
        ref = hf(ids).logits[:, -1]
        cache = model.new_cache(2)
        got = model.forward_step(ids, cache)
    torch.testing.assert_close(got.float(), ref.float(), rtol=2e-3,
                               atol=2e-3)


def test_opt_hf_decode_cache_matches():
    """KV-cached decode logits track HF full-context logits step by step
    (argmax equality is meaningless on random-init near-flat logits, so
    the assertion is on the logits themselves)."""
    hf = _tiny_hf_opt().eval()
    model = load_opt_hf(hf)
    g = torch.Generator().manual_seed(4)
    ids = torch.randint(0, 128, (1, 8), generator=g)
    with torch.no_grad():
        cache = model.new_cache(1)
        got = model.forward_step(ids, cache)
        ctx = ids
        for step in range(4):
            ref = hf(ctx).logits[:, -1]
            torch.testing.assert_close(got.float(), ref.float(),
                                       rtol=5e-3, atol=5e-3)
            nxt = torch.randint(0, 128, (1, 1), generator=g)
            ctx = torch.cat([ctx, nxt], dim=1)
            got = model.forward_step(nxt, cache)


def test_bloom_hf_logits_match():
    hf = _tiny_hf_bloom().eval()
    model = load_bloom_hf(hf)
    ids = torch.randint(0, 128, (2, 10),
                        generator=torch.Generator().manual_seed(5))
    with torch.no_grad():
        ref = hf(ids).logits[:, -1]
        cache = model.new_cache(2)
        got = model.forward_step(ids, cache)
    torch.testing.assert_close(got.float(), ref.float(), rtol=2e-3,
                               atol=2e-3)


def _opt_tp_worker(rank, world_size, tmp):
    hf = _tiny_hf_opt().eval()
    mesh = aa.DeviceMesh(list(range(world_size)), (1, world_size))
    model = load_opt_hf(hf, mesh=mesh, axis=1)
    ids = torch.randint(0, 128, (2, 10),
                        generator=torch.Generator().manual_seed(3))
    with torch.no_grad():
        ref = hf(ids).logits[:, -1]
        cache = model.new_cache(2)
        got = model.forward_step(ids, cache)  # [B, vocab/tp]
    # vocab-sharded logits: this rank holds its slice
    v = 128 // world_size
    torch.testing.assert_close(got.float(), ref[:, rank * v:(rank + 1) * v],
                               rtol=2e-3, atol=2e-3)
    return True


def test_opt_hf_tp2_shards_match():
    run_distributed(_opt_tp_worker, world_size=2, args=("",), timeout=300)


def test_fp8_checkpoint_roundtrip(tmp_path):
    """save_fp8_checkpoint halves the linear-weight bytes; loading into
    a fresh model reproduces logits to within e4m3 quantization error,
    and the lm_head stays full-precision (bit-exact)."""
    import os
    import torch
    from alpa_amd.models.opt import OPTConfig, OPTModel
    from alpa_amd.serve.weights import (load_fp8_checkpoint,
                                        save_fp8_checkpoint)
    cfg = OPTConfig(hidden_size=128, num_layers=2, num_heads=4,
                    ffn_mult=4, vocab_size=512, max_seq_len=64)
    torch.manual_seed(0)
    m0 = OPTModel(cfg, None, 1, torch.float32, torch.device("cpu"),
                  init_seed=1)
    p = str(tmp_path / "w.fp8.pt")
    save_fp8_checkpoint(m0, p)
    m1 = OPTModel(cfg, None, 1, torch.float32, torch.device("cpu"),
                  init_seed=9)  # different init
    load_fp8_checkpoint(m1, p)
    torch.testing.assert_close(m1.lm_head.weight, m0.lm_head.weight,
                               rtol=0, atol=0)
    # weight-level: per-row e4m3 keeps each quantized weight within a
    # few percent of the original
    wrel = ((m1.blocks[0].qkv.weight - m0.blocks[0].qkv.weight).abs()
            .mean() / m0.blocks[0].qkv.weight.abs().mean())
    assert wrel < 0.04, wrel.item()
    # logit-level sanity (tiny random models amplify weight noise
    # through near-cancelling logits — loose bound)
    ids = torch.randint(0, 512, (2, 16))
    with torch.no_grad():
        l0 = m0.forward_step(ids, m0.new_cache(2))
        l1 = m1.forward_step(ids, m1.new_cache(2))
    rel = (l1 - l0).abs().mean() / l0.abs().mean()
    assert rel < 0.15, rel.item()
    # size: quantized file well under the fp32 state dict
    full = str(tmp_path / "w.full.pt")
    torch.save(m0.state_dict(), full)
    # tiny test model is embedding-heavy; real serving models
    # (linears dominate) land near 0.27x of fp32
    assert os.path.getsize(p) < 0.5 * os.path.getsize(full)


def test_fp8_checkpoint_rejects_wrong_tp(tmp_path):
    import pytest
    import torch
    from alpa_amd.models.opt import OPTConfig, OPTModel
    from alpa_amd.serve.weights import (load_fp8_checkpoint,
                                        save_fp8_checkpoint)
    cfg = OPTConfig(hidden_size=64, num_layers=1, num_heads=2,
                    ffn_mult=4, vocab_size=128, max_seq_len=32)
    m = OPTModel(cfg, None, 1, torch.float32, torch.device("cpu"),
                 init_seed=0)
    p = str(tmp_path / "w.pt")
    save_fp8_checkpoint(m, p)
    blob = torch.load(p, weights_only=False)
    blob["tp"] = 2
    torch.save(blob, p)
    with pytest.raises(AssertionError, match="tp"):
        load_fp8_checkpoint(m, p)
