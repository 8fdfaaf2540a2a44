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
