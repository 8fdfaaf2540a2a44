"""MoE / expert-parallel tests: EP2 over gloo all-to-all vs serial oracle
(reference: tests around model/moe.py + suite_auto_moe; EP checklist row
SURVEY.md §2.2)."""
import pytest
import torch

from dist_utils import run_distributed

import alpa_amd as aa
from alpa_amd.models.moe import MoEConfig, MoEGPTModel
from alpa_amd.parallel.expert import ExpertParallelMLP, top2_gating

CFG = MoEConfig(hidden_size=64, num_layers=2, num_heads=4, seq_len=16,
                vocab_size=96, num_experts=4, moe_every=2,
                capacity_factor=8.0, aux_loss_weight=0.0)
BATCH = 8
STEPS = 2


def test_top2_gating_normalized():
    torch.manual_seed(0)
    logits = torch.randn(32, 8)
    w1, i1, w2, i2 = top2_gating(logits, 100)
    assert torch.all(i1 != i2)
    torch.testing.assert_close(w1 + w2, torch.ones(32))
    assert torch.all(w1 >= w2)


def test_expert_mlp_serial_forward():
    torch.manual_seed(1)
    m = ExpertParallelMLP(32, 64, 4, None, 0, capacity_factor=8.0,
                          init_seed=3)
    x = torch.randn(2, 8, 32)
    y = m(x)
    assert y.shape == x.shape
    assert m.last_aux_loss is not None
    y.sum().backward()
    assert m.w1.grad is not None and m.wg.grad is not None


def build_moe(mesh=None, axis=1, dtype=torch.float32, device=None):
    return MoEGPTModel(CFG, mesh, axis, dtype, device, init_seed=21)


def loss_fn(model, batch):
    return model.loss(batch["ids"], batch["labels"])


def make_batch(step):
    g = torch.Generator().manual_seed(700 + step)
    ids = torch.randint(0, CFG.vocab_size, (BATCH, CFG.seq_len), generator=g)
    labels = torch.randint(0, CFG.vocab_size, (BATCH, CFG.seq_len),
                           generator=g)
    return {"ids": ids, "labels": labels}


def run_serial():
    method = aa.ShardParallel(logical_mesh_shape=(1, 1))
    state = aa.TrainState.create(build_moe, method, lr=1e-3)
    step = aa.parallelize(loss_fn, method=method)
    losses = [float(step(state, make_batch(i))) for i in range(STEPS)]
    params = [p.detach().clone() for n, p in
              state.model.named_parameters() if "moe" not in n]
    return losses, params


def _ep_worker(rank, world_size):
    method = aa.ShardParallel(logical_mesh_shape=(world_size, 1))
    state = aa.TrainState.create(build_moe, method, lr=1e-3)
    step = aa.parallelize(loss_fn, method=method)
    per = BATCH // world_size
    idx = state.mesh.axis_index(0)
    losses = []
    for i in range(STEPS):
        b = make_batch(i)
        local = {k: v[idx * per:(idx + 1) * per] for k, v in b.items()}
        losses.append(float(step(state, local)))
    params = [p.detach().clone() for n, p in
              state.model.named_parameters() if "moe" not in n]
    return losses, params


def test_ep2_matches_serial():
    """dp=ep=2: experts sharded across ranks, tokens all-to-all routed.
    With no capacity drops the math equals serial exactly; non-expert
    params must stay identical to the serial run after updates."""
    serial_losses, serial_params = run_serial()
    results = run_distributed(_ep_worker, world_size=2, timeout=300)
    for i in range(STEPS):
        avg = sum(r[0][i] for r in results) / 2
        assert abs(avg - serial_losses[i]) < 3e-4, (results, serial_losses)
    # Params: expert-GEMM summation order differs between the serial
    # [E-batched] and EP layouts (~1e-5 rel on grads), and Adam's step-1
    # g/(|g|+eps) amplifies that for near-zero grads — tolerance reflects
    # fp non-associativity, not a routing error (grads were verified equal
    # to 1e-5 directly).
    for _, rank_params in results:
        for sp, rp in zip(serial_params, rank_params):
            torch.testing.assert_close(rp, sp, rtol=5e-2, atol=1e-3)
