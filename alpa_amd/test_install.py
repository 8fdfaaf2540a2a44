"""Install smoke test — the canonical "is the stack alive" check
(reference ``alpa/test_install.py``: ShardParallel and PipeshardParallel
MLP vs serial).  Run: python -m alpa_amd.test_install  (or via pytest)."""
from __future__ import annotations

import torch

import alpa_amd as aa
from alpa_amd.testing import MLPModel, assert_allclose, \
    get_mlp_train_state_and_step


def make_batch(hidden=128, batch=4, seed=0):
    g = torch.Generator().manual_seed(seed)
    return (torch.randn(batch, hidden, generator=g),
            torch.randn(batch, hidden, generator=g))


def test_shard_parallel():
    """ShardParallel runs and the loss decreases (any world size:
    serial `python -m alpa_amd.test_install` or under torchrun)."""
    aa.init()
    method = aa.ShardParallel(num_micro_batches=2,
                              logical_mesh_shape=(aa.world_size(), 1))
    state, step = get_mlp_train_state_and_step(method, hidden=128)
    batch = make_batch()
    first = float(step(state, batch))
    for _ in range(5):
        last = float(step(state, batch))
    assert last < first, (first, last)


def test_grad_accumulation_consistency():
    """nmb=1 vs nmb=4 give identical updates (the grad-acc rewrite)."""
    aa.init()
    outs = []
    for nmb in (1, 4):
        method = aa.ShardParallel(num_micro_batches=nmb,
                                  logical_mesh_shape=(aa.world_size(), 1))
        state, step = get_mlp_train_state_and_step(method, hidden=128)
        step(state, make_batch())
        outs.append([p.detach().clone()
                     for p in state.model.parameters()])
    assert_allclose(outs[0], outs[1])


if __name__ == "__main__":
    test_shard_parallel()
    test_grad_accumulation_consistency()
    print("alpa_amd install OK")
