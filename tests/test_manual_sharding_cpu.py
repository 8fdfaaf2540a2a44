"""Manual (pjit-style) sharding of PLAIN torch models (reference
shard_parallel/manual_sharding.py): pattern specs convert nn.Linear /
nn.Embedding into the TP layers; sharded forward == the original model.
"""
import torch
import torch.nn as nn

from dist_utils import run_distributed

import alpa_amd as aa
from alpa_amd.shard_parallel.manual_sharding import apply_manual_sharding


class PlainLM(nn.Module):
    """Ordinary torch model with no alpa_amd layers."""

    def __init__(self):
        super().__init__()
        torch.manual_seed(33)
        self.embed = nn.Embedding(64, 32)
        self.blocks = nn.ModuleList([
            nn.ModuleDict({"fc1": nn.Linear(32, 128),
                           "fc2": nn.Linear(128, 32)})
            for _ in range(2)
        ])

    def forward(self, ids):
        x = self.embed(ids)
        for blk in self.blocks:
            x = x + blk["fc2"](torch.relu(blk["fc1"](x)))
        return x


SPECS = {"blocks.*.fc1": "column", "blocks.*.fc2": "row",
         "embed": "vocab"}


def _tp_worker(rank, world_size):
    mesh = aa.full_mesh((1, world_size))
    m = apply_manual_sharding(PlainLM(), SPECS, mesh, axis=1)
    torch.manual_seed(7)
    ids = torch.randint(0, 64, (2, 8))
    y = m(ids)
    y.square().mean().backward()
    fc1 = m.blocks[0]["fc1"]
    assert fc1.weight.shape[0] == 128 // world_size  # actually sharded
    return y.detach()


def test_manual_sharding_tp2_matches_plain():
    plain = PlainLM()
    torch.manual_seed(7)
    ids = torch.randint(0, 64, (2, 8))
    want = plain(ids)
    for r in run_distributed(_tp_worker, world_size=2, timeout=300):
        torch.testing.assert_close(torch.as_tensor(r), want.detach(),
                                   rtol=1e-5, atol=1e-5)


def test_manual_sharding_serial_noop():
    plain = PlainLM()
    torch.manual_seed(7)
    ids = torch.randint(0, 64, (2, 8))
    want = plain(ids)
    m = apply_manual_sharding(PlainLM(), SPECS, None)
    torch.testing.assert_close(m(ids), want)


class _FakeMesh:
    """Just enough mesh surface to exercise the spec-matching check."""
    is_member = True

    def axis_size(self, axis):
        return 2

    def axis_index(self, axis):
        return 0


def test_manual_sharding_unmatched_spec_raises():
    import pytest
    with pytest.raises(AssertionError, match="matched no module"):
        apply_manual_sharding(PlainLM(), {"nope.*": "column"},
                              _FakeMesh())
