"""Captured-path auto-sharding on hardware: a plain bf16 model traces,
solves and trains on the GPU (tp degenerates to 1 on a single device —
this guards the capture + plan application machinery under ROCm/bf16,
complementing the ws=2 gloo parity tests)."""
import pytest
import torch
import torch.nn as nn
import torch.nn.functional as F

pytestmark = pytest.mark.gpu


def test_create_auto_trains_on_gpu():
    import alpa_amd as aa

    class Plain(nn.Module):
        def __init__(self):
            super().__init__()
            torch.manual_seed(0)
            self.emb = nn.Embedding(512, 256)
            self.fc1 = nn.Linear(256, 1024)
            self.fc2 = nn.Linear(1024, 256)
            self.head = nn.Linear(256, 512)

        def forward(self, ids):
            x = self.emb(ids)
            x = x + self.fc2(F.gelu(self.fc1(x)))
            return self.head(x)

    method = aa.ShardParallel()
    ids = torch.randint(0, 512, (4, 64))
    state = aa.TrainState.create_auto(lambda: Plain(), ids, method,
                                      lr=1e-3)
    step = aa.parallelize(
        lambda m, b: F.cross_entropy(
            m(b["ids"]).reshape(-1, 512).float(),
            b["labels"].reshape(-1)),
        method=method)
    dev_ids = ids.to("cuda")
    losses = [float(step(state, {"ids": dev_ids, "labels": dev_ids}))
              for _ in range(5)]
    assert all(l == l for l in losses), losses  # finite
    assert losses[-1] < losses[0], losses
    assert next(state.model.parameters()).dtype == torch.bfloat16
