"""mark_pipeline_boundary / spec_from_module / remat API (VERDICT r1
item 8; reference primitive_def.py:18 mark_pipeline_boundary,
layer_construction.py:542,571 manual_remat/automatic_remat): a PLAIN
torch model with boundary markers in its forward becomes a pipelined
training job, matching serial."""
import torch
import torch.nn as nn
import torch.nn.functional as F

from dist_utils import run_distributed

import alpa_amd as aa

H = 32


class MarkedNet(nn.Module):
    """Plain model, not in the zoo; boundaries marked in forward."""

    def __init__(self):
        super().__init__()
        torch.manual_seed(6)
        self.a = nn.Linear(H, 2 * H)
        self.b = nn.Linear(2 * H, 2 * H)
        self.c = nn.Linear(2 * H, H)

    def forward(self, x):
        x = torch.relu(self.a(x))
        x = aa.mark_pipeline_boundary(x)
        x = torch.relu(self.b(x))
        x = aa.mark_pipeline_boundary(x)
        return self.c(x)


def _mb(i, b=4):
    g = torch.Generator().manual_seed(70 + i)
    return {"x": torch.randn(b, H, generator=g),
            "y": torch.randn(b, H, generator=g)}


def _loss(out, mb):
    return F.mse_loss(out.float(), mb["y"].float())


def test_split_matches_serial_forward():
    spec = aa.spec_from_module(lambda: MarkedNet(), _mb(0), _loss)
    assert spec.num_layers == 3
    s0 = spec.build_stage((0, 1), True, False, None, 1, torch.float32, None)
    s1 = spec.build_stage((1, 3), False, True, None, 1, torch.float32, None)
    mb = _mb(1)
    loss = s1(s0(None, mb), mb)
    ref = _loss(MarkedNet()(mb["x"]), mb)
    assert abs(float(loss) - float(ref)) < 1e-6


def _serial(steps):
    net = MarkedNet()
    opt = aa.AdamW(net.parameters(), lr=1e-3, weight_decay=0.0)
    out = []
    for i in range(steps):
        mb = _mb(i)
        loss = _loss(net(mb["x"]), mb)
        opt.zero_grad()
        loss.backward()
        opt.step()
        out.append(float(loss.detach()))
    return out


def _pipeline_worker(rank, world_size):
    spec = aa.spec_from_module(lambda: MarkedNet(), _mb(0), _loss)
    method = aa.PipeshardParallel(num_micro_batches=2,
                                  num_stages=world_size,
                                  stage_mesh_shape=(1, 1))
    state = aa.TrainState.create(spec, method, lr=1e-3, weight_decay=0.0)
    step = aa.parallelize(lambda m, b: None, method=method)
    return [float(step(state, _mb(i, b=4))) for i in range(3)]


def test_marked_model_pipelines_to_serial_parity():
    """The clustering DP assigns the 3 marked segments to 2 stages; the
    1F1B engine trains to the serial trajectory."""
    serial = _serial(3)
    results = run_distributed(_pipeline_worker, world_size=2, timeout=300)
    for r in results:
        for a, b in zip(r, serial):
            assert abs(a - b) < 3e-4, (r, serial)


def test_manual_remat_parity():
    torch.manual_seed(1)
    inner = nn.Sequential(nn.Linear(H, H), nn.GELU(), nn.Linear(H, H))
    import copy
    ref = copy.deepcopy(inner)
    rm = aa.manual_remat(inner)
    x = torch.randn(4, H, requires_grad=True)
    xr = x.detach().clone().requires_grad_(True)
    rm(x).sum().backward()
    ref(xr).sum().backward()
    torch.testing.assert_close(x.grad, xr.grad)
    for p, q in zip(rm.inner.parameters(), ref.parameters()):
        torch.testing.assert_close(p.grad, q.grad)


def test_automatic_remat_clusters_blocks():
    class Tower(nn.Module):
        # self-contained blocks: forward iterates `x = b(x)` only (the
        # automatic_remat grouping contract — per-block extra ops in
        # forward would change under grouping)
        def __init__(self):
            super().__init__()
            torch.manual_seed(2)
            self.blocks = nn.ModuleList(
                [nn.Sequential(nn.Linear(H, H), nn.ReLU())
                 for _ in range(6)])

        def forward(self, x):
            for b in self.blocks:
                x = b(x)
            return x

    import copy
    t = Tower()
    ref = copy.deepcopy(t)
    t = aa.automatic_remat(t, num_layers=3)
    from alpa_amd.pipeline_parallel.boundary import _Remat
    assert len(t.blocks) == 3
    assert all(isinstance(b, _Remat) for b in t.blocks)
    x = torch.randn(4, H, requires_grad=True)
    xr = x.detach().clone().requires_grad_(True)
    t(x).sum().backward()
    ref(xr).sum().backward()
    torch.testing.assert_close(x.grad, xr.grad)
