"""Testing library: pytree-recursive assert_allclose + model fixtures
(reference ``alpa/testing.py``: assert_allclose:28, MLPModel:54,
BertLayerModel:109, PipelineBasicTest:233)."""
from __future__ import annotations

from typing import Any

import torch
import torch.nn as nn

from .parallel.layers import ColumnParallelLinear, RowParallelLinear


def assert_allclose(a: Any, b: Any, rtol: float = 1e-4, atol: float = 1e-5,
                    path: str = ""):
    """Recursively compare nested dicts/lists/tuples of tensors/scalars
    (reference assert_allclose, testing.py:28)."""
    if isinstance(a, dict):
        assert isinstance(b, dict) and set(a) == set(b), path
        for k in a:
            assert_allclose(a[k], b[k], rtol, atol, f"{path}.{k}")
    elif isinstance(a, (list, tuple)):
        assert len(a) == len(b), path
        for i, (x, y) in enumerate(zip(a, b)):
            assert_allclose(x, y, rtol, atol, f"{path}[{i}]")
    elif torch.is_tensor(a) or torch.is_tensor(b):
        ta = a if torch.is_tensor(a) else torch.tensor(a)
        tb = b if torch.is_tensor(b) else torch.tensor(b)
        torch.testing.assert_close(ta, tb.to(ta.dtype), rtol=rtol, atol=atol,
                                   msg=lambda m: f"{path}: {m}")
    else:
        assert abs(float(a) - float(b)) <= atol + rtol * abs(float(b)), \
            (path, a, b)


class MLPModel(nn.Module):
    """The canonical test MLP (reference MLPModel, testing.py:54)."""

    def __init__(self, hidden: int = 256, num_layers: int = 4, mesh=None,
                 axis: int = 1, dtype=torch.float32, device=None):
        super().__init__()
        layers = []
        for i in range(num_layers):
            if i % 2 == 0:
                layers.append(ColumnParallelLinear(
                    hidden, 4 * hidden, mesh, axis, gelu=True, dtype=dtype,
                    device=device, init_seed=0, init_tag=f"mlp{i}"))
            else:
                layers.append(RowParallelLinear(
                    4 * hidden, hidden, mesh, axis, dtype=dtype,
                    device=device, init_seed=0, init_tag=f"mlp{i}"))
        self.layers = nn.ModuleList(layers)

    def forward(self, x):
        for l in self.layers:
            x = l(x)
        return x

    def loss(self, x, y):
        return ((self.forward(x) - y) ** 2).mean()


def get_mlp_train_state_and_step(method, hidden: int = 256, lr: float = 1e-3):
    """(state, step_fn) fixture (reference get_mlp_train_state_and_step,
    testing.py:72)."""
    import alpa_amd as aa

    def build(mesh=None, axis=1, dtype=torch.float32, device=None):
        return MLPModel(hidden=hidden, mesh=mesh, axis=axis, dtype=dtype,
                        device=device)

    state = aa.TrainState.create(build, method, lr=lr)
    step = aa.parallelize(lambda m, b: m.loss(*b), method=method)
    return state, step
