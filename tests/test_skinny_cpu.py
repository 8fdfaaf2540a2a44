"""CPU-checkable properties of the skinny decode GEMV host logic."""
import pytest
import torch

from alpa_amd.ops import _skinny_splits, skinny_ok


@pytest.mark.parametrize("N,K", [(5120, 5120), (15360, 5120),
                                 (20480, 5120), (5120, 20480),
                                 (27648, 9216), (9216, 36864),
                                 (640, 512), (1280, 1024)])
@pytest.mark.parametrize("M", [1, 4, 8])
def test_splits_legal(N, K, M):
    s = _skinny_splits(N, K, M)
    assert s is not None
    rounds = K // 8 // s
    MT = 4
    while MT < M:
        MT *= 2
    assert (K // 8) % s == 0          # launcher divisibility
    assert rounds % 8 == 0            # kernel's 8-deep pipeline
    assert rounds * 16 * MT <= 65536  # LDS x-slice fits 64 KB


def test_gate_is_inference_only_and_shape_gated():
    from alpa_amd.global_env import global_config
    w = torch.randn(5120, 5120)
    x = torch.randn(4, 1, 5120)

    class M(torch.nn.Module):
        pass
    m = M()
    # CPU tensors never take the HIP path
    with torch.no_grad():
        assert not skinny_ok(x, w, m)
    # the envelope check itself (device-independent part): big square
    # bf16 shapes are excluded in auto mode unless fp8-packed
    old = global_config.fp8_gemm
    global_config.fp8_gemm = False
    try:
        wbig = torch.randn(9216, 9216)
        with torch.no_grad():
            assert not skinny_ok(torch.randn(4, 1, 9216), wbig, m)
    finally:
        global_config.fp8_gemm = old
