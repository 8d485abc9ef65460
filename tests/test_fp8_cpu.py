"""fp8 Linear plumbing that runs without a GPU (fallback paths)."""
import pytest
import torch

from deepspeed_amd.ops.fp8_linear import (E4M3_MAX, Fp8Linear, _fp8_ok,
                                          _quant, _t8, bump_fp8_version)


def test_quant_roundtrip_error_bound():
    torch.manual_seed(0)
    if not hasattr(torch, "float8_e4m3fn"):
        pytest.skip("no fp8 dtypes in this torch")
    x = torch.randn(64, 32, dtype=torch.bfloat16) * 5
    q, scale = _quant(x, torch.float8_e4m3fn, E4M3_MAX)
    rec = q.float() * scale
    err = (rec - x.float()).abs().max()
    assert err <= x.abs().max() * 2 ** -3 + 1e-3


def test_t8_transpose():
    if not hasattr(torch, "float8_e4m3fn"):
        pytest.skip("no fp8 dtypes")
    x = torch.randn(8, 16, dtype=torch.bfloat16)
    q, _ = _quant(x, torch.float8_e4m3fn, E4M3_MAX)
    qt = _t8(q)
    assert qt.shape == (16, 8)
    assert torch.equal(qt.view(torch.int8),
                       q.view(torch.int8).t().contiguous())


def test_fp8_linear_falls_back_on_cpu():
    lin = torch.nn.Linear(32, 16)
    lin.__class__ = Fp8Linear
    x = torch.randn(4, 32)
    assert not _fp8_ok(x, lin.weight)  # cpu => ineligible
    y = lin(x)  # plain bf16/fp32 path
    assert y.shape == (4, 16)
    bump_fp8_version()  # version bump is harmless without a cache
