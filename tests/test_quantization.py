"""Quantizer + OptimizedLinear/LoRA (CPU fallbacks; GPU kernels in
test_ops_gpu-style gpu test below)."""
import pytest
import torch

from deepspeed_amd.ops import quantizer as Q


def test_int8_roundtrip_cpu():
    torch.manual_seed(0)
    x = torch.randn(4096, dtype=torch.bfloat16)
    q, s = Q.quantize_int8(x, 512)
    back = Q.dequantize_int8(q, s, 512)
    err = (back.float() - x.float()).abs().max()
    assert err < x.float().abs().max() / 50, err


def test_fp8_roundtrip_cpu():
    torch.manual_seed(0)
    x = torch.randn(4096, dtype=torch.bfloat16)
    q, s = Q.quantize_fp8(x, 512)
    back = Q.dequantize_fp8(q, s, 512)
    rel = ((back.float() - x.float()).abs() /
           (x.float().abs() + 1e-3)).median()
    assert rel < 0.1, rel


def test_optimized_linear_lora():
    from deepspeed_amd.linear import LoRAConfig, OptimizedLinear
    torch.manual_seed(0)
    lin = OptimizedLinear(32, 64, lora_config=LoRAConfig(lora_r=8),
                          dtype=torch.float32)
    x = torch.randn(4, 32)
    out = lin(x)
    assert out.shape == (4, 64)
    # LoRA B initialized to zero: output == base at init
    base = torch.nn.functional.linear(x, lin.full_weight())
    assert torch.allclose(out, base, atol=1e-6)
    # only lora params train
    out.sum().backward()
    assert lin.lora_weight_1.weight.grad is not None
    assert lin.base_weight.grad is None


def test_quantized_linear():
    from deepspeed_amd.linear import QuantizedLinear
    torch.manual_seed(0)
    ref = torch.nn.Linear(64, 32, dtype=torch.bfloat16)
    qlin = QuantizedLinear(ref, quantization="fp8", group_size=64)
    x = torch.randn(4, 64, dtype=torch.bfloat16)
    out = qlin(x)
    expected = ref(x)
    rel = (out.float() - expected.float()).abs().mean() / \
        expected.float().abs().mean()
    assert rel < 0.1, rel


@pytest.mark.gpu
def test_quantizers_gpu():
    torch.manual_seed(0)
    x = torch.randn(1 << 20, dtype=torch.bfloat16, device="cuda")
    for quant, dequant, tol in ((Q.quantize_int8, Q.dequantize_int8, 0.02),
                                (Q.quantize_fp8, Q.dequantize_fp8, 0.12)):
        q, s = quant(x, 2048)
        back = dequant(q, s, 2048)
        rel = ((back.float() - x.float()).abs() /
               (x.float().abs() + 1e-3)).median().item()
        assert rel < tol, (quant.__name__, rel)
