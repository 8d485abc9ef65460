"""Checkpoint/inference weight quantization (ref runtime/weight_quantizer.py)."""
import torch

from deepspeed_amd.runtime.weight_quantizer import WeightQuantization


def test_quantize_data_roundtrip_error_bound():
    torch.manual_seed(0)
    wq = WeightQuantization()
    w = torch.randn(256, 128)
    q, s = wq.quantize_data(w, quantize_bits=8, groups=16)
    assert q.dtype == torch.int8 and s.shape == (16,)
    back = wq.dequantize_data(q, s, 16)
    # int8 symmetric: error <= scale/2 per group
    err = (back - w).abs().reshape(16, -1).amax(1)
    assert (err <= s / 2 + 1e-6).all()


def test_quantize_data_4bit_range():
    wq = WeightQuantization()
    q, s = wq.quantize_data(torch.randn(64, 64), quantize_bits=4, groups=4)
    assert q.max() <= 7 and q.min() >= -8


def test_sd_quantize_roundtrip():
    torch.manual_seed(1)
    sd = {"layer.mlp.weight": torch.randn(64, 32),
          "layer.attn.weight": torch.randn(64, 64),
          "layer.attn.bias": torch.randn(64),
          "step": 3}
    wq = WeightQuantization(mlp_extra_grouping=True)
    qsd = wq.sd_quantize(sd, quantize_groups=4)
    assert qsd["layer.mlp.weight"].dtype == torch.int8
    assert qsd["layer.attn.bias"].dtype == torch.float32  # untouched
    assert wq.scales["layer.mlp.weight"][1] == 8   # extra grouping
    assert wq.scales["layer.attn.weight"][1] == 4
    back = wq.sd_dequantize(qsd, dtype=torch.float32)
    assert (back["layer.mlp.weight"] -
            sd["layer.mlp.weight"]).abs().max() < 0.05
    assert back["step"] == 3


def test_model_quantize_swaps_linears():
    torch.manual_seed(2)
    m = torch.nn.Sequential(torch.nn.Linear(32, 64), torch.nn.GELU(),
                            torch.nn.Linear(64, 32))
    x = torch.randn(4, 32)
    ref = m(x)
    m, n = WeightQuantization().model_quantize(m, groups=8)
    from deepspeed_amd.linear.quantization import QuantizedLinear
    assert n == 2 and isinstance(m[0], QuantizedLinear)
    out = m(x.bfloat16()).float()
    assert (out - ref).abs().mean() < 0.1
