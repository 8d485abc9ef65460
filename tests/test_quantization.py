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


# ------------------------------------------------------- FP_Quantize family
def _fp_roundtrip_cpu(q_bits, tol):
    import torch
    from deepspeed_amd.ops.fp_quantizer import FP_Quantize
    torch.manual_seed(0)
    x = torch.randn(1000) * 3
    fpq = FP_Quantize(group_size=128, q_bits=q_bits)
    q = fpq.quantize(x)
    y = fpq.dequantize(q).float()
    rel = (x - y).abs().max() / x.abs().max()
    assert rel < tol, f"q{q_bits}: rel err {rel}"


def test_fp4_roundtrip_cpu():
    _fp_roundtrip_cpu(4, 0.20)


def test_fp6_roundtrip_cpu():
    _fp_roundtrip_cpu(6, 0.08)


def test_fp12_roundtrip_cpu():
    _fp_roundtrip_cpu(12, 0.005)


def test_fp_quantize_exact_values_cpu():
    """Codec spot checks: e2m1 grid is {0,.5,1,1.5,2,3,4,6} x sign."""
    import torch
    from deepspeed_amd.ops.fp_quantizer import _decode_torch, _encode_torch
    vals = torch.tensor([0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0,
                         -1.5, -6.0])
    bits = _encode_torch(vals, 2, 1)
    back = _decode_torch(bits, 2, 1)
    assert torch.equal(back, vals)
    # rounding: 2.4 -> 2, 2.6 -> 3, 7 -> saturate 6
    r = _decode_torch(_encode_torch(torch.tensor([2.4, 2.6, 7.0]), 2, 1),
                      2, 1)
    assert torch.equal(r, torch.tensor([2.0, 3.0, 6.0]))


@pytest.mark.gpu
@pytest.mark.parametrize("q_bits", [4, 6, 12])
def test_fp_quantize_gpu_matches_cpu_codec(q_bits):
    import torch
    from deepspeed_amd.ops.fp_quantizer import FP_Quantize
    torch.manual_seed(0)
    x = torch.randn(4096, dtype=torch.bfloat16) * 2
    cpu = FP_Quantize(group_size=256, q_bits=q_bits)
    yc = cpu.dequantize(cpu.quantize(x.float()))
    gpu = FP_Quantize(group_size=256, q_bits=q_bits)
    yg = gpu.dequantize(gpu.quantize(x.cuda())).cpu()
    # GPU path returns bf16; compare after identical final rounding
    assert torch.allclose(yc.to(torch.bfloat16).float(), yg.float(),
                          atol=1e-5), \
        (yc.to(torch.bfloat16).float() - yg.float()).abs().max()


def test_int4_roundtrip_cpu():
    from deepspeed_amd.ops.quantizer import dequantize_int4, quantize_int4
    torch.manual_seed(0)
    x = (torch.randn(33, 64) * 2).bfloat16()
    q, scales, shape = quantize_int4(x.float(), group_size=128)
    y = dequantize_int4(q, scales, shape, group_size=128)
    rel = (x.float() - y.float()).abs().max() / x.float().abs().max()
    assert y.shape == x.shape
    assert rel < 0.12, rel


@pytest.mark.gpu
def test_int4_gpu_matches_cpu():
    from deepspeed_amd.ops.quantizer import dequantize_int4, quantize_int4
    torch.manual_seed(0)
    x = (torch.randn(4096) * 3).bfloat16()
    qc, sc, shp = quantize_int4(x.float(), group_size=256)
    yc = dequantize_int4(qc, sc, shp, group_size=256)
    qg, sg, _ = quantize_int4(x.cuda(), group_size=256)
    yg = dequantize_int4(qg, sg, shp, group_size=256).cpu()
    assert torch.allclose(yc.float(), yg.float(), atol=1e-5), \
        (yc.float() - yg.float()).abs().max()


@pytest.mark.parametrize("bits", [4, 6, 12])
def test_fp_code_bit_packing(bits):
    import torch
    from deepspeed_amd.ops.fp_quantizer import pack_codes, unpack_codes
    torch.manual_seed(0)
    n = 1001
    codes = torch.randint(0, 1 << bits, (n,), dtype=torch.uint16)
    packed = pack_codes(codes, bits)
    # dense: ceil(n*bits/8) bytes (up to one group of padding)
    assert packed.numel() <= (n * bits + 7) // 8 + 3
    back = unpack_codes(packed, bits, n)
    assert torch.equal(back, codes)


def test_fp6_roundtrip_packed():
    import torch
    from deepspeed_amd.ops.fp_quantizer import (FP_Quantize, pack_codes,
                                                unpack_codes)
    torch.manual_seed(0)
    x = torch.randn(777) * 2
    fpq = FP_Quantize(group_size=128, q_bits=6)
    q = fpq.quantize(x)
    packed = pack_codes(q, 6)
    assert packed.numel() < q.numel()  # denser than u16 container
    q2 = unpack_codes(packed, 6, q.numel())
    y = fpq.dequantize(q2)
    y_ref = fpq.dequantize(q)
    assert torch.equal(y, y_ref)


def test_moq_quantize_training_config():
    """quantize_training config: weights fake-quantize after steps once
    the annealing schedule reaches target bits (ref engine.py:2257)."""
    import torch
    import torch.distributed as tdist
    import deepspeed_amd
    from deepspeed_amd.comm import groups
    groups.reset_groups()
    if not tdist.is_initialized():
        tdist.init_process_group("gloo",
                                 init_method="tcp://127.0.0.1:29641",
                                 rank=0, world_size=1)
    torch.manual_seed(0)
    model = torch.nn.Sequential(torch.nn.Linear(32, 32), torch.nn.Tanh(),
                                torch.nn.Linear(32, 8))
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config={
        "train_micro_batch_size_per_gpu": 4,
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
        "bf16": {"enabled": True},
        "zero_optimization": {"stage": 2},
        "quantize_training": {
            "enabled": True,
            "quantize_bits": {"start_bits": 16, "target_bits": 8},
            "schedule": {"quantize_period": 1},
        }})
    assert engine.quantizer is not None
    x = torch.randn(4, 32).bfloat16()
    y = torch.randn(4, 8).bfloat16()
    for _ in range(4):  # period 1: bits 16 -> 8 after first steps
        loss = (engine(x) - y).float().pow(2).mean()
        engine.backward(loss)
        engine.step()
    assert engine.quantizer.bits[0] == 8
    # int8 fake-quant leaves at most 257 distinct levels per bucket
    for b in engine.optimizer.buckets:
        assert b.flat16.unique().numel() <= 257, b.flat16.unique().numel()
    assert torch.isfinite(torch.tensor(loss.item()))


def test_moq_eigenvalue_modulated_annealing():
    """Layers with larger Hessian eigenvalues stretch their period and
    anneal LATER (MoQ scheduling rule)."""
    from deepspeed_amd.runtime.quantize import Quantizer
    q = Quantizer(layer_num=2, start_bits=16, target_bits=8,
                  quantize_period=4, q_eigenvalue=True)
    eig = {0: 0.1, 1: 1.0}  # layer 1 is 10x more sensitive
    import torch
    params = [[torch.randn(8, 8)], [torch.randn(8, 8)]]
    for _ in range(7):
        q.quantize(params, eigenvalue_enabled=True, block_eigenvalue=eig)
    # after 7 steps: layer 0 (period ~4.4) has halved once; layer 1
    # (period 8) has not
    assert q.bits[0] < 16 and q.bits[1] == 16, q.bits
    for _ in range(10):
        q.quantize(params, eigenvalue_enabled=True, block_eigenvalue=eig)
    assert q.bits == [8, 8], q.bits  # both reach target eventually
    assert not q.any_precision_switch()
