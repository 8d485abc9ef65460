"""fp8 Linear (hipBLASLt tensorwise _scaled_mm) numerics on hardware."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_fp8_linear_forward_close_to_bf16():
    from deepspeed_amd.ops.fp8_linear import Fp8Linear, _fp8_ok
    torch.manual_seed(0)
    lin = torch.nn.Linear(256, 512, bias=True, device="cuda",
                          dtype=torch.bfloat16)
    x = torch.randn(64, 256, device="cuda", dtype=torch.bfloat16)
    assert _fp8_ok(x, lin.weight)
    ref = lin(x)
    lin.__class__ = Fp8Linear
    got = lin(x)
    rel = (got.float() - ref.float()).abs().max() / \
        (ref.float().abs().max() + 1e-6)
    assert rel < 0.12, f"fp8 forward rel err {rel.item()}"


def test_fp8_linear_grads_flow_and_match():
    from deepspeed_amd.ops.fp8_linear import Fp8Linear
    torch.manual_seed(1)
    lin_ref = torch.nn.Linear(128, 64, bias=False, device="cuda",
                              dtype=torch.bfloat16)
    lin8 = torch.nn.Linear(128, 64, bias=False, device="cuda",
                           dtype=torch.bfloat16)
    with torch.no_grad():
        lin8.weight.copy_(lin_ref.weight)
    lin8.__class__ = Fp8Linear
    x1 = torch.randn(32, 128, device="cuda", dtype=torch.bfloat16,
                     requires_grad=True)
    x2 = x1.detach().clone().requires_grad_(True)
    g = torch.randn(32, 64, device="cuda", dtype=torch.bfloat16)
    lin_ref(x1).backward(g)
    lin8(x2).backward(g)
    for got, want, name in ((x2.grad, x1.grad, "dx"),
                            (lin8.weight.grad, lin_ref.weight.grad, "dw")):
        rel = (got.float() - want.float()).abs().max() / \
            (want.float().abs().max() + 1e-6)
        assert rel < 0.15, f"{name} rel err {rel.item()}"


def test_fp8_linear_trains_tiny_llama_mlp():
    """fp8 MLP on llama-tiny: loss decreases and stays close to bf16 run."""
    import os
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    os.environ.setdefault("LOCAL_RANK", "0")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29519")
    import deepspeed_amd
    from deepspeed_amd.models.llama import LLAMA_CONFIGS, LlamaForCausalLM
    from deepspeed_amd.ops.fp8_linear import Fp8Linear

    def run(fp8):
        torch.manual_seed(0)
        cfg = LLAMA_CONFIGS["llama-tiny"]
        with torch.device("cuda"):
            model = LlamaForCausalLM(cfg)
        if fp8:
            n = Fp8Linear.convert(model, include=["gate_proj", "up_proj",
                                                  "down_proj"])
            assert n == cfg.num_hidden_layers * 3
        config = {
            "train_micro_batch_size_per_gpu": 2,
            "optimizer": {"type": "AdamW", "params": {"lr": 3e-4}},
            "zero_optimization": {"stage": 3},
            "bf16": {"enabled": True},
        }
        engine, _, _, _ = deepspeed_amd.initialize(model=model,
                                                   config=config)
        g = torch.Generator().manual_seed(7)
        data = torch.randint(0, cfg.vocab_size, (2, 128), generator=g).cuda()
        losses = []
        for _ in range(8):
            loss = engine(data, labels=data)
            engine.backward(loss)
            engine.step()
            losses.append(loss.item())
        engine.destroy()
        return losses

    l_bf16 = run(False)
    l_fp8 = run(True)
    assert l_fp8[-1] < l_fp8[0] * 0.9, f"fp8 no progress: {l_fp8}"
    assert abs(l_fp8[-1] - l_bf16[-1]) < 0.5, \
        f"fp8 diverged from bf16: {l_fp8[-1]} vs {l_bf16[-1]}"


def test_fp8_quant_kernels_match_torch():
    """HIP amax/cast/cast_transpose vs the torch-op reference."""
    from deepspeed_amd.ops.loader import get_ext
    from deepspeed_amd.ops.fp8_linear import E4M3_MAX, E5M2_MAX, _quant
    ext = get_ext()
    torch.manual_seed(0)
    for M, K in ((128, 256), (192, 4096), (64, 48)):
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16) * 3
        amax = ext.fp8_amax(x)
        assert abs(amax.item() - x.abs().amax().item()) < 1e-3, \
            (amax.item(), x.abs().amax().item())
        for e5m2, fmax, dt in ((False, E4M3_MAX, torch.float8_e4m3fn),
                               (True, E5M2_MAX, torch.float8_e5m2)):
            scale = (amax[0] / fmax).clamp(min=1e-12)
            ref8, _ = _quant(x, dt, fmax)
            y = ext.fp8_cast(x, scale, e5m2).view(dt)
            # rounding mode at halfway points may differ by one ULP;
            # check the DEQUANTIZED error against the format's ULP bound
            recon_err = (y.float() * scale - x.float()).abs().max().item()
            ulp_bound = amax.item() * (2 ** -2 if e5m2 else 2 ** -3)
            assert recon_err <= ulp_bound, \
                f"e5m2={e5m2}: recon err {recon_err} > {ulp_bound}"
            mismatch = (y.view(torch.int8) != ref8.view(torch.int8)) \
                .float().mean().item()
            # e5m2 (2 mantissa bits) hits round-to-nearest-even halfway
            # cases more often; one-ULP disagreements are benign (the
            # dequantized-error bound above is the real check)
            assert mismatch < (0.05 if e5m2 else 0.02), \
                f"e5m2={e5m2} mismatch {mismatch}"
            yt_pair = ext.fp8_cast_transpose(x, scale, e5m2)
            y2, yt = yt_pair[0].view(dt), yt_pair[1].view(dt)
            assert torch.equal(y2.view(torch.int8), y.view(torch.int8))
            assert torch.equal(yt.view(torch.int8),
                               y.view(torch.int8).t().contiguous())
