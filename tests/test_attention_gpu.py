"""Flash attention HIP kernel vs fp32 torch reference (asymmetric random
inputs per guide G9 — catches operand/output transposes)."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu


def _ref_attention(q, k, v, causal):
    """fp32 eager reference with GQA."""
    B, S, Hq, D = q.shape
    Hk = k.shape[2]
    G = Hq // Hk
    q32 = q.float().permute(0, 2, 1, 3)           # [B,Hq,S,D]
    k32 = k.float().permute(0, 2, 1, 3)           # [B,Hk,S,D]
    v32 = v.float().permute(0, 2, 1, 3)
    k32 = k32.repeat_interleave(G, dim=1)
    v32 = v32.repeat_interleave(G, dim=1)
    s = q32 @ k32.transpose(-1, -2) / math.sqrt(D)
    if causal:
        mask = torch.triu(torch.ones(S, S, device=q.device, dtype=torch.bool),
                          diagonal=1)
        s = s.masked_fill(mask, float("-inf"))
    p = torch.softmax(s, dim=-1)
    return (p @ v32).permute(0, 2, 1, 3)


@pytest.mark.parametrize("B,S,Hq,Hk", [(2, 256, 8, 2), (1, 1024, 4, 4),
                                       (1, 333, 8, 2)])
@pytest.mark.parametrize("causal", [True, False])
def test_flash_fwd(B, S, Hq, Hk, causal):
    torch.manual_seed(0)
    D = 128
    q = torch.randn(B, S, Hq, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, S, Hk, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, S, Hk, D, device="cuda", dtype=torch.bfloat16)
    from deepspeed_amd.ops.loader import get_ext
    out, lse = get_ext().flash_attn_fwd(q, k, v, causal, 1.0 / math.sqrt(D))
    ref = _ref_attention(q, k, v, causal)
    diff = (out.float() - ref).abs()
    assert diff.max() < 3e-2, f"max diff {diff.max().item()}"
    # LSE check
    s = (q.float().permute(0, 2, 1, 3) @
         k.float().permute(0, 2, 1, 3).repeat_interleave(
             Hq // Hk, dim=1).transpose(-1, -2)) / math.sqrt(D)
    if causal:
        mask = torch.triu(torch.ones(S, S, device=q.device, dtype=torch.bool),
                          diagonal=1)
        s = s.masked_fill(mask, float("-inf"))
    lse_ref = torch.logsumexp(s, dim=-1)
    assert (lse - lse_ref).abs().max() < 2e-2


@pytest.mark.parametrize("B,S,Hq,Hk", [(1, 512, 8, 2)])
def test_flash_bwd(B, S, Hq, Hk):
    torch.manual_seed(1)
    D = 128
    from deepspeed_amd.ops.attention import _FlashAttnFn
    q = torch.randn(B, S, Hq, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn(B, S, Hk, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    v = torch.randn(B, S, Hk, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    out = _FlashAttnFn.apply(q, k, v, True, 1.0 / math.sqrt(D))
    g = torch.randn_like(out)
    out.backward(g)

    q32 = q.detach().float().requires_grad_(True)
    k32 = k.detach().float().requires_grad_(True)
    v32 = v.detach().float().requires_grad_(True)
    ref = _ref_attention(q32, k32, v32, True)
    ref.backward(g.float())

    assert (out.float() - ref).abs().max() < 3e-2
    for got, exp, name in [(q.grad, q32.grad, "dq"), (k.grad, k32.grad, "dk"),
                           (v.grad, v32.grad, "dv")]:
        d = (got.float() - exp).abs()
        rel = d.max() / (exp.abs().max() + 1e-6)
        assert rel < 5e-2, f"{name}: max abs {d.max().item()} rel {rel}"


def _ref_attention_mask(q, k, v, causal, kvmask):
    """fp32 reference with additive kv-column mask [B,S]."""
    B, S, Hq, D = q.shape
    Hk = k.shape[2]
    G = Hq // Hk
    q32 = q.float().permute(0, 2, 1, 3)
    k32 = k.float().permute(0, 2, 1, 3).repeat_interleave(G, dim=1)
    v32 = v.float().permute(0, 2, 1, 3).repeat_interleave(G, dim=1)
    s = q32 @ k32.transpose(-1, -2) / math.sqrt(D)
    if kvmask is not None:
        s = s + kvmask.view(B, 1, 1, S)
    if causal:
        cm = torch.triu(torch.ones(S, S, device=q.device, dtype=torch.bool),
                        diagonal=1)
        s = s.masked_fill(cm, float("-inf"))
    p = torch.softmax(s, dim=-1)
    return (p @ v32).permute(0, 2, 1, 3)


@pytest.mark.parametrize("D", [64, 128])
@pytest.mark.parametrize("causal", [True, False])
def test_flash_fwd_headdim(D, causal):
    """Templated head-dim kernel (BERT-class D=64) vs fp32 reference."""
    torch.manual_seed(1)
    B, S, Hq, Hk = 2, 384, 8, 4
    q = torch.randn(B, S, Hq, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, S, Hk, D, device="cuda", dtype=torch.bfloat16) * 0.5
    v = torch.randn(B, S, Hk, D, device="cuda", dtype=torch.bfloat16) * 2.0
    from deepspeed_amd.ops.loader import get_ext
    out, lse = get_ext().flash_attn_fwd(q, k, v, causal, 1.0 / math.sqrt(D))
    ref = _ref_attention(q, k, v, causal)
    diff = (out.float() - ref).abs()
    assert diff.max() < 3e-2, f"D={D}: max diff {diff.max().item()}"


@pytest.mark.parametrize("D", [64, 128])
def test_flash_fwd_kv_padding_mask(D):
    """Additive [B,S] kv padding mask (-inf on padded columns)."""
    torch.manual_seed(2)
    B, S, Hq, Hk = 2, 320, 4, 4
    q = torch.randn(B, S, Hq, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, S, Hk, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, S, Hk, D, device="cuda", dtype=torch.bfloat16)
    kvmask = torch.zeros(B, S, device="cuda")
    kvmask[0, 250:] = float("-inf")   # pad tail of sequence 0
    kvmask[1, 100:180] = -5.0         # soft bias mid-sequence
    from deepspeed_amd.ops.loader import get_ext
    out, lse = get_ext().flash_attn_fwd(q, k, v, False, 1.0 / math.sqrt(D),
                                        kvmask)
    ref = _ref_attention_mask(q, k, v, False, kvmask)
    diff = (out.float() - ref).abs()
    assert diff.max() < 3e-2, f"D={D}: max diff {diff.max().item()}"


@pytest.mark.parametrize("D", [64, 128])
def test_flash_bwd_headdim_and_mask(D):
    """Backward parity vs autograd on the fp32 reference (with mask)."""
    torch.manual_seed(3)
    B, S, Hq, Hk = 1, 256, 4, 2
    q = torch.randn(B, S, Hq, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn(B, S, Hk, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    v = torch.randn(B, S, Hk, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    kvmask = torch.zeros(B, S, device="cuda")
    kvmask[:, 200:] = float("-inf")
    from deepspeed_amd.ops.attention import _FlashAttnFn
    out = _FlashAttnFn.apply(q, k, v, True, 1.0 / math.sqrt(D), kvmask)
    g = torch.randn_like(out)
    out.backward(g)

    # reference graph on fp32 clones
    q3 = q.detach().float().requires_grad_(True)
    k3 = k.detach().float().requires_grad_(True)
    v3 = v.detach().float().requires_grad_(True)
    s = (q3.permute(0, 2, 1, 3) @
         k3.permute(0, 2, 1, 3).repeat_interleave(Hq // Hk, dim=1)
         .transpose(-1, -2)) / math.sqrt(D)
    s = s + kvmask.view(B, 1, 1, S)
    cm = torch.triu(torch.ones(S, S, device="cuda", dtype=torch.bool), 1)
    s = s.masked_fill(cm, float("-inf"))
    refo = (torch.softmax(s, -1) @
            v3.permute(0, 2, 1, 3).repeat_interleave(Hq // Hk, dim=1)) \
        .permute(0, 2, 1, 3)
    refo.backward(g.float())
    for got, want, name in ((q.grad, q3.grad, "dq"), (k.grad, k3.grad, "dk"),
                            (v.grad, v3.grad, "dv")):
        d = (got.float() - want).abs().max().item()
        scale_ref = want.abs().max().item() + 1e-6
        assert d / scale_ref < 6e-2, f"D={D} {name}: rel err {d/scale_ref}"
