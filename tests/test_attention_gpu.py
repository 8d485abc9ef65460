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
