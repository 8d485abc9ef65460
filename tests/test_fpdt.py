"""FPDT chunked attention == full attention (with and without offload)."""
import math

import torch

from deepspeed_amd.sequence.fpdt import fpdt_attention, update_out_and_lse


def _full_attention(q, k, v, causal=True):
    qt, kt, vt = (t.permute(0, 2, 1, 3).float() for t in (q, k, v))
    G = qt.shape[1] // kt.shape[1]
    if G > 1:
        kt = kt.repeat_interleave(G, dim=1)
        vt = vt.repeat_interleave(G, dim=1)
    s = qt @ kt.transpose(-1, -2) / math.sqrt(q.shape[-1])
    if causal:
        S = q.shape[1]
        m = torch.triu(torch.ones(S, S, dtype=torch.bool), 1)
        s = s.masked_fill(m, float("-inf"))
    p = torch.softmax(s, -1)
    return (p @ vt).permute(0, 2, 1, 3)


def test_lse_merge():
    torch.manual_seed(0)
    B, S, H, D = 1, 8, 2, 4
    o1 = torch.randn(B, S, H, D)
    o2 = torch.randn(B, S, H, D)
    l1 = torch.randn(B, H, S)
    l2 = torch.randn(B, H, S)
    out, lse = update_out_and_lse(None, None, o1, l1)
    out, lse = update_out_and_lse(out, lse, o2, l2)
    w1 = torch.exp(l1 - lse).permute(0, 2, 1).unsqueeze(-1)
    w2 = torch.exp(l2 - lse).permute(0, 2, 1).unsqueeze(-1)
    ref = o1 * w1 + o2 * w2
    assert torch.allclose(out, ref, atol=1e-5)
    assert torch.allclose(torch.exp(lse), torch.exp(l1) + torch.exp(l2),
                          atol=1e-4)


def test_fpdt_matches_full():
    torch.manual_seed(0)
    B, S, H, D = 2, 64, 4, 16
    q = torch.randn(B, S, H, D)
    k = torch.randn(B, S, 2, D)
    v = torch.randn(B, S, 2, D)
    ref = _full_attention(q, k, v, causal=True)
    got = fpdt_attention(q, k, v, chunk_size=16, causal=True)
    assert torch.allclose(got, ref, atol=1e-4), (got - ref).abs().max()


def test_fpdt_offload_matches():
    torch.manual_seed(1)
    B, S, H, D = 1, 48, 2, 8
    q = torch.randn(B, S, H, D)
    k = torch.randn(B, S, H, D)
    v = torch.randn(B, S, H, D)
    ref = _full_attention(q, k, v)
    got = fpdt_attention(q, k, v, chunk_size=16, offload_to_host=True)
    assert torch.allclose(got, ref, atol=1e-4)


def test_fpdt_ffn_and_logits_loss_match_dense():
    import math
    import torch
    from deepspeed_amd.sequence.fpdt import fpdt_ffn, fpdt_logits_loss
    torch.manual_seed(0)
    ffn = torch.nn.Sequential(torch.nn.Linear(32, 64), torch.nn.GELU(),
                              torch.nn.Linear(64, 32))
    x = torch.randn(2, 64, 32, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    y_ref = ffn(x)
    y = fpdt_ffn(ffn, x2, chunk_size=16)
    assert torch.allclose(y, y_ref, atol=1e-6)
    g = torch.randn_like(y)
    y_ref.backward(g)
    y.backward(g)
    assert torch.allclose(x.grad, x2.grad, atol=1e-6)

    V = 128
    w = torch.randn(V, 32) * 0.05
    h = torch.randn(2, 64, 32, requires_grad=True)
    h2 = h.detach().clone().requires_grad_(True)
    labels = torch.randint(0, V, (2, 64))
    ref = torch.nn.functional.cross_entropy(
        (h[:, :-1] @ w.t()).reshape(-1, V).float(),
        labels[:, 1:].reshape(-1))
    got = fpdt_logits_loss(h2, w, labels, chunk_size=16)
    assert abs(ref.item() - got.item()) < 1e-4, (ref.item(), got.item())
    ref.backward()
    got.backward()
    assert torch.allclose(h.grad, h2.grad, atol=1e-5), \
        (h.grad - h2.grad).abs().max()
