"""CPU-path numerics of op wrappers vs plain fp32 torch."""
import torch
import pytest

from deepspeed_amd.ops import functional as Fops


def test_rms_norm_cpu():
    torch.manual_seed(0)
    x = torch.randn(4, 8, 32, requires_grad=True)
    w = torch.randn(32, requires_grad=True)
    y = Fops.rms_norm(x, w, eps=1e-5)
    ref = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + 1e-5) * w
    assert torch.allclose(y, ref, atol=1e-5)
    y.sum().backward()
    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    ref2 = x2 * torch.rsqrt(x2.pow(2).mean(-1, keepdim=True) + 1e-5) * w2
    ref2.sum().backward()
    assert torch.allclose(x.grad, x2.grad, atol=1e-4)
    assert torch.allclose(w.grad, w2.grad, atol=1e-4)


def test_rope_cpu():
    torch.manual_seed(0)
    B, S, H, D = 2, 16, 4, 32
    cos, sin = Fops.build_rope_cache(S, D, base=10000.0)
    t = torch.randn(B, S, H, D, requires_grad=True)
    out = Fops.apply_rope(t, cos, sin)
    # rotation preserves pair norms
    half = D // 2
    n_in = t.detach()[..., :half]**2 + t.detach()[..., half:]**2
    n_out = out.detach()[..., :half]**2 + out.detach()[..., half:]**2
    assert torch.allclose(n_in, n_out, atol=1e-4)
    # grad of rotation = inverse rotation
    g = torch.randn_like(out)
    out.backward(g)
    gref = Fops._rope_torch(g, cos, sin, sign=-1.0)
    assert torch.allclose(t.grad, gref, atol=1e-5)


def test_swiglu_cpu():
    torch.manual_seed(0)
    g = torch.randn(64, requires_grad=True)
    u = torch.randn(64, requires_grad=True)
    y = Fops.swiglu(g, u)
    ref = torch.nn.functional.silu(g) * u
    assert torch.allclose(y, ref, atol=1e-6)
    y.sum().backward()
    g2 = g.detach().clone().requires_grad_(True)
    u2 = u.detach().clone().requires_grad_(True)
    (torch.nn.functional.silu(g2) * u2).sum().backward()
    assert torch.allclose(g.grad, g2.grad, atol=1e-5)
    assert torch.allclose(u.grad, u2.grad, atol=1e-5)


def test_fused_cross_entropy_cpu():
    torch.manual_seed(0)
    N, V = 24, 100
    logits = torch.randn(N, V, requires_grad=True)
    targets = torch.randint(0, V, (N,))
    targets[3] = -100
    loss = Fops.fused_cross_entropy(logits, targets)
    ref = torch.nn.functional.cross_entropy(logits.detach(), targets,
                                            ignore_index=-100)
    assert torch.allclose(loss, ref, atol=1e-5)
    loss.backward()
    l2 = logits.detach().clone().requires_grad_(True)
    torch.nn.functional.cross_entropy(l2, targets,
                                      ignore_index=-100).backward()
    assert torch.allclose(logits.grad, l2.grad, atol=1e-5)


def test_layer_norm_cpu():
    torch.manual_seed(0)
    x = torch.randn(6, 32, requires_grad=True)
    w = torch.randn(32, requires_grad=True)
    b = torch.randn(32, requires_grad=True)
    y = Fops.layer_norm(x, w, b, eps=1e-5)
    ref = torch.nn.functional.layer_norm(x, (32,), w, b, eps=1e-5)
    assert torch.allclose(y, ref, atol=1e-5)
    y.sum().backward()
    x2, w2, b2 = (t.detach().clone().requires_grad_(True) for t in (x, w, b))
    torch.nn.functional.layer_norm(x2, (32,), w2, b2, 1e-5).sum().backward()
    assert torch.allclose(x.grad, x2.grad, atol=1e-4)
    assert torch.allclose(w.grad, w2.grad, atol=1e-4)
    assert torch.allclose(b.grad, b2.grad, atol=1e-4)


def test_fused_adam_cpu_matches_torch():
    torch.manual_seed(0)
    from deepspeed_amd.ops.adam import FusedAdam
    p1 = torch.randn(100, requires_grad=True)
    p2 = p1.detach().clone().requires_grad_(True)
    o1 = FusedAdam([p1], lr=1e-2, weight_decay=0.01)
    o2 = torch.optim.AdamW([p2], lr=1e-2, weight_decay=0.01)
    for i in range(5):
        g = torch.randn(100)
        p1.grad = g.clone()
        p2.grad = g.clone()
        o1.step()
        o2.step()
    assert torch.allclose(p1, p2, atol=1e-6), (p1 - p2).abs().max()


def test_tiny_llama_cpu_trains():
    from deepspeed_amd.models.llama import LLAMA_CONFIGS, LlamaForCausalLM
    cfg = LLAMA_CONFIGS["llama-tiny"]
    torch.manual_seed(0)
    model = LlamaForCausalLM(cfg).float()
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    data = torch.randint(0, cfg.vocab_size, (2, 32))
    losses = []
    for _ in range(5):
        opt.zero_grad()
        loss = model(data, labels=data)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0]
