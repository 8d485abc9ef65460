"""Tiled MLP / tiled logits loss: exact match vs untiled."""
import torch

from deepspeed_amd.sequence.tiled import (TiledMLP, sequence_tiled_compute,
                                          tiled_logits_loss)


def test_sequence_tiled_compute_matches():
    torch.manual_seed(0)
    lin = torch.nn.Linear(16, 16)
    x = torch.randn(2, 12, 16, requires_grad=True)
    out = sequence_tiled_compute(lin, x, 3, compute_params=True)
    ref = lin(x)
    assert torch.allclose(out, ref, atol=1e-6)
    out.sum().backward()
    g1 = (x.grad.clone(), lin.weight.grad.clone())
    x.grad = None
    lin.weight.grad = None
    lin(x).sum().backward()
    assert torch.allclose(g1[0], x.grad, atol=1e-6)
    assert torch.allclose(g1[1], lin.weight.grad, atol=1e-6)


def test_tiled_mlp():
    torch.manual_seed(0)
    mlp = torch.nn.Sequential(torch.nn.Linear(8, 32), torch.nn.GELU(),
                              torch.nn.Linear(32, 8))
    tiled = TiledMLP(mlp, num_shards=4)
    x = torch.randn(1, 16, 8)
    assert torch.allclose(tiled(x), mlp(x), atol=1e-6)


def test_tiled_logits_loss_matches():
    torch.manual_seed(0)
    B, S, H, V = 2, 17, 16, 100
    hidden = torch.randn(B, S, H, requires_grad=True)
    w = torch.randn(V, H, requires_grad=True)
    labels = torch.randint(0, V, (B, S))
    labels[0, 3] = -100
    loss = tiled_logits_loss(hidden, w, labels, num_shards=4)
    # reference: full logits CE
    logits = torch.nn.functional.linear(hidden[:, :-1], w)
    ref = torch.nn.functional.cross_entropy(
        logits.reshape(-1, V), labels[:, 1:].reshape(-1),
        ignore_index=-100)
    assert torch.allclose(loss, ref, atol=1e-5), (loss, ref)
    loss.backward()
    h2 = hidden.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    logits2 = torch.nn.functional.linear(h2[:, :-1], w2)
    torch.nn.functional.cross_entropy(
        logits2.reshape(-1, V), labels[:, 1:].reshape(-1),
        ignore_index=-100).backward()
    assert torch.allclose(hidden.grad, h2.grad, atol=1e-5)
    assert torch.allclose(w.grad, w2.grad, atol=1e-4)


def test_enable_tiled_mlp_for_hf_matches_dense():
    import torch
    from deepspeed_amd.sequence.tiled import enable_tiled_mlp_for_hf

    class MLP(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.gate_proj = torch.nn.Linear(16, 32, bias=False)
            self.up_proj = torch.nn.Linear(16, 32, bias=False)
            self.down_proj = torch.nn.Linear(32, 16, bias=False)

        def forward(self, x):
            return self.down_proj(
                torch.nn.functional.silu(self.gate_proj(x)) *
                self.up_proj(x))

    torch.manual_seed(0)
    m1 = MLP()
    m2 = MLP()
    m2.load_state_dict(m1.state_dict())
    n = enable_tiled_mlp_for_hf(m2, num_shards=4)
    assert n == 1
    x1 = torch.randn(2, 64, 16, requires_grad=True)
    x2 = x1.detach().clone().requires_grad_(True)
    y1 = m1(x1)
    y2 = m2(x2)
    assert torch.allclose(y1, y2, atol=1e-6)
    g = torch.randn_like(y1)
    y1.backward(g)
    y2.backward(g)
    assert torch.allclose(x1.grad, x2.grad, atol=1e-6)
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1.grad, p2.grad, atol=1e-6)


def _agree_body():
    import torch.distributed as dist
    from deepspeed_amd.sequence.tiled import agree_num_shards
    mine = 3 if dist.get_rank() == 0 else 7
    return agree_num_shards(mine)


def test_agree_num_shards_world2():
    from tests.common import run_distributed
    outs = run_distributed(_agree_body, world_size=2)
    assert outs == [7, 7]
