"""Activation checkpointing: grads match non-checkpointed; cpu offload."""
import torch

from deepspeed_amd.runtime import activation_checkpointing as ac


def _block(x, w1, w2):
    return torch.tanh(x @ w1) @ w2


def test_checkpoint_matches_plain():
    torch.manual_seed(0)
    x = torch.randn(8, 16, requires_grad=True)
    w1 = torch.randn(16, 16, requires_grad=True)
    w2 = torch.randn(16, 16, requires_grad=True)
    out = ac.checkpoint(_block, x, w1, w2)
    out.sum().backward()
    g = (x.grad.clone(), w1.grad.clone(), w2.grad.clone())

    x2 = x.detach().clone().requires_grad_(True)
    w12 = w1.detach().clone().requires_grad_(True)
    w22 = w2.detach().clone().requires_grad_(True)
    _block(x2, w12, w22).sum().backward()
    assert torch.allclose(g[0], x2.grad, atol=1e-6)
    assert torch.allclose(g[1], w12.grad, atol=1e-6)
    assert torch.allclose(g[2], w22.grad, atol=1e-6)


def test_checkpoint_cpu_offload():
    ac.configure(checkpoint_in_cpu=True)
    try:
        torch.manual_seed(0)
        x = torch.randn(4, 8, requires_grad=True)
        w = torch.randn(8, 8, requires_grad=True)
        out = ac.checkpoint(lambda a, b: (a @ b).relu(), x, w)
        out.sum().backward()
        assert x.grad is not None and w.grad is not None
    finally:
        ac.configure(checkpoint_in_cpu=False)


def test_rng_tracker_deterministic_dropout():
    ac.model_parallel_cuda_manual_seed(1234)

    def block(x):
        with ac.get_cuda_rng_tracker().fork():
            return torch.dropout(x, 0.5, True)

    torch.manual_seed(3)
    x = torch.randn(16, 16, requires_grad=True)
    out = ac.checkpoint(block, x)
    out.sum().backward()  # recompute must reproduce the same dropout mask
    assert x.grad is not None
