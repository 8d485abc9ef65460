"""Test fixtures (mirrors role of reference tests/unit/simple_model.py)."""
import torch


class SimpleModel(torch.nn.Module):
    def __init__(self, hidden_dim=32, nlayers=3, empty_grad=False):
        super().__init__()
        self.linears = torch.nn.ModuleList(
            [torch.nn.Linear(hidden_dim, hidden_dim) for _ in range(nlayers)])
        self.norm = torch.nn.LayerNorm(hidden_dim)
        self.loss_fn = torch.nn.MSELoss()

    def forward(self, x, y):
        h = x
        for l in self.linears:
            h = torch.nn.functional.gelu(l(h))
        h = self.norm(h)
        return self.loss_fn(h, y)


def make_batches(n, batch, hidden, seed=7, dtype=torch.float32):
    g = torch.Generator().manual_seed(seed)
    return [(torch.randn(batch, hidden, generator=g, dtype=torch.float32).to(dtype),
             torch.randn(batch, hidden, generator=g, dtype=torch.float32).to(dtype))
            for _ in range(n)]


def reference_adamw_training(model_ctor, batches, lr=1e-3, steps=None,
                             weight_decay=0.0, grad_accum=1):
    """Plain fp32 torch AdamW full-batch reference."""
    torch.manual_seed(11)
    model = model_ctor()
    opt = torch.optim.AdamW(model.parameters(), lr=lr, betas=(0.9, 0.999),
                            eps=1e-8, weight_decay=weight_decay)
    it = iter(batches)
    steps = steps if steps is not None else len(batches) // grad_accum
    for _ in range(steps):
        opt.zero_grad()
        for _ in range(grad_accum):
            x, y = next(it)
            loss = model(x.float(), y.float()) / grad_accum
            loss.backward()
        opt.step()
    return model
