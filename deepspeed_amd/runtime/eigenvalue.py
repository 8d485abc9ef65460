"""Hessian eigenvalue estimation via power iteration (MoQ scheduling).

Parity: reference `runtime/eigenvalue.py:13`.
"""
import torch

from ..utils.logging import log_dist


class Eigenvalue:
    def __init__(self, verbose=False, max_iter=100, tol=1e-2,
                 stability=1e-6, gas_boundary_resolution=1):
        self.verbose = verbose
        self.max_iter = max_iter
        self.tol = tol
        self.stability = stability

    def compute_eigenvalue(self, module, device=None, scale=1.0):
        """Largest |eigenvalue| of the loss Hessian w.r.t. module params,
        via Hv products on the existing grads (requires create_graph)."""
        params = [p for p in module.parameters()
                  if p.requires_grad and p.grad is not None and
                  p.grad.grad_fn is not None]
        if not params:
            return 0.0
        grads = [p.grad for p in params]
        v = [torch.randn_like(p) for p in params]
        norm = torch.sqrt(sum((x * x).sum() for x in v))
        v = [x / (norm + self.stability) for x in v]
        eig = 0.0
        for _ in range(self.max_iter):
            hv = torch.autograd.grad(grads, params, grad_outputs=v,
                                     retain_graph=True, allow_unused=True)
            hv = [h if h is not None else torch.zeros_like(p)
                  for h, p in zip(hv, params)]
            new_eig = float(sum((h * x).sum() for h, x in zip(hv, v)))
            norm = torch.sqrt(sum((h * h).sum() for h in hv))
            v = [h / (norm + self.stability) for h in hv]
            if abs(new_eig - eig) < self.tol * max(abs(eig), 1e-6):
                eig = new_eig
                break
            eig = new_eig
        return abs(eig) * scale
