"""Host-memory pinning helpers.

Parity role: reference `utils/pin_memory.py` (NativePinnedMemory /
pin_memory helpers). On this stack torch's cudaHostRegister-backed
`Tensor.pin_memory()` is the one pinning path (the offload tiers pin
their slabs directly); these helpers pin existing structures in place.
"""
import torch


def pin_tensor(t):
    """Return a pinned copy (no-op if already pinned or no GPU)."""
    if not torch.cuda.is_available() or t.is_cuda or t.is_pinned():
        return t
    return t.pin_memory()


def pin_module_buffers(module):
    """Pin every CPU parameter/buffer of `module` in place; returns the
    number of tensors pinned."""
    n = 0
    with torch.no_grad():
        for p in module.parameters():
            if not p.is_cuda and not p.data.is_pinned() \
                    and torch.cuda.is_available():
                p.data = p.data.pin_memory()
                n += 1
        for b in module.buffers():
            if not b.is_cuda and not b.data.is_pinned() \
                    and torch.cuda.is_available():
                b.data = b.data.pin_memory()
                n += 1
    return n


def pin_optimizer_state(optimizer):
    """Pin CPU optimizer-state tensors (host-offloaded moments) in
    place; returns the number pinned."""
    n = 0
    for st in optimizer.state.values():
        for k, v in st.items():
            if torch.is_tensor(v) and not v.is_cuda \
                    and not v.is_pinned() and torch.cuda.is_available():
                st[k] = v.pin_memory()
                n += 1
    return n
