"""Checkpoint tensor helpers (ref checkpoint/utils.py)."""
import torch


def clone_tensors_for_torch_save(item, device=torch.device("cpu")):
    """Deep-copy every tensor in `item` onto `device` so torch.save
    doesn't serialize live views of (possibly huge, possibly shared)
    training storage (ref checkpoint/utils.py:41)."""
    if isinstance(device, str):
        device = torch.device(device)
    if torch.is_tensor(item):
        return item.detach().to(device, copy=True).clone()
    if isinstance(item, dict):
        return {k: clone_tensors_for_torch_save(v, device)
                for k, v in item.items()}
    if isinstance(item, (list, tuple)):
        t = type(item)(clone_tensors_for_torch_save(v, device)
                       for v in item)
        return t
    return item
