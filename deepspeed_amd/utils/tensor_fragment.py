"""Fragment APIs: read/modify full fp32 params & optimizer state under ZeRO.

Parity: reference `deepspeed/utils/tensor_fragment.py`
(`safe_get_full_fp32_param`, `safe_get_full_grad`,
`safe_set_full_fp32_param`, `safe_get_full_optimizer_state`).

Collective: ALL ranks of the dp group must call these together.
"""
import torch

from .. import comm as dist


def _find_owner(optimizer, param):
    """Returns ('bucket'|'subgroup', container, offset, shard_numel)."""
    if hasattr(optimizer, "param_to_bucket"):
        b = optimizer.param_to_bucket.get(param)
        if b is not None:
            return "bucket", b, b.offsets[param], None
    if hasattr(optimizer, "param_to_subgroup"):
        sg = optimizer.param_to_subgroup.get(param)
        if sg is not None:
            return "subgroup", sg, sg.offsets[param], param.ds_shard_numel
    return None, None, None, None


def _gather_flat(optimizer, kind, container, source32):
    """Reassemble the container's FULL fp32 flat tensor across dp ranks."""
    pg = getattr(container, "pg", None) or optimizer.dp_group
    world = dist.get_world_size(pg)
    if world == 1:
        return source32.detach().float()
    out = torch.empty(source32.numel() * world, dtype=torch.float32,
                      device=source32.device)
    dist.all_gather_into_tensor(out, source32.detach().float().contiguous(),
                                group=pg)
    return out


def safe_get_full_fp32_param(param, optimizer=None):
    """Full fp32 master copy of `param` (collective over the dp group)."""
    opt = optimizer
    if opt is None:
        return None
    kind, container, off, shard_numel = _find_owner(opt, param)
    if kind == "bucket":
        full = _gather_flat(opt, kind, container, container.master32)
        return full[off:off + param.numel()].reshape(param.shape)
    if kind == "subgroup":
        if container.master32 is None:
            return None  # NVMe-resident
        pg = opt.dp_group
        world = dist.get_world_size(pg)
        shard = container.master32.detach().float()[off:off + shard_numel]
        if world == 1:
            return shard[:param.ds_numel].reshape(param.ds_shape)
        out = torch.empty(shard_numel * world, dtype=torch.float32,
                          device=shard.device)
        dist.all_gather_into_tensor(out, shard.contiguous(), group=pg)
        return out[:param.ds_numel].reshape(param.ds_shape)
    return None


def safe_set_full_fp32_param(param, value, optimizer=None):
    """Write a full fp32 tensor into the distributed master (and 16-bit)."""
    opt = optimizer
    if opt is None:
        return False
    kind, container, off, shard_numel = _find_owner(opt, param)
    flat = value.reshape(-1).float()
    if kind == "bucket":
        pg = getattr(container, "pg", None) or opt.dp_group
        rank = dist.get_rank(pg)
        lo = rank * container.shard_numel
        hi = lo + container.shard_numel
        # param occupies [off, off+numel) of the bucket flat buffer
        s = max(lo, off)
        e = min(hi, off + param.numel())
        if e > s:
            container.master32.data[s - lo:e - lo].copy_(
                flat[s - off:e - off])
            container.shard16[s - lo:e - lo].copy_(
                flat[s - off:e - off].to(container.shard16.dtype))
        dist.all_gather_into_tensor(container.flat16, container.shard16,
                                    group=pg)
        return True
    if kind == "subgroup":
        pg = opt.dp_group
        rank = dist.get_rank(pg)
        lo = rank * shard_numel
        hi = min(lo + shard_numel, param.ds_numel)
        if hi > lo and container.master32 is not None:
            container.master32.data[off:off + hi - lo].copy_(flat[lo:hi])
            container.flat16[off:off + hi - lo].copy_(
                flat[lo:hi].to(container.flat16.dtype))
        return True
    return False


def safe_get_full_optimizer_state(param, optim_state_key, optimizer=None):
    """Full fp32 optimizer state tensor (e.g. 'exp_avg') for `param`."""
    opt = optimizer
    if opt is None:
        return None
    kind, container, off, shard_numel = _find_owner(opt, param)
    if container is None:
        return None
    base = opt.optimizer
    st = base.state.get(container.master32)
    if not st or optim_state_key not in st:
        return None
    state32 = st[optim_state_key]
    if kind == "bucket":
        full = _gather_flat(opt, kind, container, state32)
        return full[off:off + param.numel()].reshape(param.shape)
    pg = opt.dp_group
    world = dist.get_world_size(pg)
    shard = state32.detach().float()[off:off + shard_numel]
    if world == 1:
        return shard[:param.ds_numel].reshape(param.ds_shape)
    out = torch.empty(shard_numel * world, dtype=torch.float32,
                      device=shard.device)
    dist.all_gather_into_tensor(out, shard.contiguous(), group=pg)
    return out[:param.ds_numel].reshape(param.ds_shape)


def safe_get_full_grad(param, optimizer=None):
    """Full fp32 accumulated gradient for `param` (post-reduce)."""
    opt = optimizer
    if opt is None:
        return None
    kind, container, off, shard_numel = _find_owner(opt, param)
    if kind == "bucket":
        if container.grad32 is None:
            return None
        full = _gather_flat(opt, kind, container, container.grad32)
        return full[off:off + param.numel()].reshape(param.shape)
    if kind == "subgroup":
        pg = opt.dp_group
        world = dist.get_world_size(pg)
        shard = container.grad32.detach().float()[off:off + shard_numel]
        if world == 1:
            return shard[:param.ds_numel].reshape(param.ds_shape)
        out = torch.empty(shard_numel * world, dtype=torch.float32,
                          device=shard.device)
        dist.all_gather_into_tensor(out, shard.contiguous(), group=pg)
        return out[:param.ds_numel].reshape(param.ds_shape)
    return None
