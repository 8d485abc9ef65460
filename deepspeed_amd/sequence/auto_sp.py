"""AutoSP — automatic sequence-parallelism insertion.

Parity: reference `deepspeed/sequence/auto_sp.py` + `autosp_detector.py`
(+ the DeepCompile pass `compile/passes/sp_compile.py`): detect attention
modules in a model and rewrite them for Ulysses sequence parallelism
without the user hand-wrapping each layer.

MI355X-native design: instead of a torch.compile graph pass (the
reference rides DeepCompile), AutoSP here is a MODULE-LEVEL detector —
it walks the module tree, recognizes attention blocks (either this
package's model classes via their `enable_ulysses` hook, or generic
modules exposing a `num_heads`/`num_attention_heads` attribute and a
flash/sdpa call), picks an SP degree that divides both the head count
and the world size, builds the SP process group, and wraps the attention
with `DistributedAttention` (a2a over RCCL/xGMI). Graph-level insertion
is a ROADMAP item alongside the DeepCompile-class planner.
"""
import torch

from .. import comm as dist
from ..comm import groups
from ..utils.logging import log_dist
from .layer import DistributedAttention


def _lookup(mod, names):
    for holder in (mod, getattr(mod, "cfg", None),
                   getattr(mod, "config", None)):
        if holder is None:
            continue
        for attr in names:
            v = getattr(holder, attr, None)
            if isinstance(v, int) and v > 0:
                return v
    return None


def _head_count(mod):
    # only attention-looking modules count (a bare config object on the
    # root model should not match)
    if not any(hasattr(mod, a) for a in ("q_proj", "qkv_proj", "query",
                                         "Wq", "attn_fn", "core_attention",
                                         "_dist_attn", "sp_group")):
        return None
    return _lookup(mod, ("num_heads", "num_attention_heads", "n_heads",
                         "n_head"))


def _kv_head_count(mod):
    return _lookup(mod, ("num_kv_heads", "num_key_value_heads",
                         "n_kv_heads"))


def pick_sp_degree(seq_len, num_heads, num_kv_heads=None, world_size=None,
                   seq_threshold=8192):
    """Largest SP degree that (a) is needed for this seq_len, (b) divides
    the world size, and (c) divides every head count involved."""
    if world_size is None:
        world_size = dist.get_world_size() if dist.is_initialized() else 1
    if seq_len < seq_threshold or world_size == 1:
        return 1
    # how many ways we'd LIKE to split to bring local seq under threshold
    want = max(1, (seq_len + seq_threshold - 1) // seq_threshold)
    mha = num_kv_heads is None or num_kv_heads == num_heads
    best = 1
    for d in range(1, world_size + 1):
        if world_size % d:
            continue
        if num_heads % d and not (mha and d <= num_heads):
            continue  # uneven distribution only supported for MHA
        if not mha and num_kv_heads % d and d % num_kv_heads:
            continue  # GQA: degree must divide kv heads OR replicate them
        if d <= want:
            best = d
    return best


def configure_auto_sp(model, seq_len, world_size=None, seq_threshold=8192):
    """Detect attention modules and enable Ulysses SP when profitable.

    Returns the chosen SP degree (1 = left unchanged)."""
    heads = None
    kv_heads = None
    for mod in model.modules():
        h = _head_count(mod)
        if h is not None:
            heads = h if heads is None else min(heads, h)
            kv = _kv_head_count(mod)
            if kv is not None:
                kv_heads = kv if kv_heads is None else min(kv_heads, kv)
    if heads is None:
        log_dist("AutoSP: no attention modules detected", ranks=[0])
        return 1
    degree = pick_sp_degree(seq_len, heads, kv_heads, world_size,
                            seq_threshold)
    if degree <= 1:
        return 1
    groups.initialize_sequence_parallel(degree)
    sp_group = groups.get_sequence_parallel_group()
    # package models expose enable_ulysses; generic models get their
    # attention callables wrapped
    enable = getattr(model, "enable_ulysses", None)
    if enable is None:
        from ..models.llama import enable_ulysses as enable_fn
        try:
            enable_fn(model, sp_group)
        except Exception:
            _wrap_generic(model, sp_group)
    else:
        enable(sp_group)
    log_dist(f"AutoSP: sequence parallel degree {degree} "
             f"(heads={heads}, kv={kv_heads}, seq={seq_len})", ranks=[0])
    return degree


def _wrap_generic(model, sp_group):
    for mod in model.modules():
        if _head_count(mod) is None:
            continue
        attn = getattr(mod, "core_attention", None) or \
            getattr(mod, "attn_fn", None)
        if callable(attn) and not isinstance(attn, DistributedAttention):
            wrapped = DistributedAttention(attn, sp_group)
            if hasattr(mod, "core_attention"):
                mod.core_attention = wrapped
            else:
                mod.attn_fn = wrapped


# ---------------------------------------------------------------- compile pass
def _sp_sdpa(query, key, value, *args, **kwargs):
    """SDPA with Ulysses all-to-alls spliced around it: inputs arrive
    sequence-sharded [B, H, s_local, D]; heads scatter / sequence
    gathers before the attention, inverse after."""
    import torch.nn.functional as F

    from ..comm import groups
    from .layer import _SeqAllToAll
    spg = groups.get_sequence_parallel_group()
    # [B,H,s,D] -> [B,s,H,D] for the a2a layout
    q = query.transpose(1, 2).contiguous()
    k = key.transpose(1, 2).contiguous()
    v = value.transpose(1, 2).contiguous()
    q = _SeqAllToAll.apply(spg, q, 2, 1)
    k = _SeqAllToAll.apply(spg, k, 2, 1)
    v = _SeqAllToAll.apply(spg, v, 2, 1)
    out = F.scaled_dot_product_attention(
        q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
        *args, **kwargs)
    out = out.transpose(1, 2).contiguous()
    out = _SeqAllToAll.apply(spg, out, 1, 2)
    return out.transpose(1, 2)


def autosp_backend(gm, example_inputs):
    """torch.compile backend: rewrite every scaled_dot_product_attention
    node in the captured graph to the sequence-parallel form (ref
    compile/passes/sp_compile.py — AutoSP as a graph pass, not a module
    wrapper). Use: torch.compile(model, backend=autosp_backend) on a
    model fed sequence-sharded inputs, with the SP group initialized."""
    import torch
    import torch.nn.functional as F
    n = 0
    targets = {F.scaled_dot_product_attention,
               torch.ops.aten.scaled_dot_product_attention.default}
    for node in gm.graph.nodes:
        if node.op == "call_function" and node.target in targets:
            node.target = _sp_sdpa
            n += 1
    if n:
        gm.recompile()
        log_dist(f"AutoSP compile pass: rewrote {n} attention nodes",
                 ranks=[0])
    return gm


def autosp_compile(model, sp_size=None):
    """Convenience: initialize the SP group and return
    torch.compile(model, backend=autosp_backend)."""
    import torch

    from ..comm import groups
    if sp_size is not None and not groups.sequence_parallel_is_initialized():
        groups.initialize_sequence_parallel(sp_size)
    return torch.compile(model, backend=autosp_backend)
