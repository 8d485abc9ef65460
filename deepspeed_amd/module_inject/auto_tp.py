"""AutoTP: automatic tensor-parallel sharding of model families.

Parity: reference `module_inject/auto_tp.py:198` (AutoTP: classify linears
row/col, replace with sharded layers). Round-1 scope: the native Llama /
Mixtral families — q/k/v/gate/up column-parallel, o/down row-parallel,
heads divided across the TP group; norms/embeddings/lm_head replicated.
"""
import torch

from .. import comm as dist
from ..comm import groups as grp
from ..utils.logging import log_dist
from .layers import LinearAllreduce, LinearLayer


def apply_tensor_parallel(model, tp_group=None):
    """Shard a native LlamaForCausalLM/Mixtral in place for TP."""
    from ..models.llama import LlamaAttention, LlamaMLP
    group = tp_group if tp_group is not None \
        else grp.get_tensor_parallel_group()
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    if world == 1:
        return model
    n_shard, n_rep = 0, 0
    for mod in model.modules():
        if isinstance(mod, LlamaAttention):
            cfg = mod.cfg
            assert cfg.num_attention_heads % world == 0
            assert cfg.num_key_value_heads % world == 0
            mod.q_proj = LinearLayer.from_linear(mod.q_proj, group, rank,
                                                 world)
            mod.k_proj = LinearLayer.from_linear(mod.k_proj, group, rank,
                                                 world)
            mod.v_proj = LinearLayer.from_linear(mod.v_proj, group, rank,
                                                 world)
            mod.o_proj = LinearAllreduce.from_linear(mod.o_proj, group, rank,
                                                     world)
            n_shard += 4
        elif isinstance(mod, LlamaMLP):
            mod.gate_proj = LinearLayer.from_linear(mod.gate_proj, group,
                                                    rank, world)
            mod.up_proj = LinearLayer.from_linear(mod.up_proj, group, rank,
                                                  world)
            mod.down_proj = LinearAllreduce.from_linear(mod.down_proj, group,
                                                        rank, world)
            n_shard += 3
        else:
            n_rep += 1
    log_dist(f"AutoTP: sharded {n_shard} linears across tp={world}",
             ranks=[0])
    return model


# name-pattern classification for arbitrary (HF) model families
# (ref auto_tp.py:294 tp_parser / :357 _replace)
COLUMN_PATTERNS = ("q_proj", "k_proj", "v_proj", "gate_proj", "up_proj",
                   "query_key_value", "dense_h_to_4h", "fc1", "wi_0",
                   "wi_1", "c_attn", "query", "key", "value",
                   ".w1", ".w3", "c_fc")  # mixtral experts / gpt2 mlp
ROW_PATTERNS = ("o_proj", "down_proj", "dense_4h_to_h", "out_proj",
                "fc2", "wo", "c_proj", "attention.dense", ".w2")
# per-rank head attributes HF attention modules carry
HEAD_ATTRS = ("num_heads", "num_attention_heads", "num_key_value_heads",
              "num_kv_heads", "embed_dim", "hidden_size", "split_size",
              "all_head_size")


def tp_parser(model):
    """Classify every Linear by name into column/row/replicated.
    Returns {qualified_name: "column"|"row"} (ref AutoTP.tp_parser)."""
    plan = {}
    for name, mod in model.named_modules():
        if not isinstance(mod, torch.nn.Linear):
            continue
        if any(p in name for p in ROW_PATTERNS):
            plan[name] = "row"
        elif any(p in name for p in COLUMN_PATTERNS):
            plan[name] = "column"
    return plan


def apply_tensor_parallel_hf(model, tp_group=None, plan=None):
    """Generic (HF-compatible) AutoTP: shard Linears per the name-pattern
    plan and divide per-rank head-count attributes on the parent
    attention modules."""
    group = tp_group if tp_group is not None \
        else grp.get_tensor_parallel_group()
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    if world == 1:
        return model
    plan = plan or tp_parser(model)
    mods = dict(model.named_modules())
    n = 0
    for name, kind in plan.items():
        parent_name, _, child = name.rpartition(".")
        parent = mods.get(parent_name, model)
        lin = getattr(parent, child)
        if kind == "column":
            assert lin.out_features % world == 0, name
            new = LinearLayer.from_linear(lin, group, rank, world)
        else:
            assert lin.in_features % world == 0, name
            new = LinearAllreduce.from_linear(lin, group, rank, world)
        setattr(parent, child, new)
        n += 1
    # per-rank head bookkeeping — only on ATTENTION modules (ones whose
    # sharded children look like q/k/v); dividing size attrs on MLP
    # modules would corrupt families that read them in forward
    attn_markers = ("q_proj", "query", "c_attn", "query_key_value")
    touched_parents = {}
    for name in plan:
        pname, _, child = name.rpartition(".")
        touched_parents.setdefault(pname, []).append(child)
    for pname, children in touched_parents.items():
        if not any(any(m in c for m in attn_markers) for c in children):
            continue
        pm = mods.get(pname)
        if pm is None:
            continue
        for attr in HEAD_ATTRS:
            v = getattr(pm, attr, None)
            if isinstance(v, int) and v % world == 0:
                setattr(pm, attr, v // world)
    log_dist(f"AutoTP(HF): sharded {n} linears across tp={world}",
             ranks=[0])
    return model


def add_tp_training_hooks(model, tp_group=None):
    """Training AutoTP (ref module_inject training path): replicated
    (non-sharded) params — norms, embeddings, lm_head — see the SAME
    batch on every TP rank, so their grads must all-reduce (average)
    over the TP group each backward; sharded layers' grads stay
    rank-local by construction."""
    group = tp_group if tp_group is not None \
        else grp.get_tensor_parallel_group()
    world = dist.get_world_size(group)
    if world == 1:
        return []
    sharded_params = set()
    for mod in model.modules():
        if isinstance(mod, (LinearLayer, LinearAllreduce)):
            for p in mod.parameters(recurse=False):
                sharded_params.add(id(p))

    def make_hook(pg):
        def hook(param):
            param.grad.div_(dist.get_world_size(pg))
            dist.all_reduce(param.grad, group=pg)
        return hook

    handles = []
    for p in model.parameters():
        if id(p) not in sharded_params and p.requires_grad:
            handles.append(
                p.register_post_accumulate_grad_hook(make_hook(group)))
    log_dist(f"AutoTP training: {len(handles)} replicated params "
             f"all-reduce grads over tp={world}", ranks=[0])
    return handles
