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
