"""MoE optimizer-group helpers.

Parity: reference `deepspeed/moe/utils.py`
(`split_params_into_different_moe_groups_for_optimizer:116`,
`is_moe_param:14`): expert parameters must live in their own optimizer
param groups (tagged with the expert group name) so ZeRO builds
expert-DP buckets/sub-groups for them.
"""
from collections import defaultdict

import torch


def is_moe_param(param) -> bool:
    return getattr(param, "group_name", None) is not None and \
        not getattr(param, "allreduce", True)


def has_moe_layers(module) -> bool:
    return any(is_moe_param(p) for p in module.parameters())


def split_params_into_different_moe_groups_for_optimizer(
        param_groups, max_group_size=None):
    """Split each param group into {dense} + one group per expert set.

    Accepts a dict, a list of dicts, or a tuple of dicts (the formats
    torch optimizers take); returns a list of dicts where expert groups
    carry `moe=True` and `name=<expert group>`.
    """
    if isinstance(param_groups, dict):
        param_groups = [param_groups]
    elif isinstance(param_groups, tuple):
        param_groups = list(param_groups)
    out = []
    for group in param_groups:
        dense = []
        experts = defaultdict(list)
        for p in group["params"]:
            if is_moe_param(p):
                experts[p.group_name].append(p)
            else:
                dense.append(p)
        g = dict(group)
        g["params"] = dense
        out.append(g)
        for name, ps in experts.items():
            if max_group_size is not None:
                for i in range(0, len(ps), max_group_size):
                    eg = dict(group)
                    eg.update(params=ps[i:i + max_group_size], moe=True,
                              name=name)
                    out.append(eg)
            else:
                eg = dict(group)
                eg.update(params=ps, moe=True, name=name)
                out.append(eg)
    return out
