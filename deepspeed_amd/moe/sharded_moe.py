"""GShard-style top-k gated MoE with expert-parallel all-to-all.

Parity: reference `deepspeed/moe/sharded_moe.py` (`_AllToAll:97`,
`top1gating:235`, `topkgating:434`, `TopKGate:528`, `MOELayer:618`).

MI355X note: the dispatch/combine all-to-alls are intra-node over
fully-connected xGMI (EP groups are adjacent ranks, comm/groups.py) — the
preferred collective shape for this fabric.
"""
import torch
import torch.nn.functional as F
from torch import Tensor

from .. import comm as dist


class _AllToAll(torch.autograd.Function):
    @staticmethod
    def forward(ctx, group, input_):
        ctx.group = group
        if dist.get_world_size(group) == 1:
            return input_
        input_ = input_.contiguous()
        output = torch.empty_like(input_)
        dist.all_to_all_single(output, input_, group=group)
        return output

    @staticmethod
    def backward(ctx, grad):
        return None, _AllToAll.apply(ctx.group, grad)


@torch.jit.script
def _capacity(gates: Tensor, capacity_factor: float, min_capacity: int) -> int:
    num_tokens = gates.shape[0]
    num_experts = gates.shape[1]
    capacity = int(capacity_factor * num_tokens / num_experts)
    if capacity < min_capacity:
        capacity = min_capacity
    return capacity


def topkgating(logits, k, capacity_factor, min_capacity=4, drop_tokens=True):
    """Returns (l_aux, combine_weights [S,E,C], dispatch_mask [S,E,C], C)."""
    gates = F.softmax(logits, dim=1)
    num_tokens, num_experts = gates.shape
    capacity = _capacity(gates, capacity_factor * k, min_capacity)

    topk_vals, topk_idx = torch.topk(gates, k, dim=1)  # [S, k]
    # build mask [S, E]: 1 where expert selected
    mask = torch.zeros_like(gates)
    mask.scatter_(1, topk_idx, 1.0)

    # load-balancing aux loss (GShard): E * sum_e mean(gates_e)*mean(mask_e)
    me = gates.mean(dim=0)
    ce = mask.float().mean(dim=0)
    l_aux = torch.sum(me * ce) * num_experts * num_experts / k

    # position of each token within its expert queue (by token order)
    locations = torch.cumsum(mask, dim=0) - 1  # [S, E]
    if drop_tokens:
        mask = mask * (locations < capacity).float()
    locations = (locations * mask).long()

    # renormalize selected gate values
    gates_masked = gates * mask
    denom = gates_masked.sum(dim=1, keepdim=True).clamp(min=1e-9)
    gates_norm = gates_masked / denom

    # combine weights [S, E, C]
    loc_onehot = F.one_hot(locations, num_classes=capacity).to(gates.dtype)
    combine = gates_norm.unsqueeze(-1) * loc_onehot * mask.unsqueeze(-1)
    dispatch = combine.bool()
    return l_aux, combine, dispatch, capacity


def top1gating(logits, capacity_factor, min_capacity=4, drop_tokens=True):
    return topkgating(logits, 1, capacity_factor, min_capacity, drop_tokens)


class TopKGate(torch.nn.Module):
    """Gate module (ref sharded_moe.py:528). Keeps wg in fp32."""

    def __init__(self, model_dim, num_experts, k=1, capacity_factor=1.0,
                 eval_capacity_factor=1.0, min_capacity=4, drop_tokens=True,
                 **kwargs):
        super().__init__()
        self.wg = torch.nn.Linear(model_dim, num_experts, bias=False)
        self.k = k
        self.capacity_factor = capacity_factor
        self.eval_capacity_factor = eval_capacity_factor
        self.min_capacity = min_capacity
        self.drop_tokens = drop_tokens

    def forward(self, input_):
        logits = self.wg(input_.float() if self.wg.weight.dtype ==
                         torch.float32 else input_)
        cf = self.capacity_factor if self.training \
            else self.eval_capacity_factor
        return topkgating(logits.float(), self.k, cf, self.min_capacity,
                          self.drop_tokens)


class MOELayer(torch.nn.Module):
    """gate -> dispatch (einsum) -> a2a -> experts -> a2a -> combine.

    Parity: ref MOELayer fwd @ sharded_moe.py:668.
    """

    def __init__(self, gate, experts, ep_group_name, ep_size,
                 num_local_experts):
        super().__init__()
        self.gate = gate
        self.experts = experts
        self.ep_group = None
        self.ep_group_name = ep_group_name
        self.ep_size = ep_size
        self.num_local_experts = num_local_experts
        self.l_aux = torch.tensor(0.0)
        self.exp_counts = None

    def set_ep_group(self, group):
        self.ep_group = group

    def forward(self, input_):
        d_model = input_.shape[-1]
        reshaped = input_.reshape(-1, d_model)
        self.l_aux, combine, dispatch, C = self.gate(reshaped)
        E = combine.shape[1]
        dispatched = torch.einsum("sec,sm->ecm",
                                  dispatch.to(input_.dtype), reshaped)
        if self.ep_size > 1:
            dispatched = _AllToAll.apply(self.ep_group, dispatched)
        # [E, C, M] -> [ep_size, local_experts, C, M]
        dispatched = dispatched.reshape(self.ep_size, self.num_local_experts,
                                        C, d_model)
        expert_out = self.experts(dispatched)
        expert_out = expert_out.reshape(E, C, d_model)
        if self.ep_size > 1:
            expert_out = _AllToAll.apply(self.ep_group, expert_out)
        combined = torch.einsum("sec,ecm->sm", combine.to(input_.dtype),
                                expert_out)
        return combined.reshape(input_.shape)
