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


def topkgating_indices(logits, k, capacity_factor, min_capacity=4,
                       drop_tokens=True, ep_group=None):
    """Index-form top-k gating: O(S*k) dispatch metadata instead of the
    [S, E, C] one-hot tensors (ref MOELayer fast paths, sharded_moe.py:618;
    at Mixtral scale the dense combine tensor would be S*E*C elements).

    Returns (l_aux, topk_idx [S,k], weights [S,k] fp32, locations [S,k],
    keep [S,k] bool, capacity).
    """
    gates = F.softmax(logits, dim=1)
    num_tokens, num_experts = gates.shape
    capacity = _capacity(gates, capacity_factor * k, min_capacity)

    topk_vals, topk_idx = torch.topk(gates, k, dim=1)  # [S, k]
    mask = torch.zeros_like(gates)
    mask.scatter_(1, topk_idx, 1.0)

    me = gates.mean(dim=0)
    ce = mask.float().mean(dim=0)
    l_aux = torch.sum(me * ce) * num_experts * num_experts / k

    # position of each token within its expert queue (token order)
    locations = (torch.cumsum(mask, dim=0) - 1)
    loc_sj = locations.gather(1, topk_idx).long()  # [S, k]

    if drop_tokens:
        keep = loc_sj < capacity
    else:
        # capacity grows to the longest queue; must agree across the EP
        # group so the all-to-all shapes match (ref top2gating no-drop)
        local_max = int(loc_sj.max().item()) + 1 if num_tokens else 1
        if ep_group is not None and dist.get_world_size(ep_group) > 1:
            t = torch.tensor([max(capacity, local_max)],
                             device=logits.device)
            dist.all_reduce(t, op=dist.ReduceOp.MAX, group=ep_group)
            capacity = int(t.item())
        else:
            capacity = max(capacity, local_max)
        keep = torch.ones_like(loc_sj, dtype=torch.bool)

    vals = topk_vals * keep  # drop, then renormalize over kept experts
    denom = vals.sum(dim=1, keepdim=True).clamp(min=1e-9)
    weights = vals / denom
    return l_aux, topk_idx, weights, loc_sj, keep, capacity


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
        S = reshaped.shape[0]
        gate = self.gate
        logits = gate.wg(reshaped.float() if gate.wg.weight.dtype ==
                         torch.float32 else reshaped).float()
        cf = gate.capacity_factor if self.training \
            else gate.eval_capacity_factor
        self.l_aux, idx, w, loc, keep, C = topkgating_indices(
            logits, gate.k, cf, gate.min_capacity, gate.drop_tokens,
            ep_group=self.ep_group if self.ep_size > 1 else None)
        E = logits.shape[1]
        self.exp_counts = keep.sum(0) if gate.k > 1 else None

        # index dispatch: each kept (token, choice) owns one unique slot
        # e*C + loc in the flat [E*C, M] buffer — O(S*k) work/memory, no
        # [S, E, C] one-hot tensors
        flat_pos = idx * C + loc                      # [S, k]
        keep_f = keep.reshape(-1)
        kept_pos = flat_pos.reshape(-1)[keep_f]       # [N]
        token_idx = torch.arange(S, device=reshaped.device) \
            .unsqueeze(1).expand(S, gate.k).reshape(-1)[keep_f]
        dispatched = reshaped.new_zeros(E * C, d_model).index_copy(
            0, kept_pos, reshaped.index_select(0, token_idx))
        dispatched = dispatched.reshape(E, C, d_model)

        if self.ep_size > 1:
            dispatched = _AllToAll.apply(self.ep_group, dispatched)
        # [E, C, M] -> [ep_size, local_experts, C, M]
        dispatched = dispatched.reshape(self.ep_size, self.num_local_experts,
                                        C, d_model)
        expert_out = self.experts(dispatched)
        expert_out = expert_out.reshape(E, C, d_model)
        if self.ep_size > 1:
            expert_out = _AllToAll.apply(self.ep_group, expert_out)

        # index combine: gather each choice's expert output row and take
        # the gate-weighted sum over k (dropped choices carry weight 0)
        expert_flat = expert_out.reshape(E * C, d_model)
        safe_pos = flat_pos.reshape(-1).clamp_(max=E * C - 1)
        gathered = expert_flat.index_select(0, safe_pos) \
            .reshape(S, gate.k, d_model)
        wk = (w * keep).to(gathered.dtype).unsqueeze(-1)
        combined = (wk * gathered).sum(dim=1)
        return combined.reshape(input_.shape)
