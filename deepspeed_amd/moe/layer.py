"""MoE user-facing layer (ref deepspeed/moe/layer.py:17)."""
import torch

from .. import comm as dist
from ..comm import groups
from ..utils.logging import log_dist
from .experts import Experts
from .sharded_moe import MOELayer, TopKGate


class MoE(torch.nn.Module):
    def __init__(self, hidden_size, expert, num_experts=1, ep_size=1,
                 k=1, capacity_factor=1.0, eval_capacity_factor=1.0,
                 min_capacity=4, drop_tokens=True, use_residual=False,
                 **kwargs):
        super().__init__()
        self.use_residual = use_residual
        assert num_experts % ep_size == 0, \
            f"experts {num_experts} % ep_size {ep_size} != 0"
        self.ep_size = ep_size
        self.num_experts = num_experts
        self.num_local_experts = num_experts // ep_size
        self.expert_group_name = f"ep_size_{ep_size}"

        gate = TopKGate(hidden_size, num_experts, k, capacity_factor,
                        eval_capacity_factor, min_capacity, drop_tokens)
        experts = Experts(expert, self.num_local_experts,
                          self.expert_group_name)
        self.deepspeed_moe = MOELayer(gate, experts, self.expert_group_name,
                                      self.ep_size, self.num_local_experts)
        if use_residual:
            self.mlp = expert
            self.coefficient = torch.nn.Linear(hidden_size, 2)
        self._groups_ready = False

    def _ensure_groups(self):
        if self._groups_ready:
            return
        if dist.is_initialized() and self.ep_size > 1:
            if self.expert_group_name not in \
                    groups.get_expert_parallel_group_dict():
                groups.create_expert_and_data_parallel(self.ep_size)
            self.deepspeed_moe.set_ep_group(
                groups.get_expert_parallel_group(self.expert_group_name))
        self._groups_ready = True

    def forward(self, hidden_states, used_token=None):
        self._ensure_groups()
        output = self.deepspeed_moe(hidden_states)
        if self.use_residual:
            res = self.mlp(hidden_states)
            if isinstance(res, tuple):
                res = res[0]
            coef = torch.nn.functional.softmax(
                self.coefficient(hidden_states), dim=-1)
            output = output * coef[..., 0:1] + res * coef[..., 1:2]
        return output, self.deepspeed_moe.l_aux, \
            self.deepspeed_moe.exp_counts
