"""Local expert bank (ref deepspeed/moe/experts.py:13)."""
import copy

import torch


class Experts(torch.nn.Module):
    def __init__(self, expert, num_local_experts=1, expert_group_name=None):
        super().__init__()
        self.deepspeed_experts = torch.nn.ModuleList(
            [copy.deepcopy(expert) for _ in range(num_local_experts)])
        self.num_local_experts = num_local_experts
        for exp in self.deepspeed_experts:
            for p in exp.parameters():
                p.allreduce = False
                p.group_name = expert_group_name

    def forward(self, inputs):
        """inputs [ep_size, local_experts, C, M] ->
        outputs same shape (expert e processes [:, e])."""
        chunks = inputs.chunk(self.num_local_experts, dim=1)
        outputs = []
        for chunk, expert in zip(chunks, self.deepspeed_experts):
            out = expert(chunk)
            if isinstance(out, tuple):
                out = out[0]
            outputs.append(out)
        return torch.cat(outputs, dim=1)
