"""BF16_Optimizer — bf16 params with DP-partitioned fp32 masters.

Parity: reference `deepspeed/runtime/bf16_optimizer.py` (BF16_Optimizer:
fp32 flat master partitions + fp32 grad accumulation for bf16 training
without ZeRO, used notably under pipeline parallelism).

MI355X-native design: the reference implements this as its own flat-
tensor optimizer; here it IS the stage-1 partitioned optimizer
(`ZeroStage12Optimizer`) — that class already keeps one fp32 master
shard per rank, accumulates grads into persistent fp32 buffers when
`immediate_grad_update` semantics require it, reduces with RCCL
PreMulSum averaging and rewrites bf16 shards via the fused-Adam out16
path. This wrapper pins the reference's name, constructor surface and
the attribute names tools poke at (`fp32_groups_flat_partition`,
`get_grads_for_reduction`, `update_hp_grads`), so code written against
the reference's BF16_Optimizer keeps working.
"""
import torch

from .zero.stage_1_and_2 import ZeroStage12Optimizer


class BF16_Optimizer(ZeroStage12Optimizer):
    def __init__(self, init_optimizer, param_names=None, mpu=None,
                 clip_grad=0.0, allgather_bucket_size=int(5e8),
                 dp_process_group=None, timers=None, grad_acc_dtype=None,
                 graph_harvesting=False, immediate_grad_update=True,
                 has_moe_layers=False, **kw):
        if grad_acc_dtype not in (None, torch.float32):
            raise ValueError("BF16_Optimizer accumulates grads in fp32")
        super().__init__(init_optimizer, stage=1,
                         dp_process_group=dp_process_group,
                         allgather_bucket_size=allgather_bucket_size,
                         clip_grad=clip_grad, dtype=torch.bfloat16,
                         mpu=mpu, **kw)

    # ---- reference attribute surface ------------------------------------
    @property
    def fp32_groups_flat_partition(self):
        return [b.master32 for b in self.buckets]

    def get_grads_for_reduction(self):
        return [b.grad16 for b in self.buckets]

    def update_hp_grads(self, clear_lp_grads=False):
        # grads flow into the flat fp32 buffers at the accumulation
        # boundary inside backward(); nothing to do eagerly here.
        if clear_lp_grads:
            self.zero_grad(set_to_none=False)

    @torch.no_grad()
    def update_lp_params(self):
        from .. import comm as dist
        for b in self.buckets:
            b.shard16.copy_(b.master32.detach().to(b.shard16.dtype))
            dist.all_gather_into_tensor(b.flat16, b.shard16,
                                        group=self.dp_group)
