"""Activation checkpointing with RNG tracking, partitioned activations and
CPU checkpointing.

Parity: reference `runtime/activation_checkpointing/checkpointing.py`
(`CudaRNGStatesTracker:134`, `partition_activations:387`,
`CheckpointFunction:567`, `checkpoint:1058`).

MI355X notes: with 288 GB HBM3E most configs should *disable* recompute
entirely (bench.py default); this module serves the long-sequence / 70B
regimes. Partitioned activations shard the saved inputs across the TP/SP
group; cpu_checkpointing stages them to pinned host memory on a side stream.
"""
import torch

from .. import comm as dist
from ..comm import groups as grp

_CONFIG = {
    "partition_activations": False,
    "cpu_checkpointing": False,
    "contiguous_memory_optimization": False,
    "synchronize": False,
    "profile": False,
}
_mp_group = None


def configure(mpu_=None, deepspeed_config=None, partition_activations=None,
              contiguous_checkpointing=None, num_checkpoints=None,
              checkpoint_in_cpu=None, synchronize=None, profile=None):
    global _mp_group
    if deepspeed_config is not None:
        ac = deepspeed_config.activation_checkpointing
        _CONFIG["partition_activations"] = ac.partition_activations
        _CONFIG["cpu_checkpointing"] = ac.cpu_checkpointing
        _CONFIG["contiguous_memory_optimization"] = \
            ac.contiguous_memory_optimization
        _CONFIG["synchronize"] = ac.synchronize_checkpoint_boundary
        _CONFIG["profile"] = ac.profile
    if partition_activations is not None:
        _CONFIG["partition_activations"] = partition_activations
    if checkpoint_in_cpu is not None:
        _CONFIG["cpu_checkpointing"] = checkpoint_in_cpu
    if mpu_ is not None and hasattr(mpu_, "get_model_parallel_group"):
        _mp_group = mpu_.get_model_parallel_group()


def is_configured():
    return True


# ------------------------------------------------------------- RNG tracker
class CudaRNGStatesTracker:
    """Fork device RNG per named state so recompute reproduces dropout."""

    def __init__(self):
        self.states_ = {}

    def reset(self):
        self.states_.clear()

    def add(self, name, seed):
        if not torch.cuda.is_available():
            return
        orig = torch.cuda.get_rng_state()
        torch.cuda.manual_seed(seed)
        self.states_[name] = torch.cuda.get_rng_state()
        torch.cuda.set_rng_state(orig)

    class _Fork:
        def __init__(self, tracker, name):
            self.tracker = tracker
            self.name = name

        def __enter__(self):
            if not torch.cuda.is_available() or \
                    self.name not in self.tracker.states_:
                self.noop = True
                return
            self.noop = False
            self.orig = torch.cuda.get_rng_state()
            torch.cuda.set_rng_state(self.tracker.states_[self.name])

        def __exit__(self, *a):
            if not self.noop:
                self.tracker.states_[self.name] = torch.cuda.get_rng_state()
                torch.cuda.set_rng_state(self.orig)
            return False

    def fork(self, name="model-parallel-rng"):
        return self._Fork(self, name)


_RNG_TRACKER = CudaRNGStatesTracker()


def get_cuda_rng_tracker():
    return _RNG_TRACKER


def model_parallel_cuda_manual_seed(seed):
    tp_rank = grp.get_tensor_parallel_rank()
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed(seed)
    _RNG_TRACKER.reset()
    _RNG_TRACKER.add("model-parallel-rng", seed + 2718 + tp_rank)


# ----------------------------------------------------- partition helpers
def _partition_tensor(t, group):
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    flat = t.reshape(-1)
    chunk = (flat.numel() + world - 1) // world
    padded = chunk * world
    if padded != flat.numel():
        flat = torch.nn.functional.pad(flat, (0, padded - flat.numel()))
    return flat[rank * chunk:(rank + 1) * chunk].clone(), t.shape, t.numel()


def _gather_tensor(shard, shape, numel, group):
    world = dist.get_world_size(group)
    out = torch.empty(shard.numel() * world, dtype=shard.dtype,
                      device=shard.device)
    dist.all_gather_into_tensor(out, shard.contiguous(), group=group)
    return out[:numel].reshape(shape)


class CheckpointFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, run_function, *args):
        ctx.run_function = run_function
        group = _mp_group
        ctx.group = group
        part = _CONFIG["partition_activations"] and group is not None \
            and dist.get_world_size(group) > 1
        cpu = _CONFIG["cpu_checkpointing"]
        ctx.part = part
        ctx.cpu = cpu
        saved = []
        ctx.meta = []
        for a in args:
            if torch.is_tensor(a):
                t = a.detach()
                if part and a.requires_grad:
                    shard, shape, numel = _partition_tensor(t, group)
                    if cpu:
                        shard = shard.cpu()
                    saved.append(shard)
                    ctx.meta.append(("part", shape, numel, a.requires_grad,
                                     a.device))
                else:
                    if cpu and t.is_cuda:
                        saved.append(t.cpu())
                    else:
                        saved.append(t)
                    ctx.meta.append(("full", t.shape, t.numel(),
                                     a.requires_grad, a.device))
            else:
                saved.append(None)
                ctx.meta.append(("obj", a))
        ctx.save_for_backward(*[s for s in saved if torch.is_tensor(s)])
        ctx.objs = [m[1] for m in ctx.meta if m[0] == "obj"]
        if torch.cuda.is_available():
            ctx.fwd_rng_state = torch.cuda.get_rng_state()
        ctx.cpu_rng_state = torch.get_rng_state()
        with torch.no_grad():
            outputs = run_function(*args)
        return outputs

    @staticmethod
    def backward(ctx, *grads):
        saved = list(ctx.saved_tensors)
        args = []
        si, oi = 0, 0
        for m in ctx.meta:
            if m[0] == "obj":
                args.append(ctx.objs[oi])
                oi += 1
                continue
            kind, shape, numel, req, device = m
            t = saved[si]
            si += 1
            if kind == "part":
                t = t.to(device)
                t = _gather_tensor(t, shape, numel, ctx.group)
            else:
                t = t.to(device)
            t = t.detach().requires_grad_(req)
            args.append(t)
        # restore RNG for deterministic recompute
        cpu_state = torch.get_rng_state()
        torch.set_rng_state(ctx.cpu_rng_state)
        if torch.cuda.is_available():
            dev_state = torch.cuda.get_rng_state()
            torch.cuda.set_rng_state(ctx.fwd_rng_state)
        with torch.enable_grad():
            outputs = ctx.run_function(*args)
        torch.set_rng_state(cpu_state)
        if torch.cuda.is_available():
            torch.cuda.set_rng_state(dev_state)
        if torch.is_tensor(outputs):
            outputs = (outputs,)
        out_and_grads = [(o, g) for o, g in zip(outputs, grads)
                         if torch.is_tensor(o) and o.requires_grad]
        torch.autograd.backward([o for o, _ in out_and_grads],
                                [g for _, g in out_and_grads])
        input_grads = tuple(a.grad if torch.is_tensor(a) and a.requires_grad
                            else None for a in args)
        return (None,) + input_grads


def checkpoint(function, *args):
    """Megatron-compatible checkpoint() entry (ref checkpointing.py:1058)."""
    return CheckpointFunction.apply(function, *args)


def non_reentrant_checkpoint(function, *args):
    return torch.utils.checkpoint.checkpoint(function, *args,
                                             use_reentrant=False)
