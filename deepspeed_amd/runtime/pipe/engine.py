"""PipelineEngine: executes 1F1B schedules over a PipelineModule.

Parity: reference `runtime/pipe/engine.py:60` (PipelineEngine),
`train_batch:341`, `eval_batch:431`, instruction executors @722-1244,
`_exec_schedule:1396`, `_aggregate_total_loss:596`.
"""
import torch

from ... import comm as dist
from ...utils.logging import log_dist
from ..engine import DeepSpeedEngine
from . import p2p, schedule
from .module import PipelineModule


class PipelineEngine(DeepSpeedEngine):
    def __init__(self, *args, model=None, **kwargs):
        assert isinstance(model, PipelineModule)
        super().__init__(*args, model=model, **kwargs)
        self.grid = model.mpu()
        self.stage_id = self.grid.get_stage_id()
        self.num_stages = self.grid.pipe_parallel_size
        self.is_first_stage = self.grid.is_first_stage()
        self.is_last_stage = self.grid.is_last_stage()
        self.micro_batches = self.gradient_accumulation_steps()

        self.prev_rank = (self.grid.stage_to_global(self.stage_id - 1)
                          if not self.is_first_stage else None)
        self.next_rank = (self.grid.stage_to_global(self.stage_id + 1)
                          if not self.is_last_stage else None)

        self._act_meta_sent = False
        self._recv_act_template = None
        self._recv_grad_template = None
        self.agg_train_loss = None
        self._INSTRUCTION_MAP = {
            schedule.LoadMicroBatch: self._exec_load_micro_batch,
            schedule.ForwardPass: self._exec_forward_pass,
            schedule.BackwardPass: self._exec_backward_pass,
            schedule.SendActivation: self._exec_send_activations,
            schedule.RecvActivation: self._exec_recv_activations,
            schedule.SendGrad: self._exec_send_grads,
            schedule.RecvGrad: self._exec_recv_grads,
            schedule.ReduceGrads: self._exec_reduce_grads,
            schedule.OptimizerStep: self._exec_optimizer_step,
        }

    # ------------------------------------------------------------------
    def train_batch(self, data_iter=None):
        self.module.train()
        self._data_iter = data_iter
        M = self.micro_batches
        self._bufs = {"inputs": [None] * M, "outputs": [None] * M,
                      "labels": [None] * M, "losses": [None] * M,
                      "grads": [None] * M}
        if hasattr(self.optimizer, "set_accumulation_boundary"):
            self.optimizer.set_accumulation_boundary(False)
        sched = schedule.TrainSchedule(M, self.num_stages, self.stage_id)
        self._exec_schedule(sched)
        self.global_steps += 1
        self.global_samples += self.train_batch_size()
        agg = self._aggregate_total_loss()
        self.agg_train_loss = agg
        if self.global_steps % self.steps_per_print() == 0:
            log_dist(f"step={self.global_steps} loss={agg:.4f} "
                     f"lr={self.get_lr()}", ranks=[0])
        return agg

    @torch.no_grad()
    def eval_batch(self, data_iter, compute_loss=True):
        self.module.eval()
        self._data_iter = data_iter
        M = self.micro_batches
        self._bufs = {"inputs": [None] * M, "outputs": [None] * M,
                      "labels": [None] * M, "losses": [None] * M,
                      "grads": [None] * M}
        sched = schedule.InferenceSchedule(M, self.num_stages, self.stage_id)
        self._exec_schedule(sched)
        return self._aggregate_total_loss()

    def _exec_schedule(self, sched):
        self._n_backwards = 0
        for step_cmds in sched:
            for cmd in step_cmds:
                self._INSTRUCTION_MAP[type(cmd)](cmd)

    # ----------------------------------------------------- executors
    def _exec_load_micro_batch(self, cmd):
        batch = next(self._data_iter)
        if isinstance(batch, (tuple, list)):
            inputs, labels = batch[0], batch[1]
        else:
            inputs, labels = batch, None
        if self.is_first_stage:
            x = inputs.to(self.device)
            dt = self._config.dtype
            if x.is_floating_point() and \
                    dt in (torch.float16, torch.bfloat16):
                x = x.to(dt)
            self._bufs["inputs"][cmd.buffer_id] = x
        if self.is_last_stage and labels is not None:
            self._bufs["labels"][cmd.buffer_id] = labels.to(self.device)

    def _exec_forward_pass(self, cmd):
        m = cmd.buffer_id
        x = self._bufs["inputs"][m]
        out = self.module(x)
        if self.is_last_stage:
            if self.module.loss_fn is not None and \
                    self._bufs["labels"][m] is not None:
                loss = self.module.loss_fn(out, self._bufs["labels"][m])
            else:
                loss = out.float().mean() if out.requires_grad else out
            self._bufs["losses"][m] = loss
        else:
            self._bufs["outputs"][m] = out

    def _exec_backward_pass(self, cmd):
        b = cmd.buffer_id
        self._n_backwards += 1
        is_boundary = self._n_backwards == self.micro_batches
        if getattr(self.module, "tied_comms", None):
            # Tied-weight grads must be allreduced across stages BEFORE the
            # ZeRO bucket reduce consumes the grad16 views; keep the boundary
            # off during the final backward and flush every bucket in
            # _exec_reduce_grads after allreduce_tied_weight_gradients()
            # (ref pipe/engine.py _exec_reduce_tied_grads ordering).
            is_boundary = False
        if hasattr(self.optimizer, "set_accumulation_boundary"):
            self.optimizer.set_accumulation_boundary(is_boundary)
        if hasattr(self.optimizer, "ensure_grad_views"):
            self.optimizer.ensure_grad_views()
        if self.is_last_stage:
            loss = self._bufs["losses"][b] / self.micro_batches
            # fp16: scale the loss so grads stay in half range; the
            # partitioned optimizer folds 1/scale into the fused step
            scaler = getattr(self.optimizer, "loss_scaler", None)
            ls = getattr(scaler, "loss_scale", 1.0) if scaler else 1.0
            if ls != 1.0:
                (loss.float() * ls).backward()
            else:
                loss.backward()
        else:
            out = self._bufs["outputs"][b]
            grad = self._bufs["grads"][b]
            torch.autograd.backward(tensors=(out,), grad_tensors=(grad,))
            self._bufs["outputs"][b] = None
            self._bufs["grads"][b] = None

    def _exec_send_activations(self, cmd):
        m = cmd.buffer_id
        out = self._bufs["outputs"][m]
        if not self._act_meta_sent:
            p2p.send_meta(out, self.next_rank)
            self._act_meta_sent = True
        p2p.isend(out.detach(), self.next_rank)

    def _exec_recv_activations(self, cmd):
        m = cmd.buffer_id
        if self._recv_act_template is None:
            dtype, shape = p2p.recv_meta(self.prev_rank)
            self._recv_act_template = (dtype, shape)
        dtype, shape = self._recv_act_template
        buf = torch.empty(shape, dtype=dtype, device=self.device)
        p2p.recv(buf, self.prev_rank)
        buf.requires_grad_(True)
        self._bufs["inputs"][m] = buf

    def _exec_send_grads(self, cmd):
        b = cmd.buffer_id
        x = self._bufs["inputs"][b]
        assert x.grad is not None, "input grad missing after backward"
        p2p.isend(x.grad, self.prev_rank)
        self._bufs["inputs"][b] = None

    def _exec_recv_grads(self, cmd):
        b = cmd.buffer_id
        out = self._bufs["outputs"][b]
        buf = torch.empty_like(out)
        p2p.recv(buf, self.next_rank)
        self._bufs["grads"][b] = buf

    def _exec_reduce_grads(self, cmd):
        if getattr(self.module, "tied_comms", None):
            self.module.allreduce_tied_weight_gradients()
        if hasattr(self.optimizer, "set_accumulation_boundary"):
            self.optimizer.set_accumulation_boundary(True)
        if hasattr(self.optimizer, "reduce_gradients"):
            self.optimizer.reduce_gradients()
        elif self.dp_world_size > 1:
            self.allreduce_gradients()

    def _exec_optimizer_step(self, cmd):
        self.optimizer.step()
        if isinstance(self.optimizer, torch.optim.Optimizer):
            self.optimizer.zero_grad(set_to_none=True)
        if self.lr_scheduler is not None:
            self.lr_scheduler.step()
        p2p.drain()

    # ----------------------------------------------------- loss agg
    def _aggregate_total_loss(self):
        device = self.device
        if self.is_last_stage:
            losses = [l for l in self._bufs["losses"] if l is not None]
            agg = torch.stack([l.detach().float() for l in losses]).mean() \
                if losses else torch.zeros(1, device=device).squeeze()
            agg = agg.to(device)
            if self.grid.get_data_parallel_world_size() > 1:
                dist.all_reduce(agg, group=self.grid.get_data_parallel_group())
                agg = agg / self.grid.get_data_parallel_world_size()
        else:
            agg = torch.zeros((), device=device, dtype=torch.float32)
        src = self.grid.stage_to_global(self.num_stages - 1)
        dist.broadcast(agg, src, group=self.grid.get_pipe_parallel_group())
        return agg.item()

    def forward(self, *args, **kwargs):
        raise RuntimeError("PipelineEngine: use train_batch()/eval_batch()")

    def backward(self, *args, **kwargs):
        raise RuntimeError("PipelineEngine: use train_batch()")

    def step(self, *args, **kwargs):
        raise RuntimeError("PipelineEngine: use train_batch()")
