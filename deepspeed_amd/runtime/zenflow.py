"""ZenFlow — importance-aware selective optimizer offload for ZeRO-1/2.

Parity: reference `deepspeed/runtime/zenflow/zenflow_stage_1_and_2.py`
(+ `zenflow_config.py`, `ops/adam/zenflow_cpu_adam.py`): the fp32
optimizer state lives on the host; every step the top-k most important
gradient channels are updated IMMEDIATELY on the GPU (hot set), while
the remaining gradients accumulate on the host and are applied by a
batched CPU-Adam pass every `update_interval` steps. This keeps the
update path off the critical step for ~99% of the state while the loss
trajectory tracks dense Adam closely.

MI355X-native design: the unit of selection is the flat fp32 grad shard
of a stage-1/2 bucket (already reduce-scattered over RCCL, so importance
and hot sets are rank-local — no extra comm). The hot set's master/m/v
vectors are cached on the GPU between re-selections; the lazy pass is
the OpenMP `cpu_adam_step` on the pinned host slabs, and the bf16 shard
is refreshed with one H2D per bucket only on lazy boundaries.
"""
from typing import Optional

import threading

import torch
from pydantic import BaseModel

from .. import comm as dist
from ..utils.logging import log_dist
from .zero.stage_1_and_2 import ZeroStage12Optimizer
from .zero.stage3 import ZeroStage3Optimizer


class ZenFlowConfig(BaseModel):
    topk_ratio: float = 0.01
    update_interval: int = 4
    select_interval: int = 16    # re-pick the hot set every N steps
    select_strategy: str = "auto"
    full_warm_up_rounds: int = 0
    overlap_step: bool = False   # run the lazy CPU Adam in a worker thread


class ZenFlowZeroOptimizer(ZeroStage12Optimizer):
    def __init__(self, init_optimizer, zenflow_config: Optional[dict] = None,
                 **kw):
        super().__init__(init_optimizer, **kw)
        zf = zenflow_config or {}
        self.zf = zf if isinstance(zf, ZenFlowConfig) else ZenFlowConfig(**zf)
        self._zf_step = 0
        # host-resident optimizer state (pinned when CUDA is present);
        # Bucket is __slots__, so ZenFlow state lives in a side table
        pin = torch.cuda.is_available()
        self._zf = []
        for b in self.buckets:
            cpu = b.master32.detach().to("cpu")
            st = dict(master_cpu=cpu.pin_memory() if pin else cpu,
                      acc_steps=0, hot_idx=None, hot_master=None,
                      hot_m=None, hot_v=None)
            st["m_cpu"] = torch.zeros_like(st["master_cpu"])
            st["v_cpu"] = torch.zeros_like(st["master_cpu"])
            st["acc_cpu"] = torch.zeros_like(st["master_cpu"])
            st["acc_snap"] = torch.zeros_like(st["master_cpu"]) \
                if self.zf.overlap_step else None
            st["thread"] = None
            self._zf.append(st)
            # the GPU master slab is not used by ZenFlow's dense pass
            b.master32 = None
        # drop the wrapped optimizer's references so the GPU slabs free
        for group in self.optimizer.param_groups:
            group["params"] = []
        log_dist(f"ZenFlow: topk={self.zf.topk_ratio}, "
                 f"update_interval={self.zf.update_interval}", ranks=[0])

    # -- selection --------------------------------------------------------
    def _writeback_hot(self, st):
        if st["hot_idx"] is None:
            return
        idx = st["hot_idx"].cpu()
        st["master_cpu"][idx] = st["hot_master"].cpu()
        st["m_cpu"][idx] = st["hot_m"].cpu()
        st["v_cpu"][idx] = st["hot_v"].cpu()

    def _select_hot(self, st, g):
        k = max(1, int(self.zf.topk_ratio * g.numel()))
        self._writeback_hot(st)
        st["hot_idx"] = g.abs().topk(k).indices
        idx_cpu = st["hot_idx"].cpu()
        dev = g.device
        st["hot_master"] = st["master_cpu"][idx_cpu].to(dev,
                                                        non_blocking=True)
        st["hot_m"] = st["m_cpu"][idx_cpu].to(dev, non_blocking=True)
        st["hot_v"] = st["v_cpu"][idx_cpu].to(dev, non_blocking=True)

    # -- step -------------------------------------------------------------
    @torch.no_grad()
    def step(self, closure=None):
        assert closure is None, "closure not supported"
        self._sync_comm()
        if self.dtype == torch.float16:
            self.overflow = self.has_overflow()
            self.loss_scaler.update_scale(self.overflow)
            if self.overflow:
                log_dist("ZenFlow: OVERFLOW, skipping step", ranks=[0])
                self._clear_grads()
                return
        self._zf_step += 1
        combined = self._combined_scale()
        group = self.optimizer.param_groups[0]
        lr = group["lr"]
        beta1, beta2 = group.get("betas", (0.9, 0.999))
        eps = group.get("eps", 1e-8)
        wd = group.get("weight_decay", 0.0)
        t = self._zf_step
        bc1 = 1 - beta1 ** t
        bc2 = 1 - beta2 ** t

        lazy = (self._zf_step % self.zf.update_interval) == 0
        for b, st in zip(self.buckets, self._zf):
            # publish a finished overlapped CPU step before touching state
            if st["thread"] is not None and not st["thread"].is_alive():
                self._publish_cpu_step(b, st)
            g = b.grad32
            if combined != 1.0:
                g.mul_(1.0 / combined)
            if st["hot_idx"] is None or \
                    (self._zf_step % self.zf.select_interval) == 1:
                self._select_hot(st, g)
            idx = st["hot_idx"]
            # ---- hot channels: immediate GPU Adam -----------------------
            gh = g[idx]
            st["hot_m"].mul_(beta1).add_(gh, alpha=1 - beta1)
            st["hot_v"].mul_(beta2).addcmul_(gh, gh, value=1 - beta2)
            if wd:
                st["hot_master"].mul_(1 - lr * wd)
            denom = (st["hot_v"] / bc2).sqrt().add_(eps)
            st["hot_master"].addcdiv_(st["hot_m"] / bc1, denom, value=-lr)
            flat_shard = b.shard16.view(-1)
            flat_shard[idx] = st["hot_master"].to(b.shard16.dtype)
            # ---- cold channels: accumulate on host ----------------------
            g.index_fill_(0, idx, 0.0)
            st["acc_cpu"].add_(g.to("cpu"))
            st["acc_steps"] += 1
            if lazy:
                if st["thread"] is not None:  # previous overlap still busy
                    st["thread"].join()
                    self._publish_cpu_step(b, st)
                self._writeback_hot(st)
                idx_cpu = st["hot_idx"].cpu()
                hot_restore = (idx_cpu, st["master_cpu"][idx_cpu].clone(),
                               st["m_cpu"][idx_cpu].clone(),
                               st["v_cpu"][idx_cpu].clone())
                st["acc_cpu"].div_(max(st["acc_steps"], 1))
                if self.zf.overlap_step:
                    # hand the batch to a worker; accumulation continues
                    # into the other buffer, publication happens when the
                    # thread finishes (ZenFlow's bounded staleness)
                    st["acc_cpu"], st["acc_snap"] = (st["acc_snap"],
                                                     st["acc_cpu"])
                    args = (st["master_cpu"], st["acc_snap"], st["m_cpu"],
                            st["v_cpu"], lr, beta1, beta2, eps, t, wd,
                            bc1, bc2) + hot_restore
                    st["thread"] = threading.Thread(
                        target=self._host_adam, args=args, daemon=True)
                    st["thread"].start()
                else:
                    self._host_adam(st["master_cpu"], st["acc_cpu"],
                                    st["m_cpu"], st["v_cpu"], lr, beta1,
                                    beta2, eps, t, wd, bc1, bc2,
                                    *hot_restore)
                    st["acc_cpu"].zero_()
                    self._publish_cpu_step(b, st)
                st["acc_steps"] = 0

        for b in self.buckets:
            dist.all_gather_into_tensor(
                b.flat16, b.shard16,
                group=b.pg if b.pg is not None else self.dp_group)
        self._clear_grads()

    @staticmethod
    def _host_adam(master, grad, m, v, lr, beta1, beta2, eps, t, wd,
                   bc1, bc2, hot_idx=None, hot_master=None, hot_m=None,
                   hot_v=None):
        from ..ops.loader import has_ext, get_ext
        if has_ext():
            get_ext().cpu_adam_step(master, grad, m, v, None, lr, beta1,
                                    beta2, eps, t, 1, 1, wd, 1.0)
        else:
            m.mul_(beta1).add_(grad, alpha=1 - beta1)
            v.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
            master.mul_(1 - lr * wd)
            master.addcdiv_(m / bc1, (v / bc2).sqrt().add_(eps), value=-lr)
        if hot_idx is not None:
            # hot channels are owned by the immediate GPU path: the dense
            # host pass must not apply a second (decayed-momentum) update
            # to them — restore the written-back hot state
            master[hot_idx] = hot_master
            m[hot_idx] = hot_m
            v[hot_idx] = hot_v

    def _publish_cpu_step(self, b, st):
        """H2D the freshly updated masters + refresh the hot cache."""
        st["thread"] = None
        if st["acc_snap"] is not None:
            st["acc_snap"].zero_()
        b.shard16.copy_(
            st["master_cpu"].to(b.shard16.device, non_blocking=True)
            .to(b.shard16.dtype))
        if st["hot_idx"] is not None:
            idx_cpu = st["hot_idx"].cpu()
            dev = b.shard16.device
            st["hot_master"] = st["master_cpu"][idx_cpu].to(dev)
            st["hot_m"] = st["m_cpu"][idx_cpu].to(dev)
            st["hot_v"] = st["v_cpu"][idx_cpu].to(dev)

    def _drain_threads(self):
        for b, st in zip(self.buckets, self._zf):
            if st["thread"] is not None:
                st["thread"].join()
                self._publish_cpu_step(b, st)

    # checkpointing: host masters replace the GPU ones
    def state_dict(self):
        self._drain_threads()
        return {
            "zf_step": self._zf_step,
            "single_partition_of_fp32_groups":
                [st["master_cpu"] for st in self._zf],
            "m": [st["m_cpu"] for st in self._zf],
            "v": [st["v_cpu"] for st in self._zf],
            "loss_scaler": self.loss_scaler.state_dict()
            if hasattr(self.loss_scaler, "state_dict") else None,
        }

    def load_state_dict(self, sd):
        self._zf_step = sd.get("zf_step", 0)
        for b, st, mc, m, v in zip(self.buckets, self._zf,
                                   sd["single_partition_of_fp32_groups"],
                                   sd["m"], sd["v"]):
            st["master_cpu"].copy_(mc)
            st["m_cpu"].copy_(m)
            st["v_cpu"].copy_(v)
            st["hot_idx"] = None
            b.shard16.copy_(st["master_cpu"].to(b.shard16.device)
                            .to(b.shard16.dtype))
        for b in self.buckets:
            dist.all_gather_into_tensor(
                b.flat16, b.shard16,
                group=b.pg if b.pg is not None else self.dp_group)


class ZenFlowZeroStage3Optimizer(ZeroStage3Optimizer):
    """ZenFlow over ZeRO-3 (ref runtime/zenflow/engine_stage3.py): per
    sub-group, the top-k "hot" gradient channels update immediately on
    the GPU while the cold mass accumulates on the host and takes a lazy
    (optionally worker-threaded) CPU Adam step every update_interval
    steps. Masters/moments live on the host; the device keeps only the
    16-bit shard slab + the hot working set."""

    def __init__(self, init_optimizer, zenflow_config=None, **kw):
        super().__init__(init_optimizer, **kw)
        assert self.param_swapper is None, \
            "ZenFlow manages its own host state; combine it with " \
            "offload_param cpu, not the NVMe parameter tier"
        zf = zenflow_config or {}
        self.zf = zf if isinstance(zf, ZenFlowConfig) else ZenFlowConfig(**zf)
        self._zf_step = 0
        pin = torch.cuda.is_available()
        self._zf = []
        for sg in self.sub_groups:
            cpu = sg.master32.detach().to("cpu")
            st = dict(master_cpu=cpu.pin_memory() if pin else cpu,
                      acc_steps=0, hot_idx=None, hot_master=None,
                      hot_m=None, hot_v=None, thread=None)
            st["m_cpu"] = torch.zeros_like(st["master_cpu"])
            st["v_cpu"] = torch.zeros_like(st["master_cpu"])
            st["acc_cpu"] = torch.zeros_like(st["master_cpu"])
            st["acc_snap"] = torch.zeros_like(st["master_cpu"]) \
                if self.zf.overlap_step else None
            self._zf.append(st)
            # keep state_dict/load working: master32 now IS the host copy
            sg.master32 = st["master_cpu"]
        for group in self.optimizer.param_groups:
            group["params"] = []
        log_dist(f"ZenFlow(stage3): topk={self.zf.topk_ratio}, "
                 f"update_interval={self.zf.update_interval}", ranks=[0])

    @torch.no_grad()
    def step(self, closure=None):
        assert closure is None
        self._flush_ipg()
        self._sync_comm_streams()
        self._drain_inflight()
        self._cached_norm_sq = None
        if self.dtype == torch.float16:
            self.overflow = self.has_overflow()
            self.loss_scaler.update_scale(self.overflow)
            if self.overflow:
                log_dist("ZenFlow(stage3): OVERFLOW, skipping step",
                         ranks=[0])
                self._clear_grads()
                return
        self._zf_step += 1
        combined = self._combined_scale()
        group = self.optimizer.param_groups[0]
        lr = group["lr"]
        beta1, beta2 = group.get("betas", (0.9, 0.999))
        eps = group.get("eps", 1e-8)
        wd = group.get("weight_decay", 0.0)
        t = self._zf_step
        bc1 = 1 - beta1 ** t
        bc2 = 1 - beta2 ** t
        lazy = (self._zf_step % self.zf.update_interval) == 0

        for sg, st in zip(self.sub_groups, self._zf):
            if st["thread"] is not None and not st["thread"].is_alive():
                self._publish_sg(sg, st)
            g = sg.grad32
            if combined != 1.0:
                g.mul_(1.0 / combined)
            if st["hot_idx"] is None or \
                    (self._zf_step % self.zf.select_interval) == 1:
                self._select_hot_sg(sg, st, g)
            idx = st["hot_idx"]
            gh = g[idx]
            st["hot_m"].mul_(beta1).add_(gh, alpha=1 - beta1)
            st["hot_v"].mul_(beta2).addcmul_(gh, gh, value=1 - beta2)
            if wd:
                st["hot_master"].mul_(1 - lr * wd)
            denom = (st["hot_v"] / bc2).sqrt().add_(eps)
            st["hot_master"].addcdiv_(st["hot_m"] / bc1, denom, value=-lr)
            sg.flat16.view(-1)[idx] = st["hot_master"].to(sg.flat16.dtype)
            g.index_fill_(0, idx, 0.0)
            st["acc_cpu"].add_(g.to("cpu"))
            st["acc_steps"] += 1
            if lazy:
                if st["thread"] is not None:
                    st["thread"].join()
                    self._publish_sg(sg, st)
                self._writeback_hot_sg(st)
                idx_cpu = st["hot_idx"].cpu()
                hot_restore = (idx_cpu, st["master_cpu"][idx_cpu].clone(),
                               st["m_cpu"][idx_cpu].clone(),
                               st["v_cpu"][idx_cpu].clone())
                st["acc_cpu"].div_(max(st["acc_steps"], 1))
                if self.zf.overlap_step:
                    st["acc_cpu"], st["acc_snap"] = (st["acc_snap"],
                                                     st["acc_cpu"])
                    args = (st["master_cpu"], st["acc_snap"], st["m_cpu"],
                            st["v_cpu"], lr, beta1, beta2, eps, t, wd,
                            bc1, bc2) + hot_restore
                    st["thread"] = threading.Thread(
                        target=ZenFlowZeroOptimizer._host_adam, args=args,
                        daemon=True)
                    st["thread"].start()
                else:
                    ZenFlowZeroOptimizer._host_adam(
                        st["master_cpu"], st["acc_cpu"], st["m_cpu"],
                        st["v_cpu"], lr, beta1, beta2, eps, t, wd, bc1,
                        bc2, *hot_restore)
                    st["acc_cpu"].zero_()
                    self._publish_sg(sg, st)
                st["acc_steps"] = 0
        self._clear_grads()
        self._refresh_persistent_params()

    def _writeback_hot_sg(self, st):
        if st["hot_idx"] is None:
            return
        idx = st["hot_idx"].cpu()
        st["master_cpu"][idx] = st["hot_master"].cpu()
        st["m_cpu"][idx] = st["hot_m"].cpu()
        st["v_cpu"][idx] = st["hot_v"].cpu()

    def _select_hot_sg(self, sg, st, g):
        k = max(1, int(self.zf.topk_ratio * g.numel()))
        self._writeback_hot_sg(st)
        st["hot_idx"] = g.abs().topk(k).indices
        idx_cpu = st["hot_idx"].cpu()
        dev = g.device
        st["hot_master"] = st["master_cpu"][idx_cpu].to(dev,
                                                        non_blocking=True)
        st["hot_m"] = st["m_cpu"][idx_cpu].to(dev, non_blocking=True)
        st["hot_v"] = st["v_cpu"][idx_cpu].to(dev, non_blocking=True)

    def _publish_sg(self, sg, st):
        st["thread"] = None
        if st["acc_snap"] is not None:
            st["acc_snap"].zero_()
        sg.flat16.copy_(
            st["master_cpu"].to(sg.flat16.device, non_blocking=True)
            .to(sg.flat16.dtype))
        if st["hot_idx"] is not None:
            idx_cpu = st["hot_idx"].cpu()
            dev = sg.flat16.device
            st["hot_master"] = st["master_cpu"][idx_cpu].to(dev)
            st["hot_m"] = st["m_cpu"][idx_cpu].to(dev)
            st["hot_v"] = st["v_cpu"][idx_cpu].to(dev)

    def _drain_threads(self):
        for sg, st in zip(self.sub_groups, self._zf):
            if st["thread"] is not None:
                st["thread"].join()
                self._publish_sg(sg, st)

    def state_dict(self):
        self._drain_threads()
        sd = super().state_dict()
        sd["zf_step"] = self._zf_step
        sd["zf_m"] = [st["m_cpu"] for st in self._zf]
        sd["zf_v"] = [st["v_cpu"] for st in self._zf]
        return sd

    def load_state_dict(self, sd, load_optimizer_states=True):
        super().load_state_dict(sd, load_optimizer_states)
        self._zf_step = sd.get("zf_step", 0)
        if load_optimizer_states and "zf_m" in sd:
            for st, m, v in zip(self._zf, sd["zf_m"], sd["zf_v"]):
                st["m_cpu"].copy_(m)
                st["v_cpu"].copy_(v)
                st["hot_idx"] = None

    def destroy(self):
        self._drain_threads()
        super().destroy()
