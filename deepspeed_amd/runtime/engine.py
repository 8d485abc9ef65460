"""DeepSpeedEngine — the training engine.

Parity: reference `deepspeed/runtime/engine.py:252` (DeepSpeedEngine):
`forward:2801`, `backward:3205`, `step:3405`, `save_checkpoint:4749`,
`load_checkpoint:4271`, `_broadcast_model:1715`, `allreduce_gradients:2891`.

MI355X-native: one process per GPU over RCCL; dtype default bf16; ZeRO
optimizers own the grad comm (reduce-scatter on a side stream); stage-0 fp32
falls back to bucketed allreduce at the GAS boundary.
"""
import os
from typing import Callable, Optional

import torch

from .. import comm as dist
from ..comm import groups
from ..config import DeepSpeedConfig
from ..ops.adam import FusedAdam
from ..utils.logging import log_dist, logger
from ..utils.timer import SynchronizedWallClockTimer, ThroughputTimer
from . import lr_schedules
from .loss_scaler import LossScalerBase
from .utils import (CheckOverflow, get_grad_norm, see_memory_usage,
                    clip_tensors_by_global_norm)

MEMORY_OPT_ALLREDUCE_SIZE = int(5e8)


class DeepSpeedEngine(torch.nn.Module):
    def __init__(self,
                 args=None,
                 model=None,
                 optimizer=None,
                 model_parameters=None,
                 training_data=None,
                 lr_scheduler=None,
                 mpu=None,
                 dist_init_required=None,
                 collate_fn=None,
                 config=None,
                 config_class: Optional[DeepSpeedConfig] = None,
                 dont_change_device=False):
        super().__init__()
        self.module = model
        self.client_optimizer = optimizer
        self.client_lr_scheduler = lr_scheduler
        self.training_data = training_data
        self.collate_fn = collate_fn
        self.mpu = mpu
        # batch triple is defined over DATA-parallel ranks only
        if mpu is not None and hasattr(mpu, "get_data_parallel_world_size"):
            cfg_world = mpu.get_data_parallel_world_size()
        else:
            cfg_world = dist.get_world_size()
            sp = 1
            if isinstance(config, dict):
                sp = config.get("sequence_parallel", {}).get(
                    "sequence_parallel_size", 1)
            if sp > 1:
                cfg_world = max(1, cfg_world // sp)
        self._config = config_class or DeepSpeedConfig(
            config, world_size=cfg_world)

        self.global_steps = 0
        self.global_samples = 0
        self.micro_steps = 0
        cl_cfg = getattr(self._config, "curriculum_learning", None)
        if cl_cfg:
            from .data_pipeline import CurriculumScheduler
            self.curriculum_scheduler = CurriculumScheduler(cl_cfg)
        else:
            self.curriculum_scheduler = None
        pld_cfg = getattr(self._config, "progressive_layer_drop", None)
        if pld_cfg:
            from .data_pipeline import ProgressiveLayerDrop
            self.progressive_layer_drop = ProgressiveLayerDrop(
                theta=pld_cfg.get("theta", 0.5),
                gamma=pld_cfg.get("gamma", 0.001))
        else:
            self.progressive_layer_drop = None
        self.skipped_steps = 0
        self._is_gradient_accumulation_boundary = None
        self.random_ltd_scheduler = None

        self.device = (torch.device("cuda", dist.get_local_rank())
                       if torch.cuda.is_available() else torch.device("cpu"))
        self.timers = SynchronizedWallClockTimer()
        self.tput_timer = ThroughputTimer(
            batch_size=self.train_batch_size(), start_step=2)

        if mpu is not None:
            groups.set_mpu(mpu)
        dist.configure(self._config)

        sp = self._config.ulysses.sequence_parallel_size
        if sp > 1 and not groups.sequence_parallel_is_initialized():
            groups.initialize_sequence_parallel(sp)

        # training AutoTP (ref runtime/tensor_parallel/tp_manager.py):
        # shard the module over a TP group and make ZeRO/DDP partition
        # over the orthogonal DP group
        tp = self._config.tensor_parallel.autotp_size
        if tp > 1:
            assert sp <= 1, "autotp_size with sequence parallelism: pick one"
            if groups.get_tensor_parallel_group() is None:
                groups.initialize_tensor_parallel(tp)
            from ..module_inject.auto_tp import (add_tp_training_hooks,
                                                 apply_tensor_parallel_hf)
            apply_tensor_parallel_hf(model)
            add_tp_training_hooks(model)

        self.dp_group = groups.get_sequence_data_parallel_group() \
            if sp > 1 else groups.get_data_parallel_group()
        self.dp_world_size = dist.get_world_size(self.dp_group)

        self._configure_distributed_model()

        self.monitor = self._configure_monitor()
        self.optimizer = None
        self.lr_scheduler = None
        self.basic_optimizer = None
        if optimizer is not None or self._config.optimizer is not None \
                or model_parameters is not None:
            self._configure_optimizer(optimizer, model_parameters)
            self._configure_lr_scheduler(lr_scheduler)

        self.quantizer = self._configure_quantization()

        self.training_dataloader = (self.deepspeed_io(training_data)
                                    if training_data is not None else None)

        rltd = getattr(self._config, "random_ltd", None)
        if rltd and rltd.get("enabled", True):
            self._configure_random_ltd(rltd)

        self.losses = None
        self.flops_profiler = None
        if self._config.flops_profiler.enabled:
            from ..profiling.flops_profiler import FlopsProfiler
            self.flops_profiler = FlopsProfiler(self.module)

        see_memory_usage("engine init done",
                         force=self._config.memory_breakdown)

    # ------------------------------------------------------------------ cfg
    def train_batch_size(self):
        return self._config.train_batch_size

    def train_micro_batch_size_per_gpu(self):
        return self._config.train_micro_batch_size_per_gpu

    def gradient_accumulation_steps(self):
        return self._config.gradient_accumulation_steps

    def zero_optimization_stage(self):
        return self._config.zero_config.stage

    def zero_optimization(self):
        return self._config.zero_config.stage > 0

    def fp16_enabled(self):
        return self._config.fp16.enabled

    def bfloat16_enabled(self):
        return self._config.bf16.enabled

    def gradient_clipping(self):
        return self._config.gradient_clipping

    def steps_per_print(self):
        return self._config.steps_per_print

    # --- config getter shims (reference engine API surface) ---
    def dynamic_loss_scale(self):
        return self._config.fp16.loss_scale == 0.0

    def initial_dynamic_scale(self):
        return 2.0 ** self._config.fp16.initial_scale_power

    def dynamic_loss_scale_args(self):
        f = self._config.fp16
        return {"init_scale": 2.0 ** f.initial_scale_power,
                "scale_window": f.loss_scale_window,
                "min_scale": f.min_loss_scale,
                "hysteresis": f.hysteresis}

    def zero_reduce_bucket_size(self):
        return self._config.zero_config.reduce_bucket_size

    def zero_allgather_bucket_size(self):
        return self._config.zero_config.allgather_bucket_size

    def zero_overlap_comm(self):
        return self._config.zero_config.overlap_comm

    def zero_prefetch_bucket_size(self):
        return self._config.zero_config.prefetch_bucket_size

    def zero_param_persistence_threshold(self):
        return self._config.zero_config.param_persistence_threshold

    def zero_max_live_parameters(self):
        return self._config.zero_config.max_live_parameters

    def zero_max_reuse_distance(self):
        return self._config.zero_config.max_reuse_distance

    def zero_sub_group_size(self):
        return self._config.zero_config.sub_group_size

    def zero_offload_optimizer(self):
        return self._config.zero_config.offload_optimizer

    def zero_offload_param(self):
        return self._config.zero_config.offload_param

    def zero_quantized_weights(self):
        return self._config.zero_config.zero_quantized_weights

    def zero_quantized_gradients(self):
        return self._config.zero_config.zero_quantized_gradients

    def zero_gather_16bit_weights_on_model_save(self):
        return self._config.zero_config \
            .gather_16bit_weights_on_model_save

    def flops_profiler_enabled(self):
        return self._config.flops_profiler.enabled

    def wall_clock_breakdown_enabled(self):
        return self._config.wall_clock_breakdown

    def optimizer_name(self):
        return self._config.optimizer.type \
            if self._config.optimizer else None

    def optimizer_params(self):
        return dict(self._config.optimizer.params) \
            if self._config.optimizer else None

    def scheduler_name(self):
        return self._config.scheduler.type \
            if self._config.scheduler else None

    def scheduler_params(self):
        return dict(self._config.scheduler.params) \
            if self._config.scheduler else None

    @property
    def config(self):
        return self._config

    @property
    def global_rank(self):
        return dist.get_rank()

    @property
    def world_size(self):
        return dist.get_world_size()

    @property
    def local_rank(self):
        return dist.get_local_rank()

    def get_data_parallel_rank(self):
        return dist.get_rank(self.dp_group)

    # ------------------------------------------------------------ dist model
    def _configure_distributed_model(self):
        dtype = self._config.dtype
        if self.zero_optimization_stage() == 3:
            # stage-3: params may already be sharded (zero.Init); leave
            # placement to the stage-3 machinery.
            from .zero.stage3_params import module_is_sharded
            if not module_is_sharded(self.module):
                self.module.to(dtype)
        else:
            self.module.to(dtype)
            self.module.to(self.device)
            self._broadcast_model()
        self.module.train()

    def _broadcast_model(self):
        """Sync initial weights from DP rank 0 (ref engine.py:1715).

        Expert (MoE) params broadcast over their expert-DP group — each EP
        rank owns distinct experts.
        """
        if self.dp_world_size <= 1:
            return
        src_rank = dist.get_global_rank(self.dp_group, 0) \
            if self.dp_group is not None else 0
        for p in self.module.parameters():
            if not torch.is_tensor(p):
                continue
            gn = getattr(p, "group_name", None)
            if gn is not None:
                try:
                    epg = groups.get_expert_data_parallel_group(gn)
                except KeyError:
                    continue
                if dist.get_world_size(epg) > 1:
                    dist.broadcast(p.data, dist.get_global_rank(epg, 0),
                                   group=epg)
                continue
            dist.broadcast(p.data, src_rank, group=self.dp_group)
        for b in self.module.buffers():
            if torch.is_tensor(b) and b.numel() > 0:
                dist.broadcast(b.data, src_rank, group=self.dp_group)

    def _configure_monitor(self):
        from ..monitor.monitor import MonitorMaster
        return MonitorMaster(self._config)

    # -------------------------------------------------------------- optimizer
    def _offload_optimizer_enabled(self):
        zc = self._config.zero_config
        return (zc.stage >= 1 and zc.offload_optimizer is not None
                and zc.offload_optimizer.device == "cpu")

    def _configure_basic_optimizer(self, model_parameters):
        cfg = self._config.optimizer
        if cfg is None:
            return FusedAdam(model_parameters, lr=1e-3)
        name = cfg.type.lower()
        params = dict(cfg.params)
        params.pop("torch_adam", None)
        if name in ("adam", "adamw", "fusedadam"):
            if self._offload_optimizer_enabled():
                from ..ops.cpu_adam import DeepSpeedCPUAdam
                params.setdefault("adamw_mode", name != "adam")
                return DeepSpeedCPUAdam(model_parameters, **params)
            params.setdefault("adam_w_mode", name != "adam")
            return FusedAdam(model_parameters, **params)
        if name == "cpuadam" or name == "deepspeedcpuadam":
            from ..ops.cpu_adam import DeepSpeedCPUAdam
            return DeepSpeedCPUAdam(model_parameters, **params)
        if name == "lion":
            if self._offload_optimizer_enabled():
                from ..ops.lion import DeepSpeedCPULion
                return DeepSpeedCPULion(model_parameters, **params)
            from ..ops.lion import FusedLion
            return FusedLion(model_parameters, **params)
        if name == "adagrad":
            if self._offload_optimizer_enabled():
                from ..ops.adagrad import DeepSpeedCPUAdagrad
                return DeepSpeedCPUAdagrad(model_parameters, **params)
            from ..ops.adagrad import FusedAdagrad
            return FusedAdagrad(model_parameters, **params)
        if name == "lamb" or name == "fusedlamb":
            from ..ops.lamb import FusedLamb
            return FusedLamb(model_parameters, **params)
        if name == "onebitadam":
            from ..ops.onebit_adam import OnebitAdam
            return OnebitAdam(model_parameters, deepspeed=self, **params)
        if name == "zerooneadam":
            from ..ops.onebit_adam import ZeroOneAdam
            return ZeroOneAdam(model_parameters, deepspeed=self, **params)
        if name == "onebitlamb":
            from ..ops.onebit_adam import OnebitLamb
            return OnebitLamb(model_parameters, deepspeed=self, **params)
        if name == "muon":
            from ..ops.muon import Muon
            return Muon(model_parameters, **params)
        if name == "sgd":
            return torch.optim.SGD(model_parameters, **params)
        raise ValueError(f"unknown optimizer type {cfg.type}")

    def _configure_optimizer(self, client_optimizer, model_parameters):
        if client_optimizer is not None:
            if isinstance(client_optimizer, Callable):
                basic = client_optimizer(self.module.parameters())
            else:
                basic = client_optimizer
        else:
            if model_parameters is None:
                model_parameters = list(self.module.parameters())
            basic = self._configure_basic_optimizer(model_parameters)
        self.basic_optimizer = basic

        stage = self.zero_optimization_stage()
        dtype = self._config.dtype
        zc = self._config.zero_config

        if stage == 3:
            from .zero.stage3 import ZeroStage3Optimizer
            shard_group, replica_group = self.dp_group, None
            if zc.mics_shard_size > 0:
                shard_group, replica_group = groups.initialize_mics(
                    zc.mics_shard_size)
                log_dist(f"MiCS: shard_size={zc.mics_shard_size}, "
                         "grads average across replica groups", ranks=[0])
            s3_cls = ZeroStage3Optimizer
            s3_extra = {}
            if zc.zenflow is not None:
                from .zenflow import ZenFlowZeroStage3Optimizer
                s3_cls = ZenFlowZeroStage3Optimizer
                s3_extra["zenflow_config"] = zc.zenflow
            self.optimizer = s3_cls(
                basic,
                **s3_extra,
                module=self.module,
                engine=self,
                dp_process_group=shard_group,
                replica_group=replica_group,
                reduce_bucket_size=zc.reduce_bucket_size,
                prefetch_bucket_size=zc.prefetch_bucket_size,
                param_persistence_threshold=zc.param_persistence_threshold,
                model_persistence_threshold=zc.model_persistence_threshold,
                max_live_parameters=zc.max_live_parameters,
                max_reuse_distance=zc.max_reuse_distance,
                sub_group_size=zc.sub_group_size,
                overlap_comm=zc.overlap_comm,
                offload_optimizer=zc.offload_optimizer,
                offload_param=zc.offload_param,
                zero_quantized_weights=zc.zero_quantized_weights,
                zero_quantized_gradients=zc.zero_quantized_gradients,
                zero_quantized_nontrainable_weights=
                zc.zero_quantized_nontrainable_weights,
                leaf_module_names=(zc.leaf_module or {}).get("classes"),
                clip_grad=self.gradient_clipping(),
                static_loss_scale=self._static_loss_scale(),
                dynamic_loss_scale=self._dynamic_loss_scale(),
                dynamic_loss_args=self._dynamic_loss_args(),
                dtype=dtype,
                gradient_accumulation_steps=self.gradient_accumulation_steps())
        elif stage >= 1 or dtype in (torch.float16, torch.bfloat16):
            from .zero.stage_1_and_2 import ZeroStage12Optimizer
            if stage == 0:
                log_dist("dtype is 16-bit with ZeRO disabled: using "
                         "stage-1 partitioned fp32-master optimizer "
                         "(identical numerics, lower memory)", ranks=[0])
            opt_cls = ZeroStage12Optimizer
            extra = {}
            if zc.zenflow is not None:
                from .zenflow import ZenFlowZeroOptimizer
                opt_cls = ZenFlowZeroOptimizer
                extra["zenflow_config"] = zc.zenflow
            self.optimizer = opt_cls(
                basic,
                **extra,
                engine=self,
                stage=max(stage, 1),
                dp_process_group=self.dp_group,
                reduce_bucket_size=zc.reduce_bucket_size,
                allgather_bucket_size=zc.allgather_bucket_size,
                overlap_comm=zc.overlap_comm,
                clip_grad=self.gradient_clipping(),
                static_loss_scale=self._static_loss_scale(),
                dynamic_loss_scale=self._dynamic_loss_scale(),
                dynamic_loss_args=self._dynamic_loss_args(),
                dtype=dtype,
                zero_quantized_gradients=zc.zero_quantized_gradients,
                gradient_predivide_factor=self._config
                .gradient_predivide_factor,
                communication_data_type=self._comm_dtype(),
                gradient_accumulation_steps=self.gradient_accumulation_steps())
        else:
            self.optimizer = basic
        log_dist(f"optimizer: {type(self.optimizer).__name__} "
                 f"(basic {type(basic).__name__}, stage {stage})", ranks=[0])

    def _static_loss_scale(self):
        ls = self._config.fp16.loss_scale
        return ls if ls > 0 else 1.0

    def _dynamic_loss_scale(self):
        return self._config.fp16.enabled and self._config.fp16.loss_scale == 0

    def _dynamic_loss_args(self):
        c = self._config.fp16
        if not self._dynamic_loss_scale():
            return None
        return dict(init_scale=2**c.initial_scale_power,
                    scale_window=c.loss_scale_window,
                    min_scale=c.min_loss_scale,
                    delayed_shift=c.hysteresis,
                    consecutive_hysteresis=c.consecutive_hysteresis)

    def _configure_lr_scheduler(self, client_scheduler):
        if client_scheduler is not None:
            if isinstance(client_scheduler, Callable):
                self.lr_scheduler = client_scheduler(self.optimizer)
            else:
                self.lr_scheduler = client_scheduler
        elif self._config.scheduler is not None:
            cls = lr_schedules.SCHEDULES[self._config.scheduler.type]
            self.lr_scheduler = cls(self.optimizer,
                                    **self._config.scheduler.params)

    # ------------------------------------------------------------- data
    def deepspeed_io(self, dataset, batch_size=None, route=None):
        from .dataloader import DeepSpeedDataLoader
        return DeepSpeedDataLoader(
            dataset,
            batch_size=batch_size or self.train_micro_batch_size_per_gpu(),
            data_parallel_world_size=self.dp_world_size,
            data_parallel_rank=dist.get_rank(self.dp_group),
            collate_fn=self.collate_fn)

    # ------------------------------------------------------------- train
    def is_gradient_accumulation_boundary(self):
        """True when the just-completed micro-step closes a GAS window.

        self.micro_steps counts COMPLETED backward passes (incremented at
        the end of engine.backward), so the boundary is micro_steps % GAS
        == 0 — evaluated between backward() and step().
        """
        if self._is_gradient_accumulation_boundary is not None:
            return self._is_gradient_accumulation_boundary
        if self.micro_steps == 0:
            return False
        return self.micro_steps % self.gradient_accumulation_steps() == 0

    def set_gradient_accumulation_boundary(self, is_boundary):
        self._is_gradient_accumulation_boundary = is_boundary

    def torch_autocast_enabled(self):
        return self._config.torch_autocast.enabled

    def forward(self, *inputs, **kwargs):
        if self.flops_profiler is not None and \
                self.global_steps == self._config.flops_profiler.profile_step:
            self.flops_profiler.start_profile(ignore_list=None)
        if self.wall_clock_breakdown():
            self.timers("forward").start()
        if self.torch_autocast_enabled():
            ac_dtype = (torch.bfloat16 if "bf" in
                        self._config.torch_autocast.dtype else torch.float16)
            dev = "cuda" if torch.cuda.is_available() else "cpu"
            with torch.autocast(device_type=dev, dtype=ac_dtype):
                loss = self.module(*inputs, **kwargs)
        else:
            loss = self.module(*inputs, **kwargs)
        if self.wall_clock_breakdown():
            self.timers("forward").stop()
        return loss

    def _configure_random_ltd(self, cfg):
        """Config-driven random-LTD (ref data_routing): wraps decoder
        layers and steps the token schedule each optimizer step.

        config: {"random_ltd": {"enabled": true, "layer_class":
        "LlamaDecoderLayer", "min_tokens": 128, "max_tokens": 4096,
        "schedule_steps": 1000, "layer_ids": [1,2,...]}}"""
        from .random_ltd import convert_to_random_ltd
        cls_name = cfg.get("layer_class", "LlamaDecoderLayer")
        layer_cls = None
        for m in self.module.modules():
            if type(m).__name__ == cls_name:
                layer_cls = type(m)
                break
        if layer_cls is None:
            logger.warning(f"random_ltd: no {cls_name} modules found")
            return
        convert_to_random_ltd(self.module, layer_cls,
                              cfg.get("min_tokens", 128),
                              cfg.get("max_tokens", 100000),
                              cfg.get("schedule_steps", 1000),
                              layer_ids=cfg.get("layer_ids"))
        self.random_ltd_scheduler = self.module.random_ltd_scheduler

    def _comm_dtype(self):
        cdt = self._config.communication_data_type
        if cdt is None:
            return None
        return {"fp32": torch.float32, "fp16": torch.float16,
                "bf16": torch.bfloat16}.get(cdt, None)

    def wall_clock_breakdown(self):
        return self._config.wall_clock_breakdown

    def backward(self, loss, retain_graph=False, scale_wrt_gas=True):
        if self.gradient_accumulation_steps() > 1 and scale_wrt_gas:
            loss = loss / self.gradient_accumulation_steps()
        if self.wall_clock_breakdown():
            self.timers("backward").start()
        if hasattr(self.optimizer, "backward"):
            self.optimizer.backward(loss, retain_graph=retain_graph)
        else:
            loss.backward(retain_graph=retain_graph)
            # This micro-step has not been counted yet (micro_steps += 1
            # happens below), so the boundary test must look one ahead —
            # otherwise the allreduce fires one micro-batch late and the
            # boundary step consumes unreduced rank-local grads
            # (ref engine.py:3284 uses (micro_steps + 1) % GAS == 0).
            if self._is_gradient_accumulation_boundary is not None:
                at_boundary = self._is_gradient_accumulation_boundary
            else:
                at_boundary = (self.micro_steps + 1) % \
                    self.gradient_accumulation_steps() == 0
            if at_boundary:
                self.allreduce_gradients()
        if self.wall_clock_breakdown():
            self.timers("backward").stop()
        self.micro_steps += 1
        self.global_samples += self.train_micro_batch_size_per_gpu()
        try:  # remembered for the step-boundary monitor events
            self.losses = float(loss.detach().float())
        except Exception:
            pass
        return loss

    def allreduce_gradients(self, bucket_size=MEMORY_OPT_ALLREDUCE_SIZE):
        """ZeRO-0 fp32 fallback: bucketed allreduce (ref engine.py:3784);
        sparse grads (sparse embeddings) average via (indices, values)
        exchange instead of densifying (ref sparse_allreduce:3815)."""
        if self.dp_world_size <= 1:
            return
        sparse = [p for p in self.module.parameters()
                  if p.grad is not None and p.grad.is_sparse]
        if sparse:
            from .sparse_tensor import SparseTensor, sparse_allreduce
            for p in sparse:
                st = sparse_allreduce(SparseTensor(p.grad),
                                      dp_group=self.dp_group,
                                      dp_world_size=self.dp_world_size)
                p.grad = st.to_coo_tensor().coalesce()
        grads = [p.grad for p in self.module.parameters()
                 if p.grad is not None and not p.grad.is_sparse]
        bucket, numel = [], 0
        for g in grads:
            bucket.append(g)
            numel += g.numel()
            if numel >= bucket_size:
                self._allreduce_bucket(bucket)
                bucket, numel = [], 0
        if bucket:
            self._allreduce_bucket(bucket)

    def _allreduce_bucket(self, bucket):
        from .utils import flatten_dense_tensors, unflatten_dense_tensors
        flat = flatten_dense_tensors(bucket)
        flat.div_(self.dp_world_size)
        dist.all_reduce(flat, group=self.dp_group)
        for buf, synced in zip(bucket, unflatten_dense_tensors(flat, bucket)):
            buf.copy_(synced)

    def step(self, lr_kwargs=None):
        if self.curriculum_scheduler is not None:
            self.curriculum_scheduler.update_difficulty(self.global_steps + 1)
        if self.progressive_layer_drop is not None:
            self.progressive_layer_drop.update_state(self.global_steps + 1)
        if self.wall_clock_breakdown():
            self.timers("step").start()
        if self.is_gradient_accumulation_boundary():
            self._take_model_step(lr_kwargs)
        if self.wall_clock_breakdown():
            self.timers("step").stop()
        if self.flops_profiler is not None and \
                self.global_steps == self._config.flops_profiler.profile_step + 1:
            self.flops_profiler.print_model_profile(
                profile_step=self.global_steps,
                output_file=self._config.flops_profiler.output_file)
            self.flops_profiler.end_profile()
            self.flops_profiler = None

    def _configure_quantization(self):
        """MoQ quantize-aware training (ref engine.py:2257): fake-int
        quantize the 16-bit weights after each step with annealing bit
        widths. Stage 1/2 quantizes the flat buckets; stage 0 the
        module params; stage 3 is unsupported (sharded weights)."""
        q = self._config.quantize_training
        if not q or not q.get("enabled"):
            return None
        if self.zero_optimization_stage() == 3:
            log_dist("quantize_training: stage 3 not supported; "
                     "disabled", ranks=[0])
            return None
        from .quantize import Quantizer
        bits = q.get("quantize_bits", {})
        sched = q.get("schedule", {})
        n_groups = len(getattr(self.optimizer, "buckets", [])) or \
            sum(1 for p in self.module.parameters() if p.requires_grad)
        return Quantizer(
            q_groups=q.get("quantize_groups", 1),
            q_verbose=q.get("quantize_verbose", False),
            start_bits=bits.get("start_bits", 16),
            target_bits=bits.get("target_bits", 8),
            quantize_period=sched.get("quantize_period", 1000),
            layer_num=n_groups)

    def _apply_moq(self):
        overflow = getattr(self.optimizer, "overflow", False)
        if hasattr(self.optimizer, "buckets"):  # ZeRO 1/2 flat slabs
            tensors = [b.flat16 for b in self.optimizer.buckets]
        else:
            tensors = [p for p in self.module.parameters()
                       if p.requires_grad and p.dim() >= 2]
        self.quantizer.quantize(tensors, overflow=overflow)

    def _take_model_step(self, lr_kwargs=None):
        self.optimizer.step()
        if self.quantizer is not None:
            self._apply_moq()
        if self.random_ltd_scheduler is not None:
            self.random_ltd_scheduler.update(self.global_steps + 1)
        from ..ops.fp8_linear import bump_fp8_version
        bump_fp8_version()  # invalidate cached fp8 weight copies
        overflow = getattr(self.optimizer, "overflow", False)
        if not isinstance(self.optimizer, torch.optim.Optimizer):
            pass
        else:
            self.optimizer.zero_grad(set_to_none=True)
        if overflow:
            self.skipped_steps += 1
        else:
            if self.lr_scheduler is not None:
                self.lr_scheduler.step(**(lr_kwargs or {}))
        self.global_steps += 1
        if self.global_steps % self.steps_per_print() == 0:
            self._report_progress()
        if self.monitor.enabled:
            events = [("Train/lr", self.get_lr()[0] if self.get_lr()
                       else 0.0, self.global_steps)]
            if self.losses is not None:
                events.append(("Train/loss", self.losses,
                               self.global_steps))
            scale = getattr(self.optimizer, "loss_scale", None)
            if scale is not None:
                events.append(("Train/loss_scale", float(scale),
                               self.global_steps))
            norm = self.get_global_grad_norm()
            if norm:
                events.append(("Train/grad_norm", float(norm),
                               self.global_steps))
            self.monitor.write_events(events)

    def _report_progress(self):
        lr = [g["lr"] for g in self.optimizer.param_groups] \
            if hasattr(self.optimizer, "param_groups") else []
        loss_scale = getattr(self.optimizer, "loss_scale", 1.0)
        norm = getattr(self.optimizer, "get_global_grad_norm", lambda: 0.0)()
        log_dist(f"step={self.global_steps}, skipped={self.skipped_steps}, "
                 f"lr={lr}, scale={loss_scale}, grad_norm={norm:.4f}",
                 ranks=[0])
        if self.wall_clock_breakdown() and \
                hasattr(self.optimizer, "partition_stats"):
            st = self.optimizer.partition_stats()
            if st["fetches"]:
                log_dist(
                    f"zero3 partition traffic: {st['fetches']} fetches, "
                    f"{st['prefetch_hits']} prefetch hits, "
                    f"{st['demand_gathers']} demand gathers "
                    f"({st['gathered_numel'] / 1e6:.1f}M elems), "
                    f"{st['releases']} releases", ranks=[0])
        self.monitor.write_events([
            ("Train/lr", lr[0] if lr else 0.0, self.global_steps),
        ])

    def train(self, mode=True):
        self.module.train(mode)
        return self

    def eval(self):
        self.module.eval()
        return self

    def offload_states(self, include=None, device="cpu",
                       pin_memory=False, non_blocking=False):
        """ZeRO-3: push engine states to host to free HBM (ref
        engine.offload_states)."""
        assert hasattr(self.optimizer, "offload_states"), \
            "offload_states requires ZeRO stage 3"
        self.optimizer.offload_states(include, device, pin_memory,
                                      non_blocking)

    def reload_states(self, non_blocking=False):
        self.optimizer.reload_states(non_blocking)

    def zero_grad(self, set_to_none=True):
        if hasattr(self.optimizer, "zero_grad"):
            self.optimizer.zero_grad(set_to_none=set_to_none)

    # ---------------------------------------------------------- checkpoint
    def save_checkpoint(self, save_dir, tag=None, client_state=None,
                        save_latest=True, exclude_frozen_parameters=False):
        from .checkpointing import save_checkpoint as _save
        return _save(self, save_dir, tag=tag, client_state=client_state,
                     save_latest=save_latest,
                     exclude_frozen_parameters=exclude_frozen_parameters)

    def load_checkpoint(self, load_dir, tag=None, load_module_strict=True,
                        load_optimizer_states=True, load_lr_scheduler_states=True,
                        load_module_only=False):
        if getattr(self._config, "load_universal_checkpoint", False):
            # checkpoint.load_universal: true -> load_dir points at a
            # ds_to_universal output (per-param fp32 fragments), loadable
            # into ANY topology (ref universal_checkpoint.py:149)
            self.load_universal_checkpoint(load_dir)
            return load_dir, {}
        from .checkpointing import load_checkpoint as _load
        return _load(self, load_dir, tag=tag,
                     load_module_strict=load_module_strict,
                     load_optimizer_states=load_optimizer_states,
                     load_lr_scheduler_states=load_lr_scheduler_states,
                     load_module_only=load_module_only)

    def compile(self, sample_input=None, sample_labels=None,
                backend="hipgraph", num_warmup_iters=3):
        """Compile the training step (ref engine.compile:5706).

        backend "hipgraph": capture the module's fwd+bwd as hipGraphs
        (see deepspeed_amd/compile.py — ZeRO comm stays eager/overlapped).
        """
        if backend == "inductor":
            # plain torch.compile of the module; composes with the ZeRO
            # grad hooks (tested: test_torch_compile_with_zero2)
            self.module = torch.compile(self.module)
            return self
        if backend == "autosp":
            from ..sequence.auto_sp import autosp_backend
            self.module = torch.compile(self.module,
                                        backend=autosp_backend)
            return self
        if backend != "hipgraph":
            raise ValueError(
                f"unknown compile backend {backend!r}: choices are "
                "'hipgraph' (fwd+bwd graph capture), 'inductor' "
                "(torch.compile), 'autosp' (SDPA->Ulysses graph pass)")
        if sample_input is None:
            raise ValueError("compile() needs a static-shape sample_input")
        from ..compile import engine_compile
        return engine_compile(self, sample_input, sample_labels,
                              num_warmup_iters)

    def generate(self, input_ids, **kwargs):
        """RLHF-style generation with the training weights (hybrid engine:
        gathers ZeRO-3 shards for the rollout, then releases)."""
        from .hybrid_engine import generate
        return generate(self, input_ids, **kwargs)

    def load_universal_checkpoint(self, universal_dir):
        """Resume from a universal checkpoint at ANY dp world size."""
        from ..checkpoint.universal import load_universal_into_optimizer
        load_universal_into_optimizer(self.optimizer, universal_dir)

    def save_16bit_model(self, save_dir, save_filename="pytorch_model.bin",
                         exclude_frozen_parameters=False):
        """Consolidated 16-bit weights (gathers ZeRO-3 shards on rank 0).

        Ref engine.py:5567 / _zero3_consolidated_16bit_state_dict:5491.
        """
        from ..comm import groups as _grp
        if _grp.get_tensor_parallel_world_size() > 1:
            raise NotImplementedError(
                "save_16bit_model under tensor parallelism would write "
                "one TP shard as if it were the full model; use "
                "save_checkpoint (per-mp-rank files) instead")
        sd = self._consolidated_16bit_state_dict()
        if dist.get_rank() == 0 and sd is not None:
            os.makedirs(save_dir, exist_ok=True)
            torch.save(sd, os.path.join(save_dir, save_filename))
        if dist.is_initialized():
            dist.barrier()
        return True

    def _consolidated_16bit_state_dict(self):
        if self.zero_optimization_stage() != 3:
            return self.module.state_dict() if \
                self.get_data_parallel_rank() == 0 else None
        from .hybrid_engine import gathered_for_generation
        with gathered_for_generation(self):
            if self.get_data_parallel_rank() == 0:
                return {k: v.detach().clone()
                        for k, v in self.module.state_dict().items()}
        return None

    def save_fp16_model(self, save_dir, save_filename="pytorch_model.bin"):
        """Reference-compat alias of save_16bit_model."""
        return self.save_16bit_model(save_dir, save_filename)

    def set_lr(self, lr):
        for g in self.optimizer.param_groups:
            g["lr"] = lr

    def get_loss_scale(self):
        return getattr(self.optimizer, "loss_scale", 1.0)

    def empty_partition_cache(self):
        if hasattr(self.optimizer, "empty_partition_cache"):
            self.optimizer.empty_partition_cache()

    def module_state_dict(self, exclude_frozen_parameters=False):
        return self.module.state_dict()

    def load_module_state_dict(self, state_dict, strict=True):
        self.module.load_state_dict(state_dict, strict=strict)

    # ------------------------------------------------------------- misc
    def get_lr(self):
        return [g["lr"] for g in self.optimizer.param_groups]

    def set_lr(self, lr):
        for g in self.optimizer.param_groups:
            g["lr"] = lr

    def get_mom(self):
        """Momentum/betas per group (ref engine.get_mom)."""
        key = "betas" if any("betas" in g for g in
                             self.optimizer.param_groups) else "momentum"
        return [g.get(key) for g in self.optimizer.param_groups]

    def get_global_grad_norm(self):
        return getattr(self.optimizer, "get_global_grad_norm", lambda: 0.0)()

    def get_batch_info(self):
        """(train_batch_size, micro_batch, gradient_accumulation_steps)."""
        return (self.train_batch_size(),
                self.train_micro_batch_size_per_gpu(),
                self.gradient_accumulation_steps())

    def set_train_batch_size(self, train_batch_size):
        """Adjust the global batch by changing GAS (DP world and micro
        batch stay fixed — ref engine.set_train_batch_size)."""
        mb = self.train_micro_batch_size_per_gpu()
        denom = mb * self.dp_world_size
        if train_batch_size % denom != 0:
            raise ValueError(
                f"train_batch_size {train_batch_size} not divisible by "
                f"micro_batch*dp_world {denom}")
        self._config.gradient_accumulation_steps = train_batch_size // denom
        self._config.train_batch_size = train_batch_size

    def set_train_micro_batch_size(self, micro_batch_size):
        self._config.train_micro_batch_size_per_gpu = micro_batch_size
        self._config.train_batch_size = micro_batch_size * \
            self.dp_world_size * self.gradient_accumulation_steps()

    @property
    def loss_scale(self):
        return getattr(self.optimizer, "loss_scale", 1.0)

    def was_step_applied(self):
        """False when the last step was skipped by overflow backoff."""
        return not getattr(self.optimizer, "overflow", False)

    def no_sync(self):
        """Context manager suppressing the ZeRO-0 boundary allreduce for
        local accumulation (ref engine.no_sync). ZeRO>=1 owns its
        reduction schedule and rejects it."""
        assert self.zero_optimization_stage() == 0, \
            "no_sync is only meaningful for the ZeRO-0 fallback path"
        import contextlib

        @contextlib.contextmanager
        def ctx():
            prev = self._is_gradient_accumulation_boundary
            self.set_gradient_accumulation_boundary(False)
            try:
                yield
            finally:
                self._is_gradient_accumulation_boundary = prev
        return ctx()

    def get_sequence_parallel_group(self):
        return groups.get_sequence_parallel_group()

    def get_model_parallel_rank(self):
        return groups.get_tensor_parallel_rank()  # 0 when no TP group

    get_tensor_parallel_rank = get_model_parallel_rank

    def sparse_allreduce(self, sparse, dp_group=None):
        from .sparse_tensor import sparse_allreduce as _sa
        return _sa(sparse, dp_group or self.dp_group)

    def sparse_allreduce_bucket(self, bucket, dp_group=None):
        from .sparse_tensor import sparse_allreduce_bucket as _sab
        return _sab(bucket, dp_group or self.dp_group)

    def destroy(self):
        if hasattr(self.optimizer, "destroy"):
            self.optimizer.destroy()
