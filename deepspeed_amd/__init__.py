"""deepspeed_amd — MI355X-native large-scale training framework.

A from-scratch reimplementation of the deepspeedai/DeepSpeed feature set
(see /root/repo/SURVEY.md) designed MI355X-first: PyTorch-ROCm +
hand-written HIP/CDNA4 (gfx950) kernels + RCCL over xGMI. No accelerator
dispatch layer, no Triton, no CUDA shims.

API parity: `initialize()` mirrors reference `deepspeed/__init__.py:93`.
"""
__version__ = "0.1.0"
__git_branch__ = "main"

from . import comm  # noqa: F401
from . import ops  # noqa: F401
from .config import DeepSpeedConfig  # noqa: F401
from .runtime.engine import DeepSpeedEngine
from .runtime import zero  # noqa: F401
from .runtime.pipe import PipelineModule, LayerSpec  # noqa: F401
from .moe.layer import MoE  # noqa: F401
from .ops.adam import FusedAdam  # noqa: F401
from .utils.logging import logger, log_dist  # noqa: F401
from .utils.init_on_device import OnDevice  # noqa: F401
from .runtime.zero import Init as zero_Init  # noqa: F401 (zero.Init)
from .elasticity import compute_elastic_config  # noqa: F401


def initialize(args=None,
               model=None,
               optimizer=None,
               model_parameters=None,
               training_data=None,
               lr_scheduler=None,
               distributed_port=29500,
               mpu=None,
               dist_init_required=None,
               collate_fn=None,
               config=None,
               mesh_param=None,
               config_params=None):
    """Initialize the engine. Returns (engine, optimizer, dataloader,
    lr_scheduler) — same tuple as the reference."""
    assert model is not None, "deepspeed_amd.initialize: model is required"
    if config is None and config_params is not None:
        config = config_params
    if config is None and args is not None and \
            hasattr(args, "deepspeed_config") and args.deepspeed_config:
        config = args.deepspeed_config

    if dist_init_required is None or dist_init_required:
        comm.init_distributed(distributed_port=distributed_port)

    from .runtime.pipe.module import PipelineModule
    if isinstance(model, PipelineModule):
        from .runtime.pipe.engine import PipelineEngine
        engine = PipelineEngine(args=args,
                                model=model,
                                optimizer=optimizer,
                                model_parameters=model_parameters,
                                training_data=training_data,
                                lr_scheduler=lr_scheduler,
                                mpu=model.mpu(),
                                dist_init_required=dist_init_required,
                                collate_fn=collate_fn,
                                config=config)
    else:
        engine = DeepSpeedEngine(args=args,
                                 model=model,
                                 optimizer=optimizer,
                                 model_parameters=model_parameters,
                                 training_data=training_data,
                                 lr_scheduler=lr_scheduler,
                                 mpu=mpu,
                                 dist_init_required=dist_init_required,
                                 collate_fn=collate_fn,
                                 config=config)
    return (engine, engine.optimizer, engine.training_dataloader,
            engine.lr_scheduler)


def init_inference(model, config=None, **kwargs):
    """Inference engine (kernel-injection lite). Ref deepspeed/__init__.py:328."""
    from .inference.engine import InferenceEngine
    return InferenceEngine(model, config=config, **kwargs)


def tp_model_init(model, tp_size, dtype=None, config=None, **kwargs):
    """Shard `model` over a TP group of `tp_size` for training (ref
    deepspeed/__init__.py:408; sharding here happens immediately — the
    config path `tensor_parallel.autotp_size` does the same at
    initialize())."""
    import torch
    from .comm import groups
    from .module_inject.auto_tp import (add_tp_training_hooks,
                                        apply_tensor_parallel_hf)
    if groups.get_tensor_parallel_group() is None:
        groups.initialize_tensor_parallel(tp_size)
    apply_tensor_parallel_hf(model)
    add_tp_training_hooks(model)
    if dtype is not None:
        model.to(dtype)
    return model


def add_config_arguments(parser):
    """Ref deepspeed/__init__.py:305."""
    group = parser.add_argument_group("DeepSpeed-AMD",
                                      "DeepSpeed-AMD configurations")
    group.add_argument("--deepspeed", default=False, action="store_true",
                       help="Enable DeepSpeed-AMD (helper flag)")
    group.add_argument("--deepspeed_config", default=None, type=str,
                       help="DeepSpeed-AMD json configuration file")
    group.add_argument("--deepscale", default=False, action="store_true")
    group.add_argument("--local_rank", type=int, default=-1)
    return parser

from .accelerator import get_accelerator  # noqa: F401,E402
