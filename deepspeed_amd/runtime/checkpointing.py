"""Engine checkpoint save/load in the reference on-disk layout.

Parity: reference `runtime/engine.py:4749` (save_checkpoint),
`:4271` (load_checkpoint), name scheme `_get_zero_ckpt_prefix:4195`
(`zero_pp_rank_{dp}_mp_rank_{mp:02}_optim_states.pt`,
`mp_rank_{mp:02}_model_states.pt`, `latest` tag file).
"""
import os

import torch

from .. import comm as dist
from ..utils.logging import log_dist, logger

VERSION = "0.1.0-mi355x"


def _tag(engine, tag):
    return tag if tag is not None else f"global_step{engine.global_steps}"


def _mp_rank():
    # tensor-parallel ranks hold DIFFERENT module shards: each writes
    # its own mp_rank_XX file (ref _get_ckpt_name mp_rank placement)
    from ..comm import groups
    return groups.get_tensor_parallel_rank()


def _pp_stage(engine):
    """Pipeline stage id when the module is a multi-stage
    PipelineModule, else None — stages hold DIFFERENT layers, so each
    writes stage-qualified files (ref PipelineModule layer_* files)."""
    m = engine.module
    if getattr(m, "num_stages", 1) > 1:
        return m.stage_id
    return None


def _qualify(name, stage):
    if stage is None:
        return name
    base, ext = name.rsplit(".", 1)
    return f"{base}_pp_rank_{stage}.{ext}"


def _model_states_name(mp_rank=0):
    return f"mp_rank_{mp_rank:02d}_model_states.pt"


def _zero_ckpt_name(dp_rank, mp_rank=0):
    return f"zero_pp_rank_{dp_rank}_mp_rank_{mp_rank:02d}_optim_states.pt"


def _expert_ckpt_name(ep_rank, mp_rank=0):
    return f"expert_ep_rank_{ep_rank}_mp_rank_{mp_rank:02d}_model_states.pt"


def _ensure_expert_groups(gn):
    """Expert process groups are created lazily on first forward; a
    checkpoint before any step must create them (collective)."""
    from ..comm import groups
    if gn not in groups.get_expert_parallel_group_dict():
        groups.create_expert_and_data_parallel(int(gn.rsplit("_", 1)[1]))


def _expert_param_names(module):
    """Params tagged by the MoE Experts bank (p.allreduce=False,
    p.group_name=<ep group>). Returns {name: group_name}."""
    out = {}
    for n, p in module.named_parameters():
        gn = getattr(p, "group_name", None)
        if gn is not None and not getattr(p, "allreduce", True):
            out[n] = gn
    return out


def _expert_global_name_map(module):
    """{local_param_name: global_param_name} for every expert param:
    `deepspeed_experts.<local>` -> `deepspeed_experts.<ep_rank*nl+local>`
    (ref _get_moe_state_dict's global expert ids). Globally-unique names
    let offline tools (zero_to_fp32, universal) merge the per-EP-rank
    expert files without collisions."""
    from ..comm import groups
    from ..moe.experts import Experts
    out = {}
    for mod_name, mod in module.named_modules():
        if not isinstance(mod, Experts):
            continue
        nl = mod.num_local_experts
        gn = None
        for p in mod.parameters():
            gn = getattr(p, "group_name", None)
            if gn is not None:
                break
        if gn is None:
            continue
        _ensure_expert_groups(gn)
        ep_rank = groups.get_expert_parallel_rank(gn)
        for i, exp in enumerate(mod.deepspeed_experts):
            g = ep_rank * nl + i
            for pn, _ in exp.named_parameters():
                local = f"{mod_name}.deepspeed_experts.{i}.{pn}"
                out[local] = f"{mod_name}.deepspeed_experts.{g}.{pn}"
    return out


def save_checkpoint(engine, save_dir, tag=None, client_state=None,
                    save_latest=True, exclude_frozen_parameters=False):
    tag = _tag(engine, tag)
    ckpt_dir = os.path.join(save_dir, str(tag))
    os.makedirs(ckpt_dir, exist_ok=True)
    dp_rank = engine.get_data_parallel_rank()
    is_zero = hasattr(engine.optimizer, "state_dict") and \
        not isinstance(engine.optimizer, torch.optim.Optimizer)

    # MoE: expert params differ per expert-parallel rank — they go into
    # separate expert_ep_rank files (ref _save_moe_checkpoint,
    # runtime/engine.py:4921) and are excluded from the dense states.
    # Under ZeRO-3 the per-dp-rank optimizer shards already carry each
    # rank's expert state (experts partition over expert-DP), so the
    # separate expert files are only needed for stages <= 2 where the
    # dense model_states would otherwise lose rank-distinct experts.
    expert_names = _expert_param_names(engine.module) \
        if engine.zero_optimization_stage() != 3 else {}
    if expert_names:
        from ..comm import groups
        full_sd = engine.module.state_dict()
        gn = next(iter(expert_names.values()))
        _ensure_expert_groups(gn)
        ep_rank = groups.get_expert_parallel_rank(gn)
        edp_rank = dist.get_rank(groups.get_expert_data_parallel_group(gn))
        if edp_rank == 0:
            gmap = _expert_global_name_map(engine.module)
            esd = {gmap.get(k, k): v for k, v in full_sd.items()
                   if k in expert_names}
            torch.save({"module": esd, "ds_version": VERSION},
                       os.path.join(ckpt_dir, _expert_ckpt_name(ep_rank)))

    # model states: dp-rank 0 of each TP shard writes its mp_rank file
    if dp_rank == 0:
        state = {
            "module": engine.module_state_dict(
                exclude_frozen_parameters=exclude_frozen_parameters),
            "buffer_names": [n for n, _ in engine.module.named_buffers()],
            "optimizer": None if is_zero else (
                engine.optimizer.state_dict()
                if hasattr(engine.optimizer, "state_dict") else None),
            "lr_scheduler": (engine.lr_scheduler.state_dict()
                             if engine.lr_scheduler is not None else None),
            "ds_config": engine.config.to_dict(),
            "ds_version": VERSION,
            "global_steps": engine.global_steps,
            "global_samples": engine.global_samples,
            "skipped_steps": engine.skipped_steps,
            "micro_steps": engine.micro_steps,
        }
        if client_state:
            state.update(client_state)
        if expert_names and state.get("module"):
            state["module"] = {k: v for k, v in state["module"].items()
                               if k not in expert_names}
        torch.save(state, os.path.join(
            ckpt_dir, _qualify(_model_states_name(_mp_rank()),
                               _pp_stage(engine))))

    # zero shards: every dp rank
    if is_zero:
        zstate = {"optimizer_state_dict": engine.optimizer.state_dict(),
                  "ds_version": VERSION}
        torch.save(zstate, os.path.join(
            ckpt_dir, _qualify(_zero_ckpt_name(dp_rank, _mp_rank()),
                               _pp_stage(engine))))

    if dist.is_initialized():
        dist.barrier()
    if save_latest and dist.get_rank() == 0:
        with open(os.path.join(save_dir, "latest"), "w") as f:
            f.write(str(tag))
    log_dist(f"saved checkpoint {ckpt_dir}", ranks=[0])
    return True


def load_checkpoint(engine, load_dir, tag=None, load_module_strict=True,
                    load_optimizer_states=True, load_lr_scheduler_states=True,
                    load_module_only=False):
    if tag is None:
        latest = os.path.join(load_dir, "latest")
        if not os.path.exists(latest):
            logger.warning(f"no 'latest' file in {load_dir}")
            return None, {}
        with open(latest) as f:
            tag = f.read().strip()
    ckpt_dir = os.path.join(load_dir, str(tag))
    model_file = os.path.join(
        ckpt_dir, _qualify(_model_states_name(_mp_rank()),
                           _pp_stage(engine)))
    state = torch.load(model_file, map_location="cpu", weights_only=False)

    is_zero = hasattr(engine.optimizer, "load_state_dict") and \
        not isinstance(engine.optimizer, torch.optim.Optimizer)

    expert_names = _expert_param_names(engine.module) \
        if engine.zero_optimization_stage() != 3 else {}
    if engine.zero_optimization_stage() != 3:
        engine.load_module_state_dict(state["module"],
                                      strict=load_module_strict
                                      and not expert_names)
    if expert_names:
        from ..comm import groups
        gn = next(iter(expert_names.values()))
        _ensure_expert_groups(gn)
        ep_rank = groups.get_expert_parallel_rank(gn)
        efile = os.path.join(ckpt_dir, _expert_ckpt_name(ep_rank))
        if os.path.exists(efile):
            esd = torch.load(efile, map_location="cpu", weights_only=False)
            # expert files carry GLOBAL expert ids; map back to this
            # rank's local names (old local-named files pass through)
            inv = {g: l for l, g in
                   _expert_global_name_map(engine.module).items()}
            sd = {inv.get(k, k): v for k, v in esd["module"].items()}
            engine.module.load_state_dict(sd, strict=False)

    if load_module_only and is_zero:
        # Module-only load under ZeRO still has to reconcile the optimizer's
        # fp32 masters with the freshly loaded 16-bit weights, or the stale
        # masters overwrite them at the first step (ref engine.py:4525).
        if engine.zero_optimization_stage() == 3:
            # Stage 3: module weights live only in the zero shards — restore
            # them from the per-dp-rank zero file without optimizer state.
            dp_rank = engine.get_data_parallel_rank()
            zfile = os.path.join(
                ckpt_dir, _qualify(_zero_ckpt_name(dp_rank, _mp_rank()),
                                   _pp_stage(engine)))
            zstate = torch.load(zfile, map_location="cpu", weights_only=False)
            engine.optimizer.load_state_dict(zstate["optimizer_state_dict"],
                                             load_optimizer_states=False)
        elif hasattr(engine.optimizer, "refresh_fp32_params"):
            engine.optimizer.refresh_fp32_params()

    if not load_module_only:
        if is_zero:
            dp_rank = engine.get_data_parallel_rank()
            zfile = os.path.join(
                ckpt_dir, _qualify(_zero_ckpt_name(dp_rank, _mp_rank()),
                                   _pp_stage(engine)))
            zstate = torch.load(zfile, map_location="cpu", weights_only=False)
            engine.optimizer.load_state_dict(
                zstate["optimizer_state_dict"],
                load_optimizer_states=load_optimizer_states)
        elif load_optimizer_states and state.get("optimizer") is not None \
                and hasattr(engine.optimizer, "load_state_dict"):
            engine.optimizer.load_state_dict(state["optimizer"])
        if load_lr_scheduler_states and engine.lr_scheduler is not None \
                and state.get("lr_scheduler") is not None:
            engine.lr_scheduler.load_state_dict(state["lr_scheduler"])
        engine.global_steps = state.get("global_steps", 0)
        engine.global_samples = state.get("global_samples", 0)
        engine.skipped_steps = state.get("skipped_steps", 0)
        engine.micro_steps = state.get("micro_steps", 0)

    from ..ops.fp8_linear import bump_fp8_version
    bump_fp8_version()  # loaded weights invalidate fp8 caches
    client_state = {k: v for k, v in state.items()
                    if k not in ("module", "optimizer", "lr_scheduler",
                                 "buffer_names", "ds_config", "ds_version")}
    log_dist(f"loaded checkpoint {ckpt_dir}", ranks=[0])
    return ckpt_dir, client_state
