"""Offline checkpoint inspection.

Parity: reference `checkpoint/deepspeed_checkpoint.py`
(DeepSpeedCheckpoint) — the object the offline tools (ds_to_universal,
analysis scripts) use to answer "what is in this checkpoint dir"
without constructing an engine.
"""
import glob
import os

import torch

from ..utils.zero_to_fp32 import (_load_zero_shards, _read_tag,
                                  _reassemble)


class DeepSpeedCheckpoint:
    def __init__(self, ckpt_dir, tag=None):
        self.dir = ckpt_dir
        self.tag_dir = _read_tag(ckpt_dir, tag)
        self.tag = os.path.basename(self.tag_dir)
        self.model_files = sorted(glob.glob(
            os.path.join(self.tag_dir, "mp_rank_*_model_states.pt")))
        self.zero_files = sorted(glob.glob(
            os.path.join(self.tag_dir, "zero_pp_rank_*_optim_states.pt")))
        self.expert_files = sorted(glob.glob(
            os.path.join(self.tag_dir, "expert_ep_rank_*_model_states.pt")))
        self._shards = None
        self._model_state = None

    # -- topology -----------------------------------------------------------
    @property
    def dp_degree(self):
        return max(len(self.zero_files), 1)

    @property
    def tp_degree(self):
        return max(len(self.model_files), 1)

    @property
    def pp_degree(self):
        return 1  # pipeline stages checkpoint per-rank model files here

    @property
    def ep_degree(self):
        return max(len(self.expert_files), 1)

    @property
    def zero_stage(self):
        sh = self._zero_shards()
        if not sh:
            return 0
        return sh[0].get("zero_stage",
                         sh[0]["shard_layout"].get("stage", 0))

    # -- contents -----------------------------------------------------------
    def _zero_shards(self):
        if self._shards is None and self.zero_files:
            self._shards = _load_zero_shards(self.tag_dir)
        return self._shards or []

    def module_state(self):
        """The (dense) module state dict + metadata from mp_rank_00."""
        if self._model_state is None and self.model_files:
            self._model_state = torch.load(self.model_files[0],
                                           map_location="cpu",
                                           weights_only=False)
        return self._model_state or {}

    def parameter_names(self):
        """Every parameter reconstructable from the zero shards (MoE
        experts under their GLOBAL ids)."""
        sh = self._zero_shards()
        return [name for name, _ in _reassemble(sh)] if sh else \
            list(self.module_state().get("module", {}).keys())

    def fp32_parameters(self):
        """Iterate (name, fp32 tensor) reassembled from the shards."""
        yield from _reassemble(self._zero_shards())

    def client_state(self):
        ms = self.module_state()
        return {k: ms[k] for k in ("global_steps", "global_samples",
                                   "skipped_steps", "micro_steps",
                                   "ds_version") if k in ms}

    def summary(self):
        return {
            "tag": self.tag,
            "zero_stage": self.zero_stage,
            "dp_degree": self.dp_degree,
            "ep_degree": self.ep_degree if self.expert_files else 1,
            "n_model_files": len(self.model_files),
            "n_zero_files": len(self.zero_files),
            "n_params": len(self.parameter_names()),
        }
