"""Checkpoint shard loader with model-parallel resharding.

Parity: reference `runtime/state_dict_factory.py:21` (SDLoaderFactory,
MegatronSDLoader): given a list of checkpoint files written at one
MP degree, load the slice for (mp_world_size, mp_rank) — merging files
when shrinking MP, splitting tensors when growing it.
"""
import json
from abc import ABC, abstractmethod

import torch

AUTO_MODULE_KEY = "auto"


class SDLoaderFactory:
    @staticmethod
    def get_sd_loader_json(json_file, checkpoint_engine=None):
        if isinstance(json_file, str):
            with open(json_file) as f:
                data = json.load(f)
        else:
            assert isinstance(json_file, dict)
            data = json_file
        sd_type = data["type"]
        if sd_type.lower() in ("bloom", "ds_model"):
            return data
        return SDLoaderFactory.get_sd_loader(
            data["checkpoints"], checkpoint_engine, sd_type,
            data.get("version"))

    @staticmethod
    def get_sd_loader(ckpt_list, checkpoint_engine=None,
                      sd_type="Megatron", version=None):
        if sd_type == "Megatron":
            return MegatronSDLoader(ckpt_list, version)
        raise ValueError(f"unsupported checkpoint type {sd_type!r}")


class SDLoaderBase(ABC):
    def __init__(self, ckpt_list, version=None):
        assert ckpt_list, "empty checkpoint list"
        self.ckpt_list = list(ckpt_list)
        self.version = version
        self.module_key = AUTO_MODULE_KEY

    def _load_file(self, path):
        return torch.load(path, map_location="cpu", weights_only=False)

    def _module_sd(self, sd):
        if self.module_key == AUTO_MODULE_KEY:
            for k in ("module", "model"):
                if k in sd:
                    return sd[k]
            return sd
        return sd[self.module_key] if self.module_key else sd

    def load(self, mp_world_size, mp_rank, module_key=AUTO_MODULE_KEY):
        """Returns (load_path, merged/split client_sd)."""
        self.module_key = module_key
        n = len(self.ckpt_list)
        if n == mp_world_size:
            path = self.ckpt_list[mp_rank]
            sd = self._load_file(path)
            return path, self._module_sd(sd)
        if n > mp_world_size:
            assert n % mp_world_size == 0
            return self._merge(mp_world_size, mp_rank)
        assert mp_world_size % n == 0
        return self._split(mp_world_size, mp_rank)

    @abstractmethod
    def _merge(self, mp_world_size, mp_rank):
        ...

    @abstractmethod
    def _split(self, mp_world_size, mp_rank):
        ...


class MegatronSDLoader(SDLoaderBase):
    """Megatron-style column/row-parallel tensor resharding by key
    pattern (ref MegatronSDLoader.merge_state_dict/split_state_dict):
    qkv/mlp-in weights+biases concat/split on dim 0; row-parallel
    (dense/mlp-out) weights on dim 1; embeddings on dim 0; everything
    else replicated."""

    CAT0 = ("query_key_value", "dense_h_to_4h", "attention.qkv",
            "mlp.gate_proj", "mlp.up_proj", "embed", "lm_head")
    CAT1 = ("attention.dense", "dense_4h_to_h", "mlp.down_proj",
            "o_proj")

    def _axis(self, key, tensor):
        if any(t in key for t in self.CAT0):
            return 0
        if any(t in key for t in self.CAT1) and tensor.dim() == 2:
            return 1
        return None

    def _merge(self, mp_world_size, mp_rank):
        ratio = len(self.ckpt_list) // mp_world_size
        files = self.ckpt_list[mp_rank * ratio:(mp_rank + 1) * ratio]
        sds = [self._module_sd(self._load_file(f)) for f in files]
        out = {}
        for k in sds[0]:
            ts = [sd[k] for sd in sds]
            ax = self._axis(k, ts[0]) if torch.is_tensor(ts[0]) else None
            if ax is None:
                out[k] = ts[0]
            elif ts[0].dim() == 1:
                out[k] = torch.cat(ts, dim=0) if ax == 0 else ts[0]
            else:
                out[k] = torch.cat(ts, dim=ax)
        return files[0], out

    def _split(self, mp_world_size, mp_rank):
        ratio = mp_world_size // len(self.ckpt_list)
        fidx = mp_rank // ratio
        sub = mp_rank % ratio
        path = self.ckpt_list[fidx]
        sd = self._module_sd(self._load_file(path))
        out = {}
        for k, t in sd.items():
            ax = self._axis(k, t) if torch.is_tensor(t) else None
            if ax is None or (t.dim() == 1 and ax != 0):
                out[k] = t
            else:
                ax_eff = 0 if t.dim() == 1 else ax
                out[k] = torch.chunk(t, ratio, dim=ax_eff)[sub].contiguous()
        return path, out
