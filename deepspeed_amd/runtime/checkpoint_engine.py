"""Pluggable checkpoint serializers.

Parity: reference `runtime/checkpoint_engine/` (torch, fast, decoupled
engines; Nebula/DataStates are external services — config stubs only).

FastCheckpointEngine splits a checkpoint into a pickled metadata file plus
one raw tensor-payload file streamed through the C++ O_DIRECT thread-pool
(ops/csrc/aio.cpp) — bulk bytes bypass torch.save's pickler.
"""
import os
import pickle

import torch

from ..utils.logging import log_dist


class CheckpointEngine:
    def __init__(self, config_params=None):
        pass

    def create(self, tag):
        pass

    def save(self, state_dict, path):
        raise NotImplementedError

    def load(self, path, map_location=None):
        raise NotImplementedError

    def commit(self, tag):
        return True


class TorchCheckpointEngine(CheckpointEngine):
    def save(self, state_dict, path):
        torch.save(state_dict, path)

    def load(self, path, map_location=None):
        return torch.load(path, map_location=map_location,
                          weights_only=False)


class FastCheckpointEngine(CheckpointEngine):
    """Metadata pickle + raw tensor payload via the aio engine."""

    def __init__(self, config_params=None):
        super().__init__(config_params)
        from ..ops.loader import get_ext
        ext = get_ext(required=False)
        self.handle = (ext.aio_handle(1 << 20, 8, False, False, 8)
                       if ext is not None else None)

    def save(self, state_dict, path):
        meta, payload = self._flatten(state_dict)
        cpu_payload = torch.cat([t.reshape(-1).view(torch.uint8)
                                 for t in payload]) if payload else \
            torch.empty(0, dtype=torch.uint8)
        if self.handle is not None and cpu_payload.numel() > 0:
            self.handle.sync_pwrite(cpu_payload, path + ".bin")
        else:
            with open(path + ".bin", "wb") as f:
                f.write(cpu_payload.numpy().tobytes())
        with open(path, "wb") as f:
            pickle.dump(meta, f)

    def load(self, path, map_location=None):
        with open(path, "rb") as f:
            meta = pickle.load(f)
        size = os.path.getsize(path + ".bin")
        buf = torch.empty(size, dtype=torch.uint8)
        if self.handle is not None and size > 0:
            self.handle.sync_pread(buf, path + ".bin")
        elif size > 0:
            import numpy as np
            buf = torch.from_numpy(
                np.fromfile(path + ".bin", dtype=np.uint8))
        return self._unflatten(meta, buf)

    # -- (de)construction ---------------------------------------------------
    def _flatten(self, obj, payload=None):
        if payload is None:
            payload = []
            meta = self._flatten(obj, payload)
            return meta, payload
        if torch.is_tensor(obj):
            t = obj.detach().cpu().contiguous()
            nbytes = t.numel() * t.element_size()
            entry = {"__tensor__": True, "dtype": str(t.dtype),
                     "shape": list(t.shape), "nbytes": nbytes,
                     "offset": sum(p.numel() * p.element_size()
                                   for p in payload)}
            payload.append(t)
            return entry
        if isinstance(obj, dict):
            return {"__dict__": {k: self._flatten(v, payload)
                                 for k, v in obj.items()}}
        if isinstance(obj, (list, tuple)):
            return {"__list__": [self._flatten(v, payload) for v in obj],
                    "__tuple__": isinstance(obj, tuple)}
        return {"__obj__": obj}

    def _unflatten(self, meta, buf):
        if isinstance(meta, dict) and meta.get("__tensor__"):
            dtype = getattr(torch, meta["dtype"].replace("torch.", ""))
            nbytes = meta["nbytes"]
            raw = buf[meta["offset"]:meta["offset"] + nbytes]
            if nbytes == 0:
                return torch.empty(meta["shape"], dtype=dtype)
            return raw.clone().view(dtype).reshape(meta["shape"])
        if isinstance(meta, dict) and "__dict__" in meta:
            return {k: self._unflatten(v, buf)
                    for k, v in meta["__dict__"].items()}
        if isinstance(meta, dict) and "__list__" in meta:
            vals = [self._unflatten(v, buf) for v in meta["__list__"]]
            return tuple(vals) if meta.get("__tuple__") else vals
        return meta["__obj__"]


class DecoupledCheckpointEngine(CheckpointEngine):
    """Background-process writer (ref
    runtime/checkpoint_engine/decoupled_checkpoint_engine.py): save()
    hands the CPU-materialized state to a spawned writer process and
    returns immediately; commit(tag) blocks until every file of that tag
    is durably on disk. Training overlaps with serialization."""

    def __init__(self, config_params=None):
        super().__init__(config_params)
        import multiprocessing as mp
        ctx = mp.get_context("spawn")
        self._queue = ctx.Queue()
        self._done = ctx.Queue()
        self._proc = ctx.Process(target=self._writer_loop,
                                 args=(self._queue, self._done),
                                 daemon=True)
        self._proc.start()
        self._pending = 0

    @staticmethod
    def _writer_loop(q, done):
        import torch as _torch
        while True:
            item = q.get()
            if item is None:
                break
            path, payload = item
            _torch.save(payload, path)
            done.put(path)

    @staticmethod
    def _to_cpu(obj):
        if torch.is_tensor(obj):
            return obj.detach().cpu().clone()
        if isinstance(obj, dict):
            return {k: DecoupledCheckpointEngine._to_cpu(v)
                    for k, v in obj.items()}
        if isinstance(obj, (list, tuple)):
            t = [DecoupledCheckpointEngine._to_cpu(v) for v in obj]
            return type(obj)(t) if not isinstance(obj, tuple) else tuple(t)
        return obj

    def save(self, state_dict, path):
        # snapshot to host BEFORE returning: the trainer may mutate the
        # live tensors on the very next step
        payload = self._to_cpu(state_dict)
        self._queue.put((path, payload))
        self._pending += 1

    def commit(self, tag):
        while self._pending:
            self._done.get()
            self._pending -= 1
        return True

    def load(self, path, map_location=None):
        self.commit(None)  # drain in-flight writes first
        return torch.load(path, map_location=map_location,
                          weights_only=False)

    def close(self):
        if self._proc.is_alive():
            self._queue.put(None)
            self._proc.join(10)

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


def make_checkpoint_engine(name="torch", config_params=None):
    if name in (None, "torch"):
        return TorchCheckpointEngine(config_params)
    if name == "fast":
        return FastCheckpointEngine(config_params)
    if name == "decoupled":
        return DecoupledCheckpointEngine(config_params)
    raise ValueError(f"unknown checkpoint engine {name} "
                     "(nebula/datastates are external services)")


class NebulaCheckpointEngine(CheckpointEngine):
    """Parity stub for reference `runtime/checkpoint_engine/nebula_*`.

    Nebula is an Azure-proprietary asynchronous checkpoint service; its
    client SDK ("torch_nebula") is not available on this stack. The
    MI355X-native equivalent of its value proposition — async, tiered
    checkpoint writes that overlap training — is FastCheckpointEngine
    (O_DIRECT aio writer thread). Instantiating this class raises with
    that pointer rather than silently degrading.
    """

    def __init__(self, config_params=None):
        raise RuntimeError(
            "Nebula requires the proprietary torch_nebula SDK (Azure). "
            "Use checkpoint.engine=fast (FastCheckpointEngine) for async "
            "checkpointing on MI355X nodes.")


class DataStatesCheckpointEngine(CheckpointEngine):
    """Parity stub for reference DataStates-LLM engine (ANL host-memory
    async checkpointing). Same guidance as Nebula: FastCheckpointEngine
    provides the in-tree async path."""

    def __init__(self, config_params=None):
        raise RuntimeError(
            "datastates-llm is not installed; use checkpoint.engine=fast "
            "(FastCheckpointEngine) for async checkpointing.")
