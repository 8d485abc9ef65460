"""Data sampling: mmap indexed dataset + curriculum-aware loader.

Parity: reference `runtime/data_pipeline/data_sampling/indexed_dataset.py`
(mmap .bin/.idx token datasets) and `data_sampler.py`
(DeepSpeedDataSampler). Lite implementations: binary token storage with an
index of document offsets, and a loader wrapper that applies the
curriculum difficulty (sequence length) per step.
"""
import os
import struct

import numpy as np
import torch

_MAGIC = b"DSAMDIDX"
_DTYPE_CODES = {np.uint16: 1, np.int32: 2, np.int64: 3}
_CODE_DTYPES = {1: np.uint16, 2: np.int32, 3: np.int64}


class IndexedDatasetBuilder:
    """Writes <path>.bin (tokens) and <path>.idx (doc offsets)."""

    def __init__(self, path, dtype=np.uint16):
        self.path = path
        self.dtype = dtype
        self._bin = open(path + ".bin", "wb")
        self.sizes = []

    def add_item(self, tokens):
        arr = np.asarray(tokens, dtype=self.dtype)
        self._bin.write(arr.tobytes())
        self.sizes.append(len(arr))

    def finalize(self):
        self._bin.close()
        with open(self.path + ".idx", "wb") as f:
            f.write(_MAGIC)
            code = next(c for d, c in _DTYPE_CODES.items()
                        if d == self.dtype)
            f.write(struct.pack("<BQ", code, len(self.sizes)))
            np.asarray(self.sizes, dtype=np.int64).tofile(f)


class IndexedDataset(torch.utils.data.Dataset):
    """mmap-backed read of an IndexedDatasetBuilder output."""

    def __init__(self, path):
        with open(path + ".idx", "rb") as f:
            assert f.read(8) == _MAGIC, "bad index file"
            code, n = struct.unpack("<BQ", f.read(9))
            self.dtype = _CODE_DTYPES[code]
            self.sizes = np.fromfile(f, dtype=np.int64, count=n)
        self.offsets = np.concatenate([[0], np.cumsum(self.sizes)])
        self.data = np.memmap(path + ".bin", dtype=self.dtype, mode="r")

    def __len__(self):
        return len(self.sizes)

    def __getitem__(self, idx):
        lo, hi = self.offsets[idx], self.offsets[idx + 1]
        return torch.from_numpy(self.data[lo:hi].astype(np.int64))


class DeepSpeedDataSampler:
    """Curriculum-aware batch loader wrapper (ref data_sampler.py:349).

    Wraps any iterable of (input_ids[, labels]) batches; truncates the
    sequence dimension to the current curriculum difficulty, stepping the
    schedule once per batch.
    """

    def __init__(self, dataloader, curriculum_scheduler, seq_dim=1):
        self.dataloader = dataloader
        self.scheduler = curriculum_scheduler
        self.seq_dim = seq_dim
        self.global_step = 0

    def _truncate(self, t, n):
        if torch.is_tensor(t) and t.dim() > self.seq_dim and \
                t.shape[self.seq_dim] > n:
            return t.narrow(self.seq_dim, 0, n)
        return t

    def __iter__(self):
        for batch in self.dataloader:
            n = self.scheduler.update_difficulty(self.global_step)
            self.global_step += 1
            if isinstance(batch, (tuple, list)):
                yield type(batch)(self._truncate(x, n) for x in batch)
            else:
                yield self._truncate(batch, n)

    def __len__(self):
        return len(self.dataloader)

    def state_dict(self):
        return {"global_step": self.global_step,
                "scheduler": self.scheduler.state_dict()}

    def load_state_dict(self, sd):
        self.global_step = sd["global_step"]
        self.scheduler.load_state_dict(sd["scheduler"])
