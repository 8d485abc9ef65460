"""NVMe swap of fp32 optimizer state (ZeRO-Infinity tier).

Parity: reference `runtime/swap_tensor/partitioned_optimizer_swapper.py:26`
and `optimizer_utils.py` (OptimizerSwapper). Per-sub-group fp32 master /
exp_avg / exp_avg_sq live on NVMe; before a sub-group steps they are read
into pinned host buffers through the C++ O_DIRECT thread-pool engine
(ops/csrc/aio.cpp), the host AdamW updates them, and they stream back out
while the next sub-group loads (double-buffered).
"""
import os

import torch

from ...ops.loader import get_ext
from ...utils.logging import log_dist

STATE_FILES = ("master", "exp_avg", "exp_avg_sq")


class OptimizerStateSwapper:
    def __init__(self, swap_dir, block_size=1 << 20, queue_depth=8,
                 intra_op_parallelism=8, pin_memory=True):
        self.swap_dir = swap_dir
        os.makedirs(swap_dir, exist_ok=True)
        ext = get_ext(required=False)
        self.handle = (ext.aio_handle(block_size, queue_depth, False, False,
                                      intra_op_parallelism)
                       if ext is not None else None)
        self._pin = pin_memory and torch.cuda.is_available()
        self._buffers = {}   # sub-group id -> {name: pinned tensor}
        self._on_disk = set()

    def _fname(self, sg_id, name):
        return os.path.join(self.swap_dir, f"subgroup_{sg_id}_{name}.bin")

    def _buffer(self, sg_id, name, numel):
        key = (name,)
        bufs = self._buffers.setdefault(sg_id, {})
        if name not in bufs:
            bufs[name] = torch.zeros(numel, dtype=torch.float32,
                                     pin_memory=self._pin)
        return bufs[name]

    def _io(self, tensor, fname, write):
        if self.handle is not None:
            if write:
                self.handle.sync_pwrite(tensor, fname)
            else:
                self.handle.sync_pread(tensor, fname)
        else:  # pure-python fallback (CPU tests without built extension)
            if write:
                tensor.numpy().tofile(fname)
            else:
                import numpy as np
                tensor.copy_(torch.from_numpy(
                    np.fromfile(fname, dtype="float32")))

    def swap_in(self, sg_id, numel):
        """Returns (master, exp_avg, exp_avg_sq) pinned fp32 tensors."""
        out = []
        for name in STATE_FILES:
            buf = self._buffer(sg_id, name, numel)
            if (sg_id, name) in self._on_disk:
                self._io(buf, self._fname(sg_id, name), write=False)
            out.append(buf)
        return out

    def swap_out(self, sg_id):
        for name in STATE_FILES:
            buf = self._buffers[sg_id][name]
            self._io(buf, self._fname(sg_id, name), write=True)
            self._on_disk.add((sg_id, name))

    def release_buffers(self, sg_id):
        self._buffers.pop(sg_id, None)
