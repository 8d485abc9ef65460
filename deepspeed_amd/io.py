"""Streaming checkpoint file writers.

Parity: reference `deepspeed/io/` (BaseFileWriter, PyFileWriter,
MockFileWriter, FastFileWriter over single/double pinned IO buffers).
`FastFileWriter` is a file-like object (pass it to `torch.save`): bytes
fill a pinned staging buffer; full halves drain to disk through the
aio engine's offset API (`async_pwrite_at`, O_DIRECT-capable thread
pool) while serialization keeps filling the other half — serialization
and disk writes overlap, which is the whole point of double buffering.
The unaligned tail goes through one plain pwrite at close and the file
is truncated to its exact length.
"""
import os
import time

import torch

ALIGN = 4096


class BaseFileWriter:
    """File-like: only `write` is required by torch.save."""

    def __init__(self, file_path):
        self.file_path = file_path
        self._stats = {"write_bytes": 0, "write_sec": 0.0,
                       "fill_count": 0, "aio_bytes": 0, "slow_bytes": 0}

    def write(self, data):  # -> bytes written
        raise NotImplementedError

    def flush(self):
        pass

    def close(self):
        pass

    def fini(self):
        self.close()
        return self._stats

    @property
    def stats(self):
        return dict(self._stats)

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()


class MockFileWriter(BaseFileWriter):
    """Counts bytes, writes nothing (serialization-cost measurement)."""

    def write(self, data):
        self._stats["write_bytes"] += len(data)
        return len(data)


class PyFileWriter(BaseFileWriter):
    """Plain buffered python file (the slow baseline)."""

    def __init__(self, file_path):
        super().__init__(file_path)
        self._f = open(file_path, "wb")

    def write(self, data):
        t0 = time.perf_counter()
        n = self._f.write(data)
        self._stats["write_sec"] += time.perf_counter() - t0
        self._stats["write_bytes"] += n
        self._stats["slow_bytes"] += n
        return n

    def close(self):
        if not self._f.closed:
            self._f.close()


class FastFileWriter(BaseFileWriter):
    """Double-buffered pinned staging -> aio offset writes.

    pinned_tensor: optional uint8 staging tensor (pinned on CUDA builds);
    its length must be a multiple of 2*ALIGN. aio handle optional — the
    fallback drains with os.pwrite (still overlapped via async thread
    when the handle exists, synchronous otherwise).
    """

    def __init__(self, file_path, handle=None, pinned_tensor=None,
                 buffer_bytes=1 << 24, double_buffer=True):
        super().__init__(file_path)
        if pinned_tensor is None:
            pin = torch.cuda.is_available()
            pinned_tensor = torch.empty(buffer_bytes, dtype=torch.uint8,
                                        pin_memory=pin)
        nbuf = 2 if double_buffer else 1
        assert pinned_tensor.numel() % (nbuf * ALIGN) == 0, \
            "staging buffer must be a multiple of 2*4096 bytes"
        half = pinned_tensor.numel() // nbuf
        self._bufs = [pinned_tensor[i * half:(i + 1) * half]
                      for i in range(nbuf)]
        self._views = [memoryview(b.numpy()) for b in self._bufs]
        self._handle = handle
        self._fill = 0          # buffer being filled
        self._fill_off = 0      # bytes filled in it
        self._file_off = 0      # where the NEXT drain lands
        self._draining = False  # an async drain is outstanding
        self._total = 0
        # create/empty the target so offset writes land in a fresh file
        open(file_path, "wb").close()
        self._fd = os.open(file_path, os.O_WRONLY)

    def write(self, data):
        mv = memoryview(data).cast("B") if not isinstance(data, memoryview) \
            else data.cast("B")
        n = len(mv)
        done = 0
        while done < n:
            buf = self._views[self._fill]
            take = min(n - done, len(buf) - self._fill_off)
            buf[self._fill_off:self._fill_off + take] = mv[done:done + take]
            self._fill_off += take
            done += take
            if self._fill_off == len(buf):
                self._drain_full()
        self._total += n
        self._stats["write_bytes"] += n
        return n

    def _drain_full(self):
        """Current buffer is full: push it at the current file offset and
        switch to the other one (waiting for its previous drain first)."""
        t0 = time.perf_counter()
        buf = self._bufs[self._fill]
        nb = buf.numel()
        if self._handle is not None:
            if self._draining:
                self._handle.wait()
            self._handle.async_pwrite_at(buf, self.file_path,
                                         self._file_off)
            self._draining = True
            if len(self._bufs) == 1:
                # single-buffer mode: the same memory is refilled next,
                # so the drain must complete before write() continues
                self._handle.wait()
                self._draining = False
            self._stats["aio_bytes"] += nb
        else:
            os.pwrite(self._fd, self._views[self._fill], self._file_off)
            self._stats["slow_bytes"] += nb
        self._file_off += nb
        self._fill = (self._fill + 1) % len(self._bufs)
        self._fill_off = 0
        self._stats["fill_count"] += 1
        self._stats["write_sec"] += time.perf_counter() - t0

    def flush(self):
        """Drain the partial tail (plain pwrite: unaligned) and settle."""
        if self._draining:
            self._handle.wait()
            self._draining = False
        if self._fill_off:
            t0 = time.perf_counter()
            os.pwrite(self._fd, self._views[self._fill][:self._fill_off],
                      self._file_off)
            self._file_off += self._fill_off
            self._stats["slow_bytes"] += self._fill_off
            self._stats["write_sec"] += time.perf_counter() - t0
            self._fill_off = 0

    def close(self):
        if self._fd is None:
            return
        self.flush()
        os.ftruncate(self._fd, self._total)  # exact length (O_DIRECT pads)
        os.close(self._fd)
        self._fd = None


def save_with_fast_writer(obj, file_path, handle=None, **kw):
    """torch.save through a FastFileWriter; returns its stats."""
    w = FastFileWriter(file_path, handle=handle, **kw)
    try:
        torch.save(obj, w)
    finally:
        w.close()
    return w.stats
