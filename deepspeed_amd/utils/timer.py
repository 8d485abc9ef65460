"""Wall-clock + device-event timers and throughput accounting.

Parity: reference `deepspeed/utils/timer.py:44` (SynchronizedWallClockTimer)
and `:199` (ThroughputTimer). MI355X-native: uses torch.cuda events directly
(HIP events on ROCm), no accelerator-abstraction layer.
"""
import time

import torch

from .logging import log_dist


def _cuda():
    return torch.cuda.is_available()


class _Timer:
    def __init__(self, name):
        self.name = name
        self.started = False
        self.elapsed_ = 0.0
        self._start_event = None
        self._stop_events = []  # (start_ev, stop_ev) pairs pending
        self._start_time = None

    def start(self):
        assert not self.started, f"timer {self.name} already started"
        if _cuda():
            ev = torch.cuda.Event(enable_timing=True)
            ev.record()
            self._start_event = ev
        else:
            self._start_time = time.time()
        self.started = True

    def stop(self, reset=False, record=True):
        assert self.started, f"timer {self.name} not started"
        if _cuda():
            ev = torch.cuda.Event(enable_timing=True)
            ev.record()
            self._stop_events.append((self._start_event, ev))
        else:
            self.elapsed_ += time.time() - self._start_time
        self.started = False

    def _drain(self):
        if self._stop_events:
            torch.cuda.synchronize()
            for s, e in self._stop_events:
                self.elapsed_ += s.elapsed_time(e) / 1000.0
            self._stop_events = []

    def elapsed(self, reset=True):
        """Elapsed seconds."""
        self._drain()
        val = self.elapsed_
        if reset:
            self.reset()
        return val

    def reset(self):
        self.elapsed_ = 0.0
        self._stop_events = []
        self.started = False

    def mean(self):
        return self.elapsed(reset=False)


class SynchronizedWallClockTimer:
    """Named timer registry; `elapsed` synchronizes the device."""

    def __init__(self):
        self.timers = {}

    def __call__(self, name):
        if name not in self.timers:
            self.timers[name] = _Timer(name)
        return self.timers[name]

    def has(self, name):
        return name in self.timers

    def log(self, names, normalizer=1.0, reset=True, ranks=None):
        assert normalizer > 0.0
        parts = []
        for name in names:
            if name in self.timers:
                ms = self.timers[name].elapsed(reset=reset) * 1000.0 / normalizer
                parts.append(f"{name}: {ms:.2f}ms")
        if parts:
            log_dist("time: " + " | ".join(parts), ranks=ranks or [0])

    @staticmethod
    def memory_usage():
        if not _cuda():
            return ""
        alloc = torch.cuda.memory_allocated() / 2**30
        max_alloc = torch.cuda.max_memory_allocated() / 2**30
        return f"mem: alloc={alloc:.2f}GB max={max_alloc:.2f}GB"


class ThroughputTimer:
    """Samples/s + TFLOPS accounting across steps (skips warmup steps)."""

    def __init__(self, batch_size, start_step=2, steps_per_output=None):
        self.batch_size = max(batch_size, 1)
        self.start_step = start_step
        self.steps_per_output = steps_per_output
        self.epoch_count = 0
        self.global_step_count = 0
        self.total_elapsed_time = 0.0
        self.step_elapsed_time = 0.0
        self._start = None
        self.started = False

    def start(self):
        self.started = True
        if _cuda():
            torch.cuda.synchronize()
        self._start = time.time()

    def stop(self, global_step=True, report_speed=True):
        if not self.started:
            return
        self.started = False
        if global_step:
            self.global_step_count += 1
        if _cuda():
            torch.cuda.synchronize()
        dur = time.time() - self._start
        if self.global_step_count > self.start_step:
            self.total_elapsed_time += dur
            self.step_elapsed_time += dur

    def avg_samples_per_sec(self):
        if self.global_step_count > self.start_step and self.total_elapsed_time > 0:
            steps = self.global_step_count - self.start_step
            return self.batch_size * steps / self.total_elapsed_time
        return 0.0
