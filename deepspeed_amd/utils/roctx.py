"""rocTX range instrumentation.

Parity: reference `deepspeed/utils/nvtx.py` (`instrument_w_nvtx`). On
torch-rocm, `torch.cuda.nvtx` maps onto rocTX, so ranges show up in
rocprofv3 timelines (`--marker-trace` domain; do NOT combine that with
--pmc in one run). No-op overhead when no profiler is attached is one
dict lookup + two library calls per wrapped function.
"""
import functools

import torch

enable_nvtx = True


def instrument_w_nvtx(func):
    """Decorator: wrap `func` in a rocTX range named after it."""

    @functools.wraps(func)
    def wrapped(*args, **kwargs):
        if enable_nvtx and torch.cuda.is_available():
            torch.cuda.nvtx.range_push(func.__qualname__)
            try:
                return func(*args, **kwargs)
            finally:
                torch.cuda.nvtx.range_pop()
        return func(*args, **kwargs)

    return wrapped


# rocTX alias — same implementation, AMD-native name
instrument_w_roctx = instrument_w_nvtx
