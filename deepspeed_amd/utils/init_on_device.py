"""Construct modules directly on a device/dtype.

Parity: reference `utils/init_on_device.py` (OnDevice). The reference
monkeypatches torch factory functions; torch 2.x has first-class
default-device/dtype plumbing, so this is a thin context manager over
`torch.set_default_device` + `set_default_dtype`. `device="meta"`
builds shape-only modules (no host RAM) for ZeRO-3 `Init` or for
memory estimation of trillion-scale configs.
"""
import torch


class OnDevice:
    def __init__(self, dtype=None, device="meta", enabled=True):
        self.dtype = dtype
        self.device = device
        self.enabled = enabled
        self._prev_dtype = None
        self._dev_ctx = None

    def __enter__(self):
        if not self.enabled:
            return self
        if self.dtype is not None and self.dtype.is_floating_point:
            self._prev_dtype = torch.get_default_dtype()
            torch.set_default_dtype(self.dtype)
        self._dev_ctx = torch.device(self.device)
        self._dev_ctx.__enter__()
        return self

    def __exit__(self, *exc):
        if self._dev_ctx is not None:
            self._dev_ctx.__exit__(*exc)
            self._dev_ctx = None
        if self._prev_dtype is not None:
            torch.set_default_dtype(self._prev_dtype)
            self._prev_dtype = None
        return False
