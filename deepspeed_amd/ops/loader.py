"""In-tree HIP extension loader.

The extension `deepspeed_amd/ops/_hip_ops*.so` is built ahead of time for
gfx950 by `setup.py build_ext --inplace` (driven by `__graft_entry__.build`).
Policy: on a GPU box the extension is REQUIRED — ops raise rather than fall
back to eager PyTorch, so a silent-slow path can never masquerade as the
native one. On CPU (unit tests, no GPU in the dev container) torch fallbacks
are used.
"""
import torch

_ext = None
_tried = False


def has_ext():
    global _ext, _tried
    if not _tried:
        _tried = True
        try:
            from . import _hip_ops  # noqa: F401
            _ext = _hip_ops
        except ImportError:
            _ext = None
    return _ext is not None


def get_ext(required=None):
    """Return the HIP extension module.

    required=None => required iff CUDA/HIP device is available.
    """
    if required is None:
        required = torch.cuda.is_available()
    if has_ext():
        return _ext
    if required:
        raise RuntimeError(
            "deepspeed_amd HIP extension (_hip_ops) is not built. "
            "Run `python setup.py build_ext --inplace` (gfx950) — ops will "
            "not silently fall back to eager PyTorch on a GPU.")
    return None
