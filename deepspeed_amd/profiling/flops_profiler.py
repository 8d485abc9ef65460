"""FLOPs / MACs / latency profiler with a per-module tree report.

Parity: reference `profiling/flops_profiler/profiler.py:30` (FlopsProfiler).
Module-hook based: counts the dominant ops (linear, matmul/sdpa via module
formulas, embeddings, norms, elementwise) and wall-clock per module.
"""
import time
from collections import defaultdict

import torch
import torch.nn as nn

from ..utils.logging import logger


def _num_to_string(num, precision=2):
    for unit, div in (("T", 1e12), ("G", 1e9), ("M", 1e6), ("K", 1e3)):
        if num >= div:
            return f"{num / div:.{precision}f} {unit}"
    return str(num)


# ---------------------------------------------------------------- functional
# patching (ref profiler.py:893 _patch_functionals, :952
# _patch_tensor_methods): counts FLOPs at the torch.nn.functional /
# torch-op level so non-module math (SDPA, einsum in MoE dispatch, bare
# matmuls) is attributed to the module executing it.
_module_stack = []
_orphan = [0]
_patched = {}


def _add_flops(f):
    if _module_stack:
        _module_stack[-1].__flops__ += f
    else:
        _orphan[0] += f


def _wrap(mod_obj, name, flop_fn):
    orig = getattr(mod_obj, name)
    _patched[(mod_obj, name)] = orig

    def wrapper(*args, **kwargs):
        out = orig(*args, **kwargs)
        try:
            _add_flops(flop_fn(out, *args, **kwargs))
        except Exception:
            pass
        return out

    setattr(mod_obj, name, wrapper)


def _linear_flops(out, x, w, bias=None, *a, **k):
    n = x.numel() // x.shape[-1]
    f = 2 * n * w.shape[0] * w.shape[1]
    if bias is not None:
        f += n * w.shape[0]
    return f


def _matmul_flops(out, a, b, *args, **k):
    # out shape [..., m, n]; contraction dim = a.shape[-1]
    return 2 * out.numel() * a.shape[-1]


def _baddbmm_flops(out, inp, a, b, *args, **k):
    return 2 * out.numel() * a.shape[-1] + out.numel()


def _sdpa_flops(out, q, k, v, *a, **kw):
    B_H = q.numel() // (q.shape[-1] * q.shape[-2])
    sq, skv, d = q.shape[-2], k.shape[-2], q.shape[-1]
    return B_H * (4 * sq * skv * d + 5 * sq * skv)


def _einsum_flops(out, eq, *ops, **k):
    if not isinstance(eq, str) or len(ops) != 2:
        return 0
    lhs, _, _ = eq.partition("->")
    terms = lhs.split(",")
    dims = {}
    for t, op in zip(terms, ops):
        for ch, n in zip(t.replace(" ", ""), op.shape):
            dims[ch] = n
    total = 1
    for n in dims.values():
        total *= n
    return 2 * total


def _patch_functionals():
    import torch.nn.functional as F
    _wrap(F, "linear", _linear_flops)
    _wrap(F, "scaled_dot_product_attention", _sdpa_flops)
    _wrap(torch, "matmul", _matmul_flops)
    _wrap(torch, "bmm", _matmul_flops)
    _wrap(torch, "baddbmm", _baddbmm_flops)
    _wrap(torch, "einsum", _einsum_flops)
    _wrap(torch.Tensor, "__matmul__", _matmul_flops)


def _unpatch_functionals():
    for (mod_obj, name), orig in list(_patched.items()):
        setattr(mod_obj, name, orig)
    _patched.clear()


def _module_flops(module, inputs, output, functional_counting=False):
    """fwd MACs*2 for common module types (0 for containers). With
    functional counting active, GEMM-backed modules return 0 (their math
    is counted at the functional level — no double count)."""
    x = inputs[0] if inputs else None
    if functional_counting and isinstance(
            module, (nn.Linear, nn.Conv1d, nn.Conv2d)):
        return 0
    if functional_counting and module.__class__.__name__ in             ("LlamaAttention",):
        return 0
    if isinstance(module, nn.Linear):
        n = x.numel() // x.shape[-1]
        f = 2 * n * module.in_features * module.out_features
        if module.bias is not None:
            f += n * module.out_features
        return f
    if isinstance(module, nn.Embedding):
        return 0
    if isinstance(module, (nn.LayerNorm,)):
        return x.numel() * 5
    if module.__class__.__name__ in ("LlamaRMSNorm", "RMSNorm"):
        return x.numel() * 4
    if isinstance(module, (nn.Conv1d, nn.Conv2d)):
        out = output[0] if isinstance(output, tuple) else output
        kernel_ops = module.in_channels // module.groups
        for k in module.kernel_size:
            kernel_ops *= k
        return 2 * out.numel() * kernel_ops
    if module.__class__.__name__ in ("LlamaAttention",):
        # qk^T + pv: 4 * B * S^2 * D * H (causal halves it)
        B, S = x.shape[0], x.shape[1]
        cfg = getattr(module, "cfg", None)
        if cfg is not None:
            return 4 * B * S * S * cfg.head_dim * \
                cfg.num_attention_heads // 2
    return 0


class FlopsProfiler:
    def __init__(self, model, ds_engine=None):
        self.model = model
        self.started = False
        self._hooks = []

    def start_profile(self, ignore_list=None, patch_functionals=True):
        self.reset()
        self._functional = patch_functionals
        if patch_functionals:
            _orphan[0] = 0
            _module_stack.clear()
            _patch_functionals()
        for name, mod in self.model.named_modules():
            if ignore_list and type(mod) in ignore_list:
                continue
            mod.__flops__ = 0
            mod.__params__ = sum(p.numel() for p in
                                 mod.parameters(recurse=False))
            mod.__latency__ = 0.0
            mod.__calls__ = 0
            h1 = mod.register_forward_pre_hook(self._pre_hook)
            h2 = mod.register_forward_hook(self._post_hook)
            self._hooks += [h1, h2]
        self.started = True

    def _pre_hook(self, mod, inputs):
        if getattr(self, "_functional", False):
            _module_stack.append(mod)
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        mod.__start__ = time.time()

    def _post_hook(self, mod, inputs, output):
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        mod.__latency__ += time.time() - getattr(mod, "__start__", time.time())
        mod.__calls__ += 1
        mod.__flops__ += _module_flops(
            mod, inputs, output,
            functional_counting=getattr(self, "_functional", False))
        if getattr(self, "_functional", False) and _module_stack and \
                _module_stack[-1] is mod:
            _module_stack.pop()

    def stop_profile(self):
        for h in self._hooks:
            h.remove()
        self._hooks = []
        if getattr(self, "_functional", False):
            _unpatch_functionals()
            _module_stack.clear()
            self._functional = False

    def reset(self):
        self.stop_profile()
        for mod in self.model.modules():
            for attr in ("__flops__", "__params__", "__latency__",
                         "__calls__", "__start__"):
                if hasattr(mod, attr):
                    delattr(mod, attr)

    def get_total_flops(self, as_string=False):
        total = sum(getattr(m, "__flops__", 0) for m in self.model.modules())
        return _num_to_string(total) + "FLOPs" if as_string else total

    def get_total_params(self, as_string=False):
        total = sum(p.numel() for p in self.model.parameters())
        return _num_to_string(total) if as_string else total

    def get_total_duration(self, as_string=False):
        root_lat = getattr(self.model, "__latency__", 0.0)
        return f"{root_lat*1000:.2f} ms" if as_string else root_lat

    def print_model_profile(self, profile_step=1, module_depth=-1,
                            top_modules=3, detailed=True, output_file=None):
        lines = []
        total_flops = self.get_total_flops()
        total_lat = max(self.get_total_duration(), 1e-9)
        lines.append("-" * 70)
        lines.append(f"Flops profiler report (step {profile_step})")
        lines.append(f"params: {self.get_total_params(True)}  "
                     f"fwd flops: {_num_to_string(total_flops)}  "
                     f"fwd latency: {total_lat*1000:.2f} ms  "
                     f"fwd FLOPS: {_num_to_string(total_flops/total_lat)}")
        if detailed:
            for name, mod in self.model.named_modules():
                depth = name.count(".")
                if module_depth >= 0 and depth > module_depth:
                    continue
                fl = getattr(mod, "__flops__", 0)
                lat = getattr(mod, "__latency__", 0.0)
                if fl == 0 and getattr(mod, "__params__", 0) == 0:
                    continue
                pad = "  " * depth
                lines.append(
                    f"{pad}{name or 'model'} ({type(mod).__name__}): "
                    f"{_num_to_string(fl)}FLOPs ({100*fl/max(total_flops,1):.1f}%), "
                    f"{lat*1000:.2f} ms ({100*lat/total_lat:.1f}%)")
        lines.append("-" * 70)
        report = "\n".join(lines)
        if output_file:
            with open(output_file, "w") as f:
                f.write(report)
        else:
            print(report)
        return report

    def end_profile(self):
        self.reset()


def get_model_profile(model, input_shape=None, args=(), kwargs=None,
                      print_profile=True, detailed=True, warm_up=1,
                      as_string=True):
    """One-shot profiling helper (ref profiler.py get_model_profile)."""
    prof = FlopsProfiler(model)
    kwargs = kwargs or {}
    if input_shape is not None:
        args = (torch.randn(*input_shape),)
    for _ in range(warm_up):
        model(*args, **kwargs)
    prof.start_profile()
    model(*args, **kwargs)
    flops = prof.get_total_flops(as_string)
    params = prof.get_total_params(as_string)
    if print_profile:
        prof.print_model_profile(detailed=detailed)
    prof.end_profile()
    return flops, None, params
