"""get_accelerator() compatibility façade.

Parity: reference `deepspeed/accelerator/real_accelerator.py`. The
reference routes every device op through an accelerator-dispatch layer;
this framework targets ONE device (MI355X / torch-rocm "cuda"), so there
is no dispatch — but user code migrating from the reference calls
`get_accelerator()` constantly. This façade exposes that surface mapped
1:1 onto torch.cuda so such code runs unchanged. It is intentionally NOT
an abstraction point: everything returns the ROCm-CUDA implementation.
"""
import torch


class MI355XAccelerator:
    _name = "cuda"

    # -- identity ---------------------------------------------------------
    def device_name(self, device_index=None):
        if device_index is None:
            return "cuda"
        return f"cuda:{device_index}"

    def device(self, device_index=None):
        return torch.device("cuda", device_index)

    def communication_backend_name(self):
        return "nccl"  # RCCL

    def is_available(self):
        return torch.cuda.is_available()

    # -- device control ---------------------------------------------------
    def current_device(self):
        return torch.cuda.current_device()

    def current_device_name(self):
        return f"cuda:{torch.cuda.current_device()}"

    def set_device(self, device_index):
        torch.cuda.set_device(device_index)

    def device_count(self):
        return torch.cuda.device_count()

    def synchronize(self, device_index=None):
        torch.cuda.synchronize(device_index)

    # -- RNG --------------------------------------------------------------
    def manual_seed(self, seed):
        torch.cuda.manual_seed(seed)

    def manual_seed_all(self, seed):
        torch.cuda.manual_seed_all(seed)

    def initial_seed(self):
        return torch.cuda.initial_seed()

    def get_rng_state(self, device_index=None):
        return torch.cuda.get_rng_state(
            device_index if device_index is not None else "cuda")

    def set_rng_state(self, state, device_index=None):
        torch.cuda.set_rng_state(
            state, device_index if device_index is not None else "cuda")

    # -- streams / events -------------------------------------------------
    def Stream(self, *a, **kw):
        return torch.cuda.Stream(*a, **kw)

    def stream(self, s):
        return torch.cuda.stream(s)

    def current_stream(self, device_index=None):
        return torch.cuda.current_stream(device_index)

    def default_stream(self, device_index=None):
        return torch.cuda.default_stream(device_index)

    def Event(self, **kw):
        return torch.cuda.Event(**kw)

    # -- memory -----------------------------------------------------------
    def empty_cache(self):
        torch.cuda.empty_cache()

    def memory_allocated(self, device_index=None):
        return torch.cuda.memory_allocated(device_index)

    def max_memory_allocated(self, device_index=None):
        return torch.cuda.max_memory_allocated(device_index)

    def reset_peak_memory_stats(self, device_index=None):
        torch.cuda.reset_peak_memory_stats(device_index)

    def memory_reserved(self, device_index=None):
        return torch.cuda.memory_reserved(device_index)

    def total_memory(self, device_index=None):
        return torch.cuda.get_device_properties(
            device_index or 0).total_memory

    def available_memory(self, device_index=None):
        free, _ = torch.cuda.mem_get_info(device_index)
        return free

    # -- dtype / capability ----------------------------------------------
    def is_bf16_supported(self):
        return True

    def is_fp16_supported(self):
        return True

    def supported_dtypes(self):
        return [torch.float32, torch.bfloat16, torch.float16]

    # -- tensor helpers ---------------------------------------------------
    def pin_memory(self, tensor, align_bytes=1):
        return tensor.pin_memory()

    def is_pinned(self, tensor):
        return tensor.is_pinned()

    def on_accelerator(self, tensor):
        return tensor.is_cuda

    # -- profiling --------------------------------------------------------
    def range_push(self, msg):
        torch.cuda.nvtx.range_push(msg)  # rocTX on ROCm

    def range_pop(self):
        torch.cuda.nvtx.range_pop()


_ACCEL = MI355XAccelerator()


def get_accelerator():
    return _ACCEL
