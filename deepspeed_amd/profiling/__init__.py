from .flops_profiler import FlopsProfiler  # noqa: F401
