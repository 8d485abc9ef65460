"""SuperOffload — full-model-state offload for coherent-memory superchips.

Parity stub: reference `runtime/superoffload/superoffload_stage3.py`
(SuperOffloadOptimizer_Stage3) targets cache-coherent CPU<->GPU memory
(NVIDIA GH200 / AMD MI300A APUs), where the optimizer can walk GPU-
resident gradients from host Adam workers without explicit transfers,
plus speculative step + rollback on overflow.

The MI355X is a DISCRETE GPU: host memory is reached over PCIe/IF, not
a coherent fabric, so the superchip fast path does not exist on this
hardware. The pieces of SuperOffload that DO apply here are implemented
elsewhere in this package:
  - bucketed, pinned, overlapped optimizer offload: ZeRO-3
    `offload_optimizer` path (runtime/zero/stage3.py) with OpenMP
    cpu_adam workers (ops/csrc/cpu_adam.cpp) and NUMA binding
    (utils/numa.py)
  - selective/asynchronous CPU updates: ZenFlow (runtime/zenflow.py)

`SuperOffloadOptimizer_Stage3` raises with that guidance so configs
written for the reference fail loudly instead of silently degrading.
"""


class SuperOffloadOptimizer_Stage3:
    def __init__(self, *a, **kw):
        raise RuntimeError(
            "SuperOffload targets coherent-memory superchips (GH200/"
            "MI300A); on discrete MI355X nodes use zero_optimization."
            "offload_optimizer (ZeRO-3 offload with OpenMP cpu_adam + "
            "NUMA binding) or zero_optimization.zenflow for selective "
            "asynchronous CPU updates.")
