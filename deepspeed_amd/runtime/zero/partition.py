"""ZeRO-3 parameter partitioning (zero.Init).

Implemented in stage3.py build phase — this module hosts the `Init` context
manager and per-parameter shard state. See stage3 docstring for design.
"""
from .stage3_params import Init, is_zero_param, ZeroParamStatus  # noqa: F401
