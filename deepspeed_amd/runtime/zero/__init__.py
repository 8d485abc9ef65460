from .stage_1_and_2 import ZeroStage12Optimizer  # noqa: F401
from .partition import Init  # noqa: F401
