from .stage_1_and_2 import ZeroStage12Optimizer  # noqa: F401
from .partition import Init  # noqa: F401
from .stage3_params import (GatheredParameters,  # noqa: F401
                            register_external_parameter,
                            unregister_external_parameter)
from .stage3 import ZeroStage3Optimizer  # noqa: F401
