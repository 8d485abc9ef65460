"""deepspeed.utils-compatible export surface."""
from .logging import logger, log_dist  # noqa: F401
from .timer import SynchronizedWallClockTimer, ThroughputTimer  # noqa: F401
from .tensor_fragment import (safe_get_full_fp32_param,  # noqa: F401
                              safe_get_full_grad,
                              safe_get_full_optimizer_state,
                              safe_set_full_fp32_param)
from .roctx import instrument_w_nvtx, instrument_w_roctx  # noqa: F401
from .init_on_device import OnDevice  # noqa: F401
