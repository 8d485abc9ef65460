from .loader import get_ext, has_ext  # noqa: F401
from .adam import FusedAdam  # noqa: F401
from .adagrad import FusedAdagrad, DeepSpeedCPUAdagrad  # noqa: F401
from .lamb import FusedLamb  # noqa: F401
from .lion import FusedLion, DeepSpeedCPULion  # noqa: F401
