from .loader import get_ext, has_ext  # noqa: F401
from .adam import FusedAdam  # noqa: F401
