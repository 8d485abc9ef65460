from .auto_tp import apply_tensor_parallel  # noqa: F401
from .layers import LinearAllreduce, LinearLayer  # noqa: F401
