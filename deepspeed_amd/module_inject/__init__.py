from .auto_tp import apply_tensor_parallel  # noqa: F401
from .layers import LinearAllreduce, LinearLayer  # noqa: F401
from .replace_module import (replace_module,  # noqa: F401
                             replace_transformer_layer)
