from .optimized_linear import OptimizedLinear, LoRAConfig, QuantizationConfig  # noqa: F401
from .quantization import QuantizedParameter, QuantizedLinear  # noqa: F401
