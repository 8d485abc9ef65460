from .layer import MoE  # noqa: F401
from .sharded_moe import TopKGate, MOELayer  # noqa: F401
from .experts import Experts  # noqa: F401
