from .layer import MoE  # noqa: F401
from .sharded_moe import TopKGate, MOELayer  # noqa: F401
from .experts import Experts  # noqa: F401
from .utils import (has_moe_layers, is_moe_param,  # noqa: F401
                    split_params_into_different_moe_groups_for_optimizer)
from .mappings import drop_tokens, gather_tokens  # noqa: F401
