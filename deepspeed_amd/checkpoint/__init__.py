from .universal import (ds_to_universal,  # noqa: F401
                        load_universal_into_optimizer)
from .utils import clone_tensors_for_torch_save  # noqa: F401
from .inspect import DeepSpeedCheckpoint  # noqa: F401
