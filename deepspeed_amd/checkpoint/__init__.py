from .universal import (ds_to_universal,  # noqa: F401
                        load_universal_into_optimizer)
