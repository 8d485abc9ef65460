"""`deepspeed.pipe` import-path alias (ref deepspeed/pipe/__init__.py)."""
from .runtime.pipe.module import (LayerSpec, PipelineModule,  # noqa: F401
                                  TiedLayerSpec)
