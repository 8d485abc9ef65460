from .module import PipelineModule, LayerSpec, TiedLayerSpec  # noqa: F401
from .topology import (PipeDataParallelTopology,  # noqa: F401
                       PipelineParallelGrid, PipeModelDataParallelTopology,
                       ProcessTopology)
