from .module import PipelineModule, LayerSpec, TiedLayerSpec  # noqa: F401
from .topology import PipelineParallelGrid  # noqa: F401
