"""PipelineModule placeholder — full 1F1B pipeline implemented later this
round (see runtime/pipe/engine.py when present)."""
import torch


class LayerSpec:
    def __init__(self, typename, *args, **kwargs):
        self.typename = typename
        self.module_args = args
        self.module_kwargs = kwargs

    def build(self):
        return self.typename(*self.module_args, **self.module_kwargs)


class PipelineModule(torch.nn.Module):
    def __init__(self, *args, **kwargs):
        raise NotImplementedError(
            "PipelineModule: pipeline parallelism lands later this round")

    def mpu(self):
        return None
