"""Inference-time model quantization entry (ref inference/quantization/
quantization.py init_quantization surface): swap eligible linears for
group-wise int8 dequant-on-the-fly modules."""
from ..runtime.weight_quantizer import WeightQuantization


def init_quantization(model, quantize_bits=8, groups=64,
                      exclude=("lm_head", "embed"),
                      mlp_extra_grouping=False):
    wq = WeightQuantization(mlp_extra_grouping=mlp_extra_grouping)
    model, n = wq.model_quantize(model, quantize_bits=quantize_bits,
                                 groups=groups, exclude=exclude)
    return model
