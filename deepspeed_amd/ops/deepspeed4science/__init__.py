from .evoformer_attn import DS4Sci_EvoformerAttention  # noqa: F401
