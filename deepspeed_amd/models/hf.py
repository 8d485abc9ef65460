"""HuggingFace checkpoint import for the native model family.

The native Llama modules deliberately mirror HF's parameter tree
(`model.layers.N.self_attn.q_proj.weight`, ...), so importing an HF
Llama checkpoint is a filtered state_dict load. This is the migration
path for users coming from the reference stack: load the HF weights,
then train with deepspeed_amd or serve with the native engines.
"""
import torch

from ..utils.logging import logger
from .llama import LlamaConfig, LlamaForCausalLM


def _hf_rope_theta(hf_config):
    # transformers >=5 keeps theta inside rope_scaling/rope_parameters;
    # older versions expose .rope_theta directly
    for holder in (getattr(hf_config, "rope_scaling", None) or {},
                   getattr(hf_config, "rope_parameters", None) or {}):
        if isinstance(holder, dict) and "rope_theta" in holder:
            return float(holder["rope_theta"])
    return float(getattr(hf_config, "rope_theta", 500000.0))


def config_from_hf(hf_config):
    """Map a transformers LlamaConfig to the native LlamaConfig."""
    return LlamaConfig(
        hidden_size=hf_config.hidden_size,
        intermediate_size=hf_config.intermediate_size,
        num_hidden_layers=hf_config.num_hidden_layers,
        num_attention_heads=hf_config.num_attention_heads,
        num_key_value_heads=getattr(hf_config, "num_key_value_heads",
                                    hf_config.num_attention_heads),
        vocab_size=hf_config.vocab_size,
        max_position_embeddings=hf_config.max_position_embeddings,
        rms_norm_eps=getattr(hf_config, "rms_norm_eps", 1e-5),
        rope_theta=_hf_rope_theta(hf_config),
        # Qwen2 configs set attention bias (qkv) on the same tree
        attention_bias=getattr(hf_config, "attention_bias",
                               getattr(hf_config, "model_type", "")
                               .startswith("qwen2")),
        tie_word_embeddings=getattr(hf_config, "tie_word_embeddings",
                                    False),
    )


def load_hf_llama(hf_model_or_state_dict, config=None):
    """Build a native LlamaForCausalLM from an HF model/state_dict."""
    if hasattr(hf_model_or_state_dict, "state_dict"):
        hf = hf_model_or_state_dict
        sd = hf.state_dict()
        if config is None:
            config = config_from_hf(hf.config)
        # HF may tie lm_head to the embedding (absent from state_dict)
        if "lm_head.weight" not in sd and \
                "model.embed_tokens.weight" in sd:
            sd = dict(sd)
            sd["lm_head.weight"] = sd["model.embed_tokens.weight"]
    else:
        sd = dict(hf_model_or_state_dict)
        assert config is not None, \
            "pass config= when loading from a raw state_dict"
    model = LlamaForCausalLM(config)
    wanted = dict(model.named_parameters())
    filtered = {k: v for k, v in sd.items() if k in wanted}
    missing = [k for k in wanted if k not in filtered]
    if missing:
        logger.warning(f"HF import: {len(missing)} params missing "
                       f"(e.g. {missing[:3]})")
    with torch.no_grad():
        model.load_state_dict(filtered, strict=False)
    return model
