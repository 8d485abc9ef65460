"""HF kernel injection: module swaps + logits parity (CPU: ops fall back
to their torch reference paths; GPU runs the HIP kernels)."""
import pytest
import torch

transformers = pytest.importorskip("transformers")


def _tiny_hf_llama():
    from transformers import LlamaConfig, LlamaForCausalLM
    cfg = LlamaConfig(hidden_size=128, intermediate_size=256,
                      num_hidden_layers=2, num_attention_heads=4,
                      num_key_value_heads=2, vocab_size=512,
                      max_position_embeddings=128)
    torch.manual_seed(0)
    return LlamaForCausalLM(cfg)


def _tiny_hf_gpt2():
    from transformers import GPT2Config, GPT2LMHeadModel
    cfg = GPT2Config(n_embd=96, n_layer=2, n_head=4, vocab_size=512,
                     n_positions=128)
    torch.manual_seed(0)
    return GPT2LMHeadModel(cfg)


def test_inject_llama_swaps_and_matches():
    from deepspeed_amd.module_inject.replace_module import (
        LlamaPolicy, policy_for, replace_transformer_layer)
    model = _tiny_hf_llama().eval()
    assert policy_for(model) is LlamaPolicy
    ids = torch.randint(0, 512, (2, 16))
    with torch.no_grad():
        ref = model(ids).logits
    counts = replace_transformer_layer(model)
    assert counts["rmsnorm"] == 2 * 2 + 1   # 2/layer + final
    assert counts["mlp"] == 2
    assert counts["attention"] == 1
    with torch.no_grad():
        got = model(ids).logits
    err = (got - ref).abs().max().item()
    assert err < 2e-4, f"logits diverged after injection: {err}"


def test_inject_gpt2_layernorms():
    from deepspeed_amd.module_inject.replace_module import (
        GPT2Policy, policy_for, replace_transformer_layer)
    model = _tiny_hf_gpt2().eval()
    assert policy_for(model) is GPT2Policy
    ids = torch.randint(0, 512, (2, 12))
    with torch.no_grad():
        ref = model(ids).logits
    counts = replace_transformer_layer(model)
    assert counts["layernorm"] == 2 * 2 + 1
    with torch.no_grad():
        got = model(ids).logits
    err = (got - ref).abs().max().item()
    assert err < 2e-4, f"logits diverged after injection: {err}"


def test_init_inference_injects_hf():
    import deepspeed_amd
    model = _tiny_hf_llama().eval()
    eng = deepspeed_amd.init_inference(model, config={"dtype": "fp32"})
    from deepspeed_amd.module_inject.replace_module import _FusedRMSNorm
    n_fused = sum(isinstance(m, _FusedRMSNorm)
                  for m in eng.module.modules())
    assert n_fused == 5
    ids = torch.randint(0, 512, (1, 8))
    out = eng.generate(ids, max_new_tokens=4)
    assert out.shape[1] >= 12 or out.shape[1] == 12


def test_inject_bloom_layernorms_no_attn():
    """Bloom: alibi attention left alone, layernorms fused."""
    from transformers import BloomConfig, BloomForCausalLM
    from deepspeed_amd.module_inject.replace_module import (
        BloomPolicy, policy_for, replace_transformer_layer)
    cfg = BloomConfig(hidden_size=64, n_layer=2, n_head=4, vocab_size=512)
    torch.manual_seed(0)
    model = BloomForCausalLM(cfg).eval()
    assert policy_for(model) is BloomPolicy
    ids = torch.randint(0, 512, (2, 10))
    with torch.no_grad():
        ref = model(ids).logits
    counts = replace_transformer_layer(model)
    assert counts["attention"] == 0 and counts["layernorm"] >= 2 * 2
    with torch.no_grad():
        got = model(ids).logits
    assert (got - ref).abs().max().item() < 2e-4


def test_inject_gpt_neox():
    from transformers import GPTNeoXConfig, GPTNeoXForCausalLM
    from deepspeed_amd.module_inject.replace_module import (
        GPTNeoXPolicy, policy_for, replace_transformer_layer)
    cfg = GPTNeoXConfig(hidden_size=64, num_hidden_layers=2,
                        num_attention_heads=4, intermediate_size=128,
                        vocab_size=512, max_position_embeddings=64)
    torch.manual_seed(0)
    model = GPTNeoXForCausalLM(cfg).eval()
    assert policy_for(model) is GPTNeoXPolicy
    ids = torch.randint(0, 512, (2, 10))
    with torch.no_grad():
        ref = model(ids).logits
    counts = replace_transformer_layer(model)
    assert counts["layernorm"] >= 4
    with torch.no_grad():
        got = model(ids).logits
    assert (got - ref).abs().max().item() < 2e-3
