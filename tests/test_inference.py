"""InferenceEngine: cached decode == uncached full forward; generation."""
import torch

import deepspeed_amd
from deepspeed_amd.models.llama import LLAMA_CONFIGS, LlamaForCausalLM


def _tiny_model():
    cfg = LLAMA_CONFIGS["llama-tiny"]
    torch.manual_seed(0)
    m = LlamaForCausalLM(cfg).float()
    m.eval()
    return m, cfg


def test_kv_cache_matches_full_forward():
    model, cfg = _tiny_model()
    engine = deepspeed_amd.init_inference(model,
                                          config={"dtype": torch.float32})
    torch.manual_seed(1)
    ids = torch.randint(0, cfg.vocab_size, (2, 16)).to(engine.device)
    # full forward logits
    full = engine.forward(ids)
    # prefill 12 then decode 4 with cache
    engine._alloc_caches(2, 32)
    logits = engine.module(ids[:, :12], kv_caches=engine._caches)
    outs = [logits]
    for t in range(12, 16):
        step_logits = engine.module(ids[:, t:t + 1], seq_offset=t,
                                    kv_caches=engine._caches)
        outs.append(step_logits)
    cached = torch.cat(outs, dim=1)
    assert torch.allclose(cached, full, atol=1e-4), \
        (cached - full).abs().max()


def test_generate_greedy_deterministic():
    model, cfg = _tiny_model()
    engine = deepspeed_amd.init_inference(model,
                                          config={"dtype": torch.float32})
    torch.manual_seed(2)
    ids = torch.randint(0, cfg.vocab_size, (1, 8))
    out1 = engine.generate(ids, max_new_tokens=8)
    out2 = engine.generate(ids, max_new_tokens=8)
    assert out1.shape[1] == 16
    assert torch.equal(out1, out2)
    # greedy continuation must be self-consistent with teacher forcing
    logits = engine.forward(out1)
    for t in range(8, 15):
        assert logits[0, t].argmax().item() == out1[0, t + 1].item()


def test_generate_sampling_runs():
    model, cfg = _tiny_model()
    engine = deepspeed_amd.init_inference(model,
                                          config={"dtype": torch.float32})
    ids = torch.randint(0, cfg.vocab_size, (2, 4))
    out = engine.generate(ids, max_new_tokens=4, temperature=0.8, top_k=10)
    assert out.shape == (2, 8)


def _init_inference_tp2():
    import torch.distributed as dist
    import deepspeed_amd as ds
    from deepspeed_amd.models.llama import LLAMA_CONFIGS, LlamaForCausalLM
    torch.manual_seed(0)
    model = LlamaForCausalLM(LLAMA_CONFIGS["llama-tiny"]).eval()
    torch.manual_seed(5)
    ids = torch.randint(0, 2000, (1, 12))
    with torch.no_grad():
        ref = model(ids).float()
    eng = ds.init_inference(model, dtype=torch.float32,
                            tensor_parallel={"tp_size": 2})
    with torch.no_grad():
        out = eng(ids).float()
    err = (out - ref).abs().max().item()
    assert err < 1e-4, err


def test_init_inference_tensor_parallel_world2():
    from tests.common import run_distributed
    run_distributed(_init_inference_tp2, world_size=2)


def test_hf_llama_import_logits_match():
    """Native model loaded from an HF checkpoint reproduces HF logits."""
    from transformers import LlamaConfig as HFConfig
    from transformers import LlamaForCausalLM as HFModel
    from deepspeed_amd.models.hf import load_hf_llama
    torch.manual_seed(0)
    hf_cfg = HFConfig(hidden_size=128, intermediate_size=256,
                      num_hidden_layers=2, num_attention_heads=8,
                      num_key_value_heads=4, vocab_size=512,
                      max_position_embeddings=128, rms_norm_eps=1e-5,
                      rope_theta=10000.0)
    hf = HFModel(hf_cfg).eval()
    native = load_hf_llama(hf).eval()
    ids = torch.randint(0, 512, (2, 16))
    with torch.no_grad():
        ref = hf(ids).logits
        out = native(ids)
    err = (out - ref).abs().max().item()
    assert err < 2e-4, err


def test_init_inference_hf_model_delegates_generate():
    from transformers import LlamaConfig as HFConfig
    from transformers import LlamaForCausalLM as HFModel
    import deepspeed_amd as ds
    torch.manual_seed(0)
    hf = HFModel(HFConfig(hidden_size=64, intermediate_size=128,
                          num_hidden_layers=2, num_attention_heads=4,
                          num_key_value_heads=2, vocab_size=128,
                          max_position_embeddings=64)).eval()
    eng = ds.init_inference(hf, dtype=torch.float32)
    ids = torch.randint(0, 128, (1, 8))
    out = eng.generate(ids, max_new_tokens=4)
    assert out.shape[1] == 12


def test_init_inference_int8_weight_residency():
    """dtype int8 (or quant.enabled) swaps linears for group-wise int8
    dequant-on-the-fly modules; generation still works (ref
    init_inference quantization_setting)."""
    import torch
    from deepspeed_amd import init_inference
    from deepspeed_amd.linear.quantization import QuantizedLinear
    from deepspeed_amd.models.llama import LLAMA_CONFIGS, LlamaForCausalLM
    torch.manual_seed(0)
    model = LlamaForCausalLM(LLAMA_CONFIGS["llama-tiny"])
    eng = init_inference(model, config={"dtype": "int8"})
    nq = sum(isinstance(m, QuantizedLinear) for m in eng.module.modules())
    assert nq > 0, "no linears quantized"
    assert eng._config.dtype == torch.bfloat16  # compute stays bf16
    ids = torch.randint(0, 512, (1, 8))
    out = eng.generate(ids, max_new_tokens=4)
    assert out.shape == (1, 12)


def test_hf_qwen2_import_logits_match():
    """Qwen2 HF checkpoint (qkv bias, tied embeddings) imports onto the
    native modules with logits parity."""
    import pytest
    import torch
    transformers = pytest.importorskip("transformers")
    from transformers import Qwen2Config, Qwen2ForCausalLM
    from deepspeed_amd.models.hf import load_hf_llama
    hf_cfg = Qwen2Config(hidden_size=128, intermediate_size=256,
                         num_hidden_layers=2, num_attention_heads=4,
                         num_key_value_heads=2, vocab_size=512,
                         max_position_embeddings=128,
                         tie_word_embeddings=True)
    torch.manual_seed(0)
    hf = Qwen2ForCausalLM(hf_cfg).eval()
    native = load_hf_llama(hf).eval()
    assert native.model.layers[0].self_attn.q_proj.bias is not None
    ids = torch.randint(0, 512, (2, 16))
    with torch.no_grad():
        ref = hf(ids).logits.float()
        got = native(ids).float()
    err = (got - ref).abs().max().item()
    assert err < 2e-3, f"qwen2 logits diverged: {err}"


def test_hf_mistral_import_logits_match():
    """Mistral shares the llama tree; import + parity (sliding window
    is a no-op below the window size)."""
    import pytest
    import torch
    transformers = pytest.importorskip("transformers")
    from transformers import MistralConfig, MistralForCausalLM
    from deepspeed_amd.models.hf import load_hf_llama
    hf_cfg = MistralConfig(hidden_size=128, intermediate_size=256,
                           num_hidden_layers=2, num_attention_heads=4,
                           num_key_value_heads=2, vocab_size=512,
                           max_position_embeddings=128,
                           sliding_window=4096)
    torch.manual_seed(0)
    hf = MistralForCausalLM(hf_cfg).eval()
    native = load_hf_llama(hf).eval()
    ids = torch.randint(0, 512, (2, 16))
    with torch.no_grad():
        ref = hf(ids).logits.float()
        got = native(ids).float()
    err = (got - ref).abs().max().item()
    assert err < 2e-3, f"mistral logits diverged: {err}"
