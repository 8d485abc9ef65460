"""DS4Sci Evoformer attention + sparse-attention layouts."""
import math

import pytest
import torch


def test_evoformer_attention_matches_manual():
    from deepspeed_amd.ops.deepspeed4science import DS4Sci_EvoformerAttention
    torch.manual_seed(0)
    B, N, S, H, D = 2, 3, 16, 4, 8
    q = torch.randn(B, N, S, H, D)
    k = torch.randn(B, N, S, H, D)
    v = torch.randn(B, N, S, H, D)
    mask_bias = torch.where(torch.rand(B, N, 1, 1, S) > 0.1, 0.0,
                            float("-inf"))
    pair_bias = torch.randn(B, 1, H, S, S)
    out = DS4Sci_EvoformerAttention(q, k, v, [mask_bias, pair_bias])
    assert out.shape == (B, N, S, H, D)
    # manual reference
    qh = q.transpose(-2, -3)
    kh = k.transpose(-2, -3)
    vh = v.transpose(-2, -3)
    logits = qh @ kh.transpose(-1, -2) / math.sqrt(D)
    logits = logits + mask_bias + pair_bias
    ref = (torch.softmax(logits, -1) @ vh).transpose(-2, -3)
    assert torch.allclose(out, ref, atol=1e-5)


def test_evoformer_pair_bias_gets_grad():
    from deepspeed_amd.ops.deepspeed4science import DS4Sci_EvoformerAttention
    q = torch.randn(1, 2, 8, 2, 4, requires_grad=True)
    k = torch.randn(1, 2, 8, 2, 4)
    v = torch.randn(1, 2, 8, 2, 4)
    pair = torch.randn(1, 1, 2, 8, 8, requires_grad=True)
    out = DS4Sci_EvoformerAttention(q, k, v, [None, pair])
    out.sum().backward()
    assert q.grad is not None and pair.grad is not None
    assert pair.grad.abs().sum() > 0


@pytest.mark.parametrize("cfg_name", ["fixed", "bigbird", "bslongformer",
                                      "variable", "sliding"])
def test_sparsity_layouts(cfg_name):
    from deepspeed_amd.ops.sparse_attention import (
        BigBirdSparsityConfig, BSLongformerSparsityConfig,
        FixedSparsityConfig, LocalSlidingWindowSparsityConfig,
        VariableSparsityConfig)
    H, S, block = 4, 128, 16
    cfg = {
        "fixed": FixedSparsityConfig(H, block, attention="unidirectional"),
        "bigbird": BigBirdSparsityConfig(H, block),
        "bslongformer": BSLongformerSparsityConfig(H, block),
        "variable": VariableSparsityConfig(H, block,
                                           attention="unidirectional"),
        "sliding": LocalSlidingWindowSparsityConfig(H, block),
    }[cfg_name]
    layout = cfg.make_layout(S)
    n = S // block
    assert layout.shape == (H, n, n)
    assert layout.sum() > 0
    frac = layout.float().mean().item()
    assert frac < 1.0  # actually sparse
    # every query block attends to something
    assert (layout.sum(-1) > 0).all()


def test_sparse_self_attention_equals_dense_when_dense():
    from deepspeed_amd.ops.sparse_attention import (DenseSparsityConfig,
                                                    SparseSelfAttention)
    torch.manual_seed(0)
    B, H, S, D = 2, 4, 64, 16
    q = torch.randn(B, H, S, D)
    k = torch.randn(B, H, S, D)
    v = torch.randn(B, H, S, D)
    attn = SparseSelfAttention(DenseSparsityConfig(H, 16))
    out = attn(q, k, v)
    ref = torch.nn.functional.scaled_dot_product_attention(q, k, v)
    assert torch.allclose(out, ref, atol=1e-5)


def test_sparse_self_attention_masks_blocks():
    from deepspeed_amd.ops.sparse_attention import (
        LocalSlidingWindowSparsityConfig, SparseSelfAttention)
    torch.manual_seed(0)
    B, H, S, D = 1, 2, 64, 8
    cfg = LocalSlidingWindowSparsityConfig(H, 16, num_sliding_window_blocks=1,
                                           attention="unidirectional")
    attn = SparseSelfAttention(cfg)
    q = torch.randn(B, H, S, D)
    k = torch.randn(B, H, S, D)
    v = torch.randn(B, H, S, D)
    # layout is block-granular; within-block causality comes from attn_mask
    causal = torch.tril(torch.ones(S, S))
    out = attn(q, k, v, attn_mask=causal)
    assert out.shape == (B, H, S, D)
    # causal => position 0 sees only itself
    ref0 = v[:, :, 0]
    assert torch.allclose(out[:, :, 0], ref0, atol=1e-5)
    # far-away block (past the sliding window) is masked: position 63
    # must not depend on v[:, :, :16]
    v2 = v.clone()
    v2[:, :, :16] = 99.0
    out2 = attn(q, k, v2, attn_mask=causal)
    assert torch.allclose(out[:, :, 63], out2[:, :, 63], atol=1e-5)
