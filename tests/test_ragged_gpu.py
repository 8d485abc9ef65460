"""Ragged decode HIP kernel vs fp32 reference (mixed sequence lengths)."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu


def _ref_decode(q, kpool, vpool, rows, lens):
    """fp32 per-row attention over each row's valid KV prefix."""
    n, _, Hq, D = q.shape
    Hk = kpool.shape[2]
    G = Hq // Hk
    outs = []
    for i in range(n):
        L = int(lens[i])
        s = int(rows[i])
        k = kpool[s, :L].float()                  # [L, Hk, D]
        v = vpool[s, :L].float()
        qi = q[i, 0].float()                      # [Hq, D]
        k = k.repeat_interleave(G, dim=1)
        v = v.repeat_interleave(G, dim=1)
        att = torch.einsum("hd,lhd->hl", qi, k) / math.sqrt(D)
        p = torch.softmax(att, dim=-1)
        outs.append(torch.einsum("hl,lhd->hd", p, v))
    return torch.stack(outs).unsqueeze(1)         # [n,1,Hq,D]


@pytest.mark.parametrize("D,G", [(128, 4), (128, 1), (64, 2), (128, 8)])
def test_ragged_decode_matches_reference(D, G):
    from deepspeed_amd.ops.loader import get_ext
    torch.manual_seed(0)
    Hk = 4
    Hq = Hk * G
    n, B, Smax = 5, 8, 700
    kpool = torch.randn(B, Smax, Hk, D, device="cuda", dtype=torch.bfloat16)
    vpool = torch.randn(B, Smax, Hk, D, device="cuda", dtype=torch.bfloat16)
    q = torch.randn(n, 1, Hq, D, device="cuda", dtype=torch.bfloat16)
    rows = torch.tensor([7, 2, 4, 0, 5], device="cuda")
    lens = torch.tensor([700, 1, 33, 512, 130], device="cuda")
    out = get_ext().ragged_decode(q, kpool, vpool, rows, lens)
    ref = _ref_decode(q, kpool, vpool, rows, lens)
    err = (out.float() - ref).abs().max().item()
    assert err < 3e-2, f"D={D} G={G}: max err {err}"


def test_ragged_decode_long_context_split():
    """Lengths spanning many split-KV chunks."""
    from deepspeed_amd.ops.loader import get_ext
    torch.manual_seed(1)
    D, Hk, G = 128, 8, 4
    Hq = Hk * G
    n, B, Smax = 3, 4, 8192
    kpool = torch.randn(B, Smax, Hk, D, device="cuda", dtype=torch.bfloat16)
    vpool = torch.randn(B, Smax, Hk, D, device="cuda", dtype=torch.bfloat16)
    q = torch.randn(n, 1, Hq, D, device="cuda", dtype=torch.bfloat16)
    rows = torch.tensor([0, 2, 3], device="cuda")
    lens = torch.tensor([8192, 4097, 640], device="cuda")
    out = get_ext().ragged_decode(q, kpool, vpool, rows, lens)
    ref = _ref_decode(q, kpool, vpool, rows, lens)
    err = (out.float() - ref).abs().max().item()
    assert err < 3e-2, f"max err {err}"


def test_serving_engine_uses_ragged_kernel():
    """Continuous-batching decode routes through the pool-direct kernel
    and generates the same tokens as the masked-SDPA fallback."""
    from deepspeed_amd.inference.serving import ContinuousBatchingEngine
    from deepspeed_amd.models.llama import LLAMA_CONFIGS, LlamaForCausalLM
    torch.manual_seed(0)
    cfg = LLAMA_CONFIGS["llama-small"]  # head_dim 64 (tiny is 32)
    with torch.device("cuda"):
        model = LlamaForCausalLM(cfg).bfloat16().eval()

    def run(force_fallback):
        eng = ContinuousBatchingEngine(model, max_batch=4)
        if force_fallback:
            for c in eng.caches:
                c._use_ragged = False
        else:
            assert all(c._use_ragged for c in eng.caches)
        g = torch.Generator().manual_seed(3)
        for i in range(3):
            eng.add_request(torch.randint(0, cfg.vocab_size, (8 + 3 * i,),
                                          generator=g),
                            max_new_tokens=12)
        out = eng.run()
        return [out[k] for k in sorted(out)]

    fast = run(False)
    slow = run(True)
    flat_f = [t for rr in fast for t in (rr.tolist()
              if hasattr(rr, "tolist") else rr)]
    flat_s = [t for rr in slow for t in (rr.tolist()
              if hasattr(rr, "tolist") else rr)]
    same = sum(int(a == b) for a, b in zip(flat_f, flat_s))
    total = len(flat_f)
    assert same / total > 0.9, f"token agreement {same}/{total}"
