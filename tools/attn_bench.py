#!/usr/bin/env python3
"""A/B attention microbenchmark: HIP flash fwd + chunked bwd vs torch SDPA."""
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import os as _os
B = int(_os.environ.get("AB_B", 1))
S = int(_os.environ.get("AB_S", 4096))
Hq, Hk, D = 32, 8, 128
it = 10


def bench(fn, *args):
    for _ in range(3):
        fn(*args)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(it):
        fn(*args)
    torch.cuda.synchronize()
    return (time.time() - t0) / it * 1000


def main():
    torch.manual_seed(0)
    q = torch.randn(B, S, Hq, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn(B, S, Hk, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    v = torch.randn(B, S, Hk, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    g = torch.randn(B, S, Hq, D, device="cuda", dtype=torch.bfloat16)

    from deepspeed_amd.ops.attention import _FlashAttnFn as _FF
    def flash_attention(q,k,v,causal=True):
        import math as _m
        return _FF.apply(q,k,v,causal,1.0/_m.sqrt(q.shape[-1]))
    from deepspeed_amd.ops.loader import get_ext
    ext = get_ext()

    # fwd only
    t_hip_f = bench(lambda: ext.flash_attn_fwd(q, k, v, True,
                                               1 / math.sqrt(D)))
    import torch.nn.functional as F
    qt, kt, vt = (t.transpose(1, 2) for t in (q, k, v))
    t_sdpa_f = bench(lambda: F.scaled_dot_product_attention(
        qt, kt, vt, is_causal=True, enable_gqa=True))

    def hip_fb():
        out = flash_attention(q, k, v, True)
        out.backward(g)
        q.grad = k.grad = v.grad = None

    def sdpa_fb():
        out = F.scaled_dot_product_attention(qt, kt, vt, is_causal=True,
                                             enable_gqa=True)
        out.backward(g.transpose(1, 2))
        q.grad = k.grad = v.grad = None

    t_hip_fb = bench(hip_fb)
    t_sdpa_fb = bench(sdpa_fb)

    flops_f = 4 * B * S * S * D * Hq / 2  # causal
    print(f"fwd:      HIP {t_hip_f:8.2f} ms ({flops_f/t_hip_f/1e9:7.1f} TF)  "
          f"SDPA {t_sdpa_f:8.2f} ms ({flops_f/t_sdpa_f/1e9:7.1f} TF)")
    print(f"fwd+bwd:  HIP {t_hip_fb:8.2f} ms  SDPA {t_sdpa_fb:8.2f} ms")


if __name__ == "__main__":
    main()
