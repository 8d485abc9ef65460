#!/usr/bin/env python3
"""Memory-bound kernel roofline check: achieved GB/s vs ~6.3 TB/s ceiling."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402


def bench(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / iters


def main():
    from deepspeed_amd.ops.loader import get_ext
    from deepspeed_amd.ops import functional as F
    ext = get_ext()
    dev = "cuda"
    R, H = 8192, 8192
    x = torch.randn(R, H, device=dev, dtype=torch.bfloat16)
    w = torch.randn(H, device=dev, dtype=torch.bfloat16)
    dy = torch.randn_like(x)

    rows = []

    t = bench(lambda: ext.rmsnorm_fwd(x, w, 1e-5))
    traffic = R * H * 2 * 2  # read x, write y
    rows.append(("rmsnorm_fwd", t, traffic))

    y, rstd = ext.rmsnorm_fwd(x, w, 1e-5)
    t = bench(lambda: ext.rmsnorm_bwd(dy, x, w, rstd))
    traffic = R * H * 2 * 5  # dy,x (x2 passes), dx
    rows.append(("rmsnorm_bwd(dx+dw)", t, traffic))

    g = torch.randn(R, H, device=dev, dtype=torch.bfloat16)
    u = torch.randn_like(g)
    t = bench(lambda: ext.swiglu_fwd(g, u))
    rows.append(("swiglu_fwd", t, R * H * 2 * 3))
    t = bench(lambda: ext.swiglu_bwd(dy, g, u))
    rows.append(("swiglu_bwd", t, R * H * 2 * 5))

    n = 1 << 28
    p = torch.randn(n, device=dev)
    gr = torch.randn(n, device=dev)
    m = torch.zeros(n, device=dev)
    v = torch.zeros(n, device=dev)
    o16 = torch.empty(n, device=dev, dtype=torch.bfloat16)
    t = bench(lambda: ext.multi_tensor_adam([p], [gr], [m], [v], 1e-3, 0.9,
                                            0.999, 1e-8, 2, 1, 1, 0.0,
                                            [o16], 1.0))
    rows.append(("fused_adam(+bf16 out)", t, n * (16 + 12 + 2)))

    t = bench(lambda: ext.l2norm_sq([p]))
    rows.append(("l2norm_sq", t, n * 4))

    t = bench(lambda: ext.accum_bf16_to_f32(m, o16, 1.0))
    rows.append(("accum_bf16_to_f32", t, n * (2 + 4 + 4)))

    NV, V = 8192, 128256
    logits = torch.randn(NV, V, device=dev, dtype=torch.bfloat16)
    tgt = torch.randint(0, V, (NV,), device=dev)
    t = bench(lambda: ext.cross_entropy_fwd(logits, tgt, -100), iters=5)
    rows.append(("cross_entropy_fwd 128k-vocab", t, NV * V * 2))

    cos, sin = F.build_rope_cache(4096, 128, device=dev)
    q = torch.randn(4, 4096, 32, 128, device=dev, dtype=torch.bfloat16)
    qo = torch.empty_like(q)
    t = bench(lambda: ext.rope(qo, q, cos, sin, 0, False))
    rows.append(("rope", t, q.numel() * 2 * 2))

    print(f"{'kernel':<30} {'ms':>8} {'GB/s':>9}  (ceiling ~6300 GB/s)")
    for name, t, traffic in rows:
        print(f"{name:<30} {t*1000:>8.3f} {traffic/t/1e9:>9.0f}")


if __name__ == "__main__":
    main()
