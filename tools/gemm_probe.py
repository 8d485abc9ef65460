#!/usr/bin/env python3
"""Measure hipBLASLt bf16 GEMM vs fp8 _scaled_mm on the Llama-3-8B bench
shapes (mb4 x seq4096 tokens). Decides whether an fp8 MLP path can beat
bf16 on this stack: CDNA4's NON-scaled fp8 MFMA runs at the bf16 rate
(~2.4 PF); only hipBLASLt kernels using the MX block-scaled path
(mfma_scale_*_f8f6f4) can exceed it.

Run on GPU: python tools/gemm_probe.py
"""
import sys

import torch

M = 4 * 4096  # tokens per micro-batch at the bench config

# (name, M, K, N) — forward shapes; backward dgrad/wgrad share sizes
SHAPES = [
    ("qkv", M, 4096, 6144),
    ("attn_out", M, 4096, 4096),
    ("gate_up", M, 4096, 28672),
    ("mlp_down", M, 14336, 4096),
    ("lm_head", M, 4096, 128256),
]


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters  # ms


def main():
    torch.manual_seed(0)
    dev = "cuda"
    print(f"torch {torch.__version__}; fp8 dtypes: "
          f"{hasattr(torch, 'float8_e4m3fn')}")
    for name, m, k, n in SHAPES:
        a = torch.randn(m, k, device=dev, dtype=torch.bfloat16)
        b = torch.randn(n, k, device=dev, dtype=torch.bfloat16)
        flops = 2.0 * m * k * n

        t_bf16 = bench(lambda: a @ b.t())
        tf_bf16 = flops / t_bf16 / 1e9

        # tensorwise fp8
        a8 = a.to(torch.float8_e4m3fn)
        b8 = b.to(torch.float8_e4m3fn)
        sa = torch.tensor(1.0, device=dev)
        sb = torch.tensor(1.0, device=dev)
        try:
            t_fp8 = bench(lambda: torch._scaled_mm(
                a8, b8.t(), scale_a=sa, scale_b=sb,
                out_dtype=torch.bfloat16))
            tf_fp8 = flops / t_fp8 / 1e9
        except Exception as ex:
            t_fp8, tf_fp8 = float("nan"), 0.0
            print(f"  fp8 tensorwise failed: {ex}", file=sys.stderr)

        # rowwise fp8
        try:
            sar = torch.ones(m, 1, device=dev)
            sbr = torch.ones(1, n, device=dev)
            t_fp8r = bench(lambda: torch._scaled_mm(
                a8, b8.t(), scale_a=sar, scale_b=sbr,
                out_dtype=torch.bfloat16))
            tf_fp8r = flops / t_fp8r / 1e9
        except Exception as ex:
            t_fp8r, tf_fp8r = float("nan"), 0.0
            print(f"  fp8 rowwise failed: {ex}", file=sys.stderr)

        # MX blockwise (torch >= 2.8 exposes e8m0 scales; may be
        # unsupported on this stack — probe and report)
        tf_mx = 0.0
        try:
            if hasattr(torch, "float8_e8m0fnu"):
                sax = torch.ones(m, k // 32, device=dev) \
                    .to(torch.float8_e8m0fnu)
                sbx = torch.ones(n, k // 32, device=dev) \
                    .to(torch.float8_e8m0fnu)
                t_mx = bench(lambda: torch._scaled_mm(
                    a8, b8.t(), scale_a=sax, scale_b=sbx,
                    out_dtype=torch.bfloat16))
                tf_mx = flops / t_mx / 1e9
        except Exception as ex:
            print(f"  fp8 MX-block failed: {type(ex).__name__}: {ex}",
                  file=sys.stderr)

        print(f"{name:9s} M{m} K{k} N{n}: bf16 {tf_bf16:7.0f} TF | "
              f"fp8 tw {tf_fp8:7.0f} TF | fp8 rw {tf_fp8r:7.0f} TF | "
              f"fp8 mx {tf_mx:7.0f} TF")


if __name__ == "__main__":
    main()
