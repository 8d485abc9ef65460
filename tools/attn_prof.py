#!/usr/bin/env python3
"""Run the flash fwd kernel alone N times (for rocprofv3 pmc capture)."""
import math
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

B, S, Hq, Hk, D = 1, 4096, 32, 8, 128


def main():
    from deepspeed_amd.ops.loader import get_ext
    ext = get_ext()
    torch.manual_seed(0)
    q = torch.randn(B, S, Hq, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, S, Hk, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, S, Hk, D, device="cuda", dtype=torch.bfloat16)
    for _ in range(int(sys.argv[1]) if len(sys.argv) > 1 else 10):
        ext.flash_attn_fwd(q, k, v, True, 1 / math.sqrt(D))
    torch.cuda.synchronize()
    print("done")


if __name__ == "__main__":
    main()
