#!/usr/bin/env python3
"""Continuous-batching decode throughput vs sequential generate."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from deepspeed_amd.inference.engine import InferenceEngine  # noqa: E402
from deepspeed_amd.inference.serving import \
    ContinuousBatchingEngine  # noqa: E402
from deepspeed_amd.models.llama import (LLAMA_CONFIGS,  # noqa: E402
                                        LlamaForCausalLM)

MODEL = os.environ.get("SB_MODEL", "llama-small")
NREQ = int(os.environ.get("SB_NREQ", 16))
NEW = int(os.environ.get("SB_NEW", 64))
BATCH = int(os.environ.get("SB_BATCH", 8))


def main():
    torch.manual_seed(0)
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    dtype = torch.bfloat16 if dev == "cuda" else torch.float32
    model = LlamaForCausalLM(LLAMA_CONFIGS[MODEL]).to(dev, dtype).eval()
    vocab = model.cfg.vocab_size
    prompts = [torch.randint(0, vocab, (32 + (i % 5) * 16,))
               for i in range(NREQ)]

    eng = InferenceEngine(model)
    # warm
    eng.generate(prompts[0].view(1, -1).to(dev), max_new_tokens=4)
    if dev == "cuda":
        torch.cuda.synchronize()
    t0 = time.time()
    for p in prompts:
        eng.generate(p.view(1, -1).to(dev), max_new_tokens=NEW)
    if dev == "cuda":
        torch.cuda.synchronize()
    seq_s = time.time() - t0

    cb = ContinuousBatchingEngine(model, max_batch=BATCH)
    t0 = time.time()
    for p in prompts:
        cb.add_request(p, max_new_tokens=NEW)
    cb.run()
    if dev == "cuda":
        torch.cuda.synchronize()
    cb_s = time.time() - t0

    tok = NREQ * NEW
    print(f"{MODEL}: {NREQ} reqs x {NEW} new tokens")
    print(f"sequential: {seq_s:6.2f} s  {tok / seq_s:8.1f} tok/s")
    print(f"continuous (batch {BATCH}): {cb_s:6.2f} s  "
          f"{tok / cb_s:8.1f} tok/s  ({seq_s / cb_s:.2f}x)")


if __name__ == "__main__":
    main()
