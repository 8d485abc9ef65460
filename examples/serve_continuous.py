#!/usr/bin/env python3
"""Continuous-batching serving: requests stream in, decode stays batched.

Run on 1 GPU:  python examples/serve_continuous.py
"""
import torch

from deepspeed_amd.inference.serving import ContinuousBatchingEngine
from deepspeed_amd.models.llama import LLAMA_CONFIGS, LlamaForCausalLM


def main():
    torch.manual_seed(0)
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    dtype = torch.bfloat16 if dev == "cuda" else torch.float32
    model = LlamaForCausalLM(LLAMA_CONFIGS["llama-small"]) \
        .to(dev, dtype).eval()
    eng = ContinuousBatchingEngine(model, max_batch=8)

    # requests arrive over time; slots free up and are reused
    for i in range(12):
        prompt = torch.randint(0, 31000, (16 + 7 * (i % 3),))
        eng.add_request(prompt, max_new_tokens=32,
                        temperature=0.7 if i % 2 else 0.0, top_k=40)
    done = eng.run()
    for rid, toks in sorted(done.items()):
        print(f"request {rid}: {len(toks)} tokens")


if __name__ == "__main__":
    main()
