#!/usr/bin/env python3
"""Decode latency: eager KV-cache loop vs hipGraph-captured step."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402


def main():
    model_name = sys.argv[1] if len(sys.argv) > 1 else "llama3-8b"
    n_tok = int(sys.argv[2]) if len(sys.argv) > 2 else 64
    import deepspeed_amd
    from deepspeed_amd.models.llama import LLAMA_CONFIGS, LlamaForCausalLM
    cfg = LLAMA_CONFIGS[model_name]
    torch.manual_seed(0)
    with torch.device("cuda:0"):
        model = LlamaForCausalLM(cfg)
    engine = deepspeed_amd.init_inference(model, config={})
    ids = torch.randint(0, cfg.vocab_size, (1, 64), device="cuda:0")

    # warmup both paths
    engine.generate(ids, max_new_tokens=4)
    engine.generate_hipgraph(ids, max_new_tokens=4)
    torch.cuda.synchronize()

    t0 = time.time()
    engine.generate(ids, max_new_tokens=n_tok)
    torch.cuda.synchronize()
    t_eager = time.time() - t0

    t0 = time.time()
    engine.generate_hipgraph(ids, max_new_tokens=n_tok)
    torch.cuda.synchronize()
    t_graph = time.time() - t0

    print(f"model={model_name} tokens={n_tok}")
    print(f"eager decode:    {t_eager/n_tok*1000:7.2f} ms/token "
          f"({n_tok/t_eager:6.1f} tok/s)")
    print(f"hipGraph decode: {t_graph/n_tok*1000:7.2f} ms/token "
          f"({n_tok/t_graph:6.1f} tok/s)  "
          f"speedup {t_eager/t_graph:.2f}x")


if __name__ == "__main__":
    main()
