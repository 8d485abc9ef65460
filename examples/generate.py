#!/usr/bin/env python3
"""KV-cache generation, optionally hipGraph-captured decode."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import deepspeed_amd
from deepspeed_amd.models.llama import LLAMA_CONFIGS, LlamaForCausalLM

cfg = LLAMA_CONFIGS["llama-small"]
torch.manual_seed(0)
with torch.device("cuda:0" if torch.cuda.is_available() else "cpu"):
    model = LlamaForCausalLM(cfg)
engine = deepspeed_amd.init_inference(model, config={})
prompt = torch.randint(0, cfg.vocab_size, (1, 32), device=engine.device)
if torch.cuda.is_available():
    out = engine.generate_hipgraph(prompt, max_new_tokens=64)
else:
    out = engine.generate(prompt, max_new_tokens=64)
print("generated", out.shape[1] - prompt.shape[1], "tokens")
