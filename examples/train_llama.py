#!/usr/bin/env python3
"""Minimal Llama training loop on deepspeed_amd.

Single node:  torchrun --nproc-per-node 8 examples/train_llama.py
          or: bin/deepspeed examples/train_llama.py
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import deepspeed_amd
from deepspeed_amd.models.llama import LLAMA_CONFIGS, LlamaForCausalLM


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama-small")
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--seq-len", type=int, default=1024)
    p.add_argument("--deepspeed_config", default=os.path.join(
        os.path.dirname(__file__), "ds_config_zero3.json"))
    p.add_argument("--local_rank", type=int, default=-1)
    args = p.parse_args()

    cfg = LLAMA_CONFIGS[args.model]
    torch.manual_seed(42)
    model = LlamaForCausalLM(cfg)
    engine, _, _, sched = deepspeed_amd.initialize(
        model=model, config=args.deepspeed_config)

    mb = engine.train_micro_batch_size_per_gpu()
    for step in range(args.steps):
        # synthetic data; swap in your tokenized corpus via
        # deepspeed_amd.runtime.data_sampling.IndexedDataset
        data = torch.randint(0, cfg.vocab_size, (mb, args.seq_len),
                             device=engine.device)
        loss = engine(data, labels=data)
        engine.backward(loss)
        engine.step()

    engine.save_checkpoint("./checkpoints")
    if engine.global_rank == 0:
        print(f"done; final loss {loss.item():.4f}")


if __name__ == "__main__":
    main()
