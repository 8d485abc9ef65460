#!/usr/bin/env python3
"""Long-context training of a HuggingFace model with Ulysses SP (ALST).

Launch (8 GPUs, sequence-parallel degree 8):
  torchrun --nproc-per-node 8 examples/train_hf_ulysses.py
"""
import torch

import deepspeed_amd as ds
from deepspeed_amd.runtime.ulysses_sp_hf import (apply_ulysses_sp_to_hf,
                                                 shard_batch_for_sp)


def main():
    from transformers import LlamaConfig, LlamaForCausalLM
    ds.comm.init_distributed()
    world = ds.comm.get_world_size()
    cfg = LlamaConfig(hidden_size=1024, intermediate_size=2816,
                      num_hidden_layers=8, num_attention_heads=16,
                      num_key_value_heads=8, vocab_size=32000,
                      max_position_embeddings=65536)
    model = LlamaForCausalLM(cfg)
    apply_ulysses_sp_to_hf(model, sp_size=world)

    engine, _, _, _ = ds.initialize(model=model, config={
        "train_micro_batch_size_per_gpu": 1,
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-4}},
        "bf16": {"enabled": True},
        "zero_optimization": {"stage": 1},
    })
    S = 32768  # full sequence; each rank holds S/world tokens
    for step in range(3):
        batch = {"input_ids": torch.randint(0, 32000, (1, S)),
                 "labels": torch.randint(0, 32000, (1, S))}
        local = shard_batch_for_sp(batch)
        out = engine(local["input_ids"].to(engine.device),
                     position_ids=local["position_ids"].to(engine.device))
        logits = out.logits
        labels = local["shift_labels"].to(engine.device)
        loss = torch.nn.functional.cross_entropy(
            logits.float().flatten(0, 1), labels.flatten(),
            ignore_index=-100)
        engine.backward(loss)
        engine.step()
        if ds.comm.get_rank() == 0:
            print(f"step {step}: loss {loss.item():.4f}")


if __name__ == "__main__":
    main()
