#!/usr/bin/env python3
"""2 RCCL ranks on ONE GPU (if RCCL permits): exercises the world>1
stage-3 stream paths (coalesced gathers, PreMulSum reduce-scatter,
side-stream overlap) without an 8-GPU box."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
import torch.multiprocessing as mp  # noqa: E402


def worker(rank):
    os.environ.update(RANK=str(rank), WORLD_SIZE="2", LOCAL_RANK="0",
                      MASTER_ADDR="127.0.0.1", MASTER_PORT="29551")
    import torch.distributed as td
    td.init_process_group("nccl", rank=rank, world_size=2)
    torch.cuda.set_device(0)
    import deepspeed_amd
    from deepspeed_amd.models.llama import LLAMA_CONFIGS, LlamaForCausalLM
    cfg = LLAMA_CONFIGS["llama-tiny"]
    torch.manual_seed(100 + rank)
    with torch.device("cuda:0"):
        model = LlamaForCausalLM(cfg)
    config = {
        "train_micro_batch_size_per_gpu": 2,
        "optimizer": {"type": "AdamW", "params": {"lr": 3e-4}},
        "zero_optimization": {"stage": 3},
        "bf16": {"enabled": True},
    }
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config=config)
    data = torch.randint(0, cfg.vocab_size, (2, 64), device="cuda:0")
    losses = []
    for _ in range(4):
        loss = engine(data, labels=data)
        engine.backward(loss)
        engine.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0], losses
    if rank == 0:
        print(f"WORLD2_OK losses={losses}", flush=True)
    td.destroy_process_group()


if __name__ == "__main__":
    mp.spawn(worker, nprocs=2)
