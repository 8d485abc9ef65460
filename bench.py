#!/usr/bin/env python3
"""Flagship benchmark: Llama-3-8B ZeRO-3 bf16 training throughput (tokens/s).

Contract (driver): `python bench.py --gpus N --steps K --warmup W`; launched
via torch.distributed.run for N>1 (one rank per GPU over RCCL). W untimed
warmup steps, then exactly K timed steps bracketed by barrier+synchronize;
MAX elapsed over ranks; rank 0 prints ONE JSON line.

Synthetic data (random tokens), random-init weights (no network access).
"""
import argparse
import json
import os
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=8)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--model", type=str, default="llama3-8b")
    p.add_argument("--seq-len", type=int, default=4096)
    p.add_argument("--micro-batch", type=int, default=4)
    p.add_argument("--grad-accum", type=int, default=1)
    p.add_argument("--zero-stage", type=int, default=3)
    p.add_argument("--act-ckpt", action="store_true",
                   help="enable activation checkpointing (default off: "
                   "288 GB HBM3E fits full activations at these configs)")
    p.add_argument("--local_rank", type=int, default=-1)
    return p.parse_args()


def main():
    args = parse_args()
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29515")

    import deepspeed_amd
    from deepspeed_amd import comm as dist
    from deepspeed_amd.models.llama import LLAMA_CONFIGS, LlamaForCausalLM

    dist.init_distributed()
    rank = dist.get_rank()
    world = dist.get_world_size()
    device = torch.device("cuda", dist.get_local_rank()) \
        if torch.cuda.is_available() else torch.device("cpu")
    if torch.cuda.is_available():
        torch.cuda.set_device(device)

    cfg = LLAMA_CONFIGS[args.model]
    cfg.activation_checkpointing = args.act_ckpt
    torch.manual_seed(1234 + rank)
    t0 = time.time()
    # build directly on device: 8B bf16 = 16 GB, fits trivially in 288 GB
    with torch.device(device):
        model = LlamaForCausalLM(cfg)
    if rank == 0:
        n_params = sum(p.numel() for p in model.parameters())
        print(f"# model {args.model}: {n_params/1e9:.2f}B params, "
              f"built in {time.time()-t0:.1f}s", flush=True)

    ds_config = {
        "train_micro_batch_size_per_gpu": args.micro_batch,
        "gradient_accumulation_steps": args.grad_accum,
        "optimizer": {"type": "AdamW",
                      "params": {"lr": 1e-4, "betas": [0.9, 0.95],
                                 "eps": 1e-8, "weight_decay": 0.1}},
        "zero_optimization": {"stage": args.zero_stage},
        "bf16": {"enabled": True},
        "gradient_clipping": 1.0,
        "steps_per_print": 1000000,
    }
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config=ds_config)

    S = args.seq_len
    B = args.micro_batch
    # rotating synthetic batches: fresh tokens each micro-step (no caching)
    n_bufs = 8
    bufs = [torch.randint(0, cfg.vocab_size, (B, S), device=device)
            for _ in range(n_bufs)]
    step_idx = [0]

    def one_step():
        for _ in range(args.grad_accum):
            data = bufs[step_idx[0] % n_bufs]
            step_idx[0] += 1
            loss = engine(data, labels=data)
            engine.backward(loss)
            engine.step()
        return loss

    for i in range(args.warmup):
        loss = one_step()
    if rank == 0:
        print(f"# warmup done, loss={loss.item():.4f}", flush=True)

    if dist.is_initialized() and world > 1:
        dist.barrier()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t_start = time.time()
    for i in range(args.steps):
        loss = one_step()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    if dist.is_initialized() and world > 1:
        dist.barrier()
    elapsed = time.time() - t_start

    # MAX over ranks
    if world > 1:
        t = torch.tensor([elapsed], device=device if
                         torch.cuda.is_available() else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    global_batch = B * args.grad_accum * world
    tokens_per_step = global_batch * S
    tokens_per_s = tokens_per_step * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    # model FLOPs (fwd+bwd+recompute): ~6*P*T (+6 for act-ckpt recompute ~8PT)
    n_params = sum(getattr(p, "ds_numel", p.numel())
                   for p in model.parameters())
    tflops_per_gpu = 6 * n_params * tokens_per_s / world / 1e12

    if rank == 0:
        print(json.dumps({
            "metric": "tokens/sec Llama-3-8B ZeRO-3",
            "value": round(tokens_per_s, 1),
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {"model": args.model, "global_batch": global_batch,
                       "seq_len": S, "parallelism": f"zero{args.zero_stage}_dp{world}",
                       "grad_accum": args.grad_accum,
                       "model_tflops_per_gpu_6PT": round(tflops_per_gpu, 1),
                       "final_loss": round(loss.item(), 4)},
        }), flush=True)


if __name__ == "__main__":
    main()
