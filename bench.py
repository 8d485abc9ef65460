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
    p.add_argument("--model", type=str, default="llama3-8b",
                   help="llama3-8b|llama3-70b|llama-small|mixtral-8x7b|...")
    p.add_argument("--seq-len", type=int, default=4096)
    p.add_argument("--sp", type=int, default=1,
                   help="Ulysses sequence-parallel degree (config #3: "
                   "--sp 8 --seq-len 32768)")
    p.add_argument("--ep", type=int, default=1,
                   help="expert-parallel degree for MoE models")
    p.add_argument("--offload-param", type=str, default="none",
                   choices=["none", "cpu", "nvme"],
                   help="ZeRO-Infinity parameter tier")
    p.add_argument("--no-pin", action="store_true",
                   help="pageable host offload buffers (skip pinning "
                        "cost at 70B scale)")
    p.add_argument("--offload", type=str, default="none",
                   choices=["none", "cpu", "nvme"])
    p.add_argument("--micro-batch", type=int, default=4)
    p.add_argument("--grad-accum", type=int, default=2)
    p.add_argument("--zero-stage", type=int, default=3)
    p.add_argument("--fp8-mlp", action="store_true",
                   help="compute MLP (gate/up/down) GEMMs in fp8 via "
                        "hipBLASLt _scaled_mm (bf16 master numerics)")
    p.add_argument("--fp8-attn-proj", action="store_true",
                   help="also run qkv/o projections in fp8")
    p.add_argument("--fp8-head", action="store_true",
                   help="also run the lm_head GEMM in fp8")
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
    from deepspeed_amd.models.llama import (LLAMA_CONFIGS, LlamaForCausalLM,
                                            enable_ulysses)
    from deepspeed_amd.models.mixtral import (MIXTRAL_CONFIGS,
                                              MixtralForCausalLM)

    dist.init_distributed()
    rank = dist.get_rank()
    world = dist.get_world_size()
    device = torch.device("cuda", dist.get_local_rank()) \
        if torch.cuda.is_available() else torch.device("cpu")
    if torch.cuda.is_available():
        torch.cuda.set_device(device)

    from deepspeed_amd.models.bert import BERT_CONFIGS, BertForPreTraining
    is_bert = args.model in BERT_CONFIGS
    if args.model in MIXTRAL_CONFIGS:
        cfg = MIXTRAL_CONFIGS[args.model]
        cfg.ep_size = min(args.ep, world) if args.ep > 1 else 1
        model_cls = MixtralForCausalLM
    elif is_bert:
        cfg = BERT_CONFIGS[args.model]
        model_cls = BertForPreTraining
        args.seq_len = min(args.seq_len, cfg.max_position_embeddings)
    else:
        cfg = LLAMA_CONFIGS[args.model]
        model_cls = LlamaForCausalLM
    if not is_bert:
        cfg.activation_checkpointing = args.act_ckpt
        if args.seq_len > cfg.max_position_embeddings:
            cfg.max_position_embeddings = args.seq_len
    torch.manual_seed(1234 + rank)
    t0 = time.time()
    # build directly on device: 8B bf16 = 16 GB, fits trivially in 288 GB
    with torch.device(device):
        model = model_cls(cfg)
    if args.fp8_mlp or args.fp8_attn_proj or args.fp8_head:
        from deepspeed_amd.ops.fp8_linear import Fp8Linear
        tags = []
        if args.fp8_mlp:
            tags += ["gate_proj", "up_proj", "down_proj"]
        if args.fp8_attn_proj:
            tags += ["q_proj", "k_proj", "v_proj", "o_proj"]
        if args.fp8_head:
            tags += ["lm_head"]
        n_conv = Fp8Linear.convert(model, include=tags)
        if rank == 0:
            print(f"# fp8 linear: {n_conv} layers converted", flush=True)
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
    if args.offload != "none":
        ds_config["zero_optimization"]["offload_optimizer"] = {
            "device": args.offload, "pin_memory": not args.no_pin}
    if args.offload_param != "none":
        ds_config["zero_optimization"]["offload_param"] = {
            "device": args.offload_param, "pin_memory": not args.no_pin}
        ds_config["zero_optimization"]["sub_group_size"] = int(5e8)
    if args.sp > 1:
        ds_config["sequence_parallel"] = {
            "sequence_parallel_size": args.sp}
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config=ds_config)

    sp_rank, sp_world = 0, 1
    if args.sp > 1:
        from deepspeed_amd.comm import groups as _grp
        spg = _grp.get_sequence_parallel_group()
        enable_ulysses(engine.module, spg)
        sp_rank = dist.get_rank(spg)
        sp_world = dist.get_world_size(spg)

    S = args.seq_len
    B = args.micro_batch
    # rotating synthetic batches: fresh tokens each micro-step (no caching)
    S_local = S // sp_world
    seq_off = sp_rank * S_local
    n_bufs = 8
    bufs = [torch.randint(0, cfg.vocab_size, (B, S_local), device=device)
            for _ in range(n_bufs)]
    step_idx = [0]

    def one_step():
        for _ in range(args.grad_accum):
            data = bufs[step_idx[0] % n_bufs]
            step_idx[0] += 1
            if sp_world > 1:
                loss = engine(data, labels=data, seq_offset=seq_off)
            else:
                loss = engine(data, labels=data)
            engine.backward(loss)
            engine.step()
        return loss

    for i in range(args.warmup):
        loss = one_step()
    if rank == 0 and args.warmup > 0:
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

    global_batch = B * args.grad_accum * (world // sp_world)
    # with SP, ranks of one SP group share a sample (S_local tokens each)
    tokens_per_step = B * args.grad_accum * world * S_local
    tokens_per_s = tokens_per_step * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    # model FLOPs (fwd+bwd+recompute): ~6*P*T (+6 for act-ckpt recompute ~8PT)
    n_params = sum(getattr(p, "ds_numel", p.numel())
                   for p in model.parameters())
    tflops_per_gpu = 6 * n_params * tokens_per_s / world / 1e12

    if rank == 0:
        metric = ("tokens/sec Llama-3-8B ZeRO-3"
                  if args.model == "llama3-8b" and args.zero_stage == 3
                  else f"tokens/sec {args.model} zero{args.zero_stage}")
        print(json.dumps({
            "metric": metric,
            "value": round(tokens_per_s, 1),
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": ("bf16+fp8gemm" if (args.fp8_mlp or args.fp8_attn_proj
                                          or args.fp8_head) else "bf16"),
            "data": "synthetic",
            "config": {"model": args.model, "global_batch": global_batch,
                       "seq_len": S,
                       "parallelism": (f"zero{args.zero_stage}_"
                                       f"dp{world // sp_world}" +
                                       (f"_offload-{args.offload}"
                                        if args.offload != "none" else "") +
                                       (f"_param-{args.offload_param}"
                                        if args.offload_param != "none"
                                        else "") +
                                       (f"_sp{sp_world}" if sp_world > 1
                                        else "") +
                                       (f"_ep{args.ep}" if args.ep > 1
                                        else "")),
                       "grad_accum": args.grad_accum,
                       "model_tflops_per_gpu_6PT": round(tflops_per_gpu, 1),
                       "final_loss": round(loss.item(), 4)},
        }), flush=True)


if __name__ == "__main__":
    main()
