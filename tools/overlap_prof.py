#!/usr/bin/env python3
"""Capture + analyze comm/compute overlap evidence for ZeRO-3.

train mode (default): 2 RCCL ranks on one GPU (or DSAMD_FORCE_STREAMS=1
single rank with --ranks 1) train llama-small ZeRO-3 bf16 for a few steps.
Wrap with rocprofv3 --kernel-trace to get per-kernel timestamps, then run

    python tools/overlap_prof.py analyze <trace_dir>

which reports what fraction of RCCL kernel time overlaps a concurrently
running compute kernel in the same process (the reduce-scatter /
all-gather streams doing their job during backward).
"""
import csv
import glob
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def worker(rank, world):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK="0",
                      MASTER_ADDR="127.0.0.1", MASTER_PORT="29553")
    import torch
    import torch.distributed as td
    td.init_process_group("nccl", rank=rank, world_size=world)
    torch.cuda.set_device(0)
    import deepspeed_amd
    from deepspeed_amd.models.llama import LLAMA_CONFIGS, LlamaForCausalLM
    cfg = LLAMA_CONFIGS["llama-small"]
    torch.manual_seed(100 + rank)
    with torch.device("cuda:0"):
        model = LlamaForCausalLM(cfg)
    config = {
        "train_micro_batch_size_per_gpu": 2,
        "optimizer": {"type": "AdamW", "params": {"lr": 3e-4}},
        "zero_optimization": {"stage": 3, "overlap_comm": True},
        "bf16": {"enabled": True},
    }
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config=config)
    data = torch.randint(0, cfg.vocab_size, (2, 1024), device="cuda:0")
    for _ in range(5):
        loss = engine(data, labels=data)
        engine.backward(loss)
        engine.step()
    torch.cuda.synchronize()
    if rank == 0:
        print(f"train done, loss={loss.item():.4f}", flush=True)


def train(world):
    if world == 1:
        os.environ["DSAMD_FORCE_STREAMS"] = "1"
        worker(0, 1)
        return
    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    ps = [ctx.Process(target=worker, args=(r, world)) for r in range(world)]
    [p.start() for p in ps]
    [p.join(600) for p in ps]
    rc = [p.exitcode for p in ps]
    print(f"worker exit codes: {rc}")
    sys.exit(0 if all(c == 0 for c in rc) else 1)


def _sniff(row, *cands):
    for c in cands:
        for k in row:
            if c.lower() in k.lower():
                return k
    return None


def analyze(trace_dir):
    files = glob.glob(os.path.join(trace_dir, "**", "*kernel_trace*.csv"),
                      recursive=True)
    if not files:
        print(f"no kernel_trace csv under {trace_dir}")
        sys.exit(1)
    for f in files:
        with open(f) as fh:
            rows = list(csv.DictReader(fh))
        if not rows:
            continue
        kname = _sniff(rows[0], "kernel_name", "name")
        kstart = _sniff(rows[0], "start")
        kend = _sniff(rows[0], "end")
        ivs_comm, ivs_comp = [], []
        for r in rows:
            try:
                s, e = int(r[kstart]), int(r[kend])
            except (ValueError, TypeError):
                continue
            n = r[kname].lower()
            (ivs_comm if ("nccl" in n or "rccl" in n or "AllGather" in r[kname]
                          or "ReduceScatter" in r[kname]) else ivs_comp) \
                .append((s, e))
        if not ivs_comm:
            print(f"{os.path.basename(f)}: no RCCL kernels "
                  f"({len(ivs_comp)} compute kernels)")
            continue
        ivs_comp.sort()
        # merge compute intervals
        merged = []
        for s, e in ivs_comp:
            if merged and s <= merged[-1][1]:
                merged[-1][1] = max(merged[-1][1], e)
            else:
                merged.append([s, e])
        total = sum(e - s for s, e in ivs_comm)
        overl = 0
        import bisect
        starts = [m[0] for m in merged]
        for s, e in ivs_comm:
            i = bisect.bisect_right(starts, e) - 1
            # walk merged intervals that intersect [s, e]
            j = max(0, bisect.bisect_right(starts, s) - 1)
            for m in merged[j:i + 1]:
                lo, hi = max(s, m[0]), min(e, m[1])
                if hi > lo:
                    overl += hi - lo
        comm_ms = total / 1e6
        print(f"{os.path.basename(f)}: {len(ivs_comm)} RCCL kernels, "
              f"{comm_ms:.2f} ms total, "
              f"{100.0 * overl / max(total, 1):.1f}% overlapped with "
              f"concurrent compute kernels")


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "analyze":
        analyze(sys.argv[2])
    else:
        world = int(sys.argv[sys.argv.index("--ranks") + 1]) \
            if "--ranks" in sys.argv else 2
        train(world)
