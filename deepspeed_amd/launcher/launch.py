"""Per-node process spawner: one process per GPU with distributed env.

Parity: reference `deepspeed/launcher/launch.py:145` (spawns per-GPU procs,
sets RANK/LOCAL_RANK/WORLD_SIZE/MASTER_*, SIGTERM kill-tree @333).
"""
import argparse
import json
import os
import signal
import subprocess
import sys

from ..utils.logging import logger


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--node_rank", type=int, default=0)
    p.add_argument("--master_addr", type=str, default="127.0.0.1")
    p.add_argument("--master_port", type=int, default=29500)
    p.add_argument("--world_info", type=str, default=None,
                   help="json: {hostname: [gpu ids]}")
    p.add_argument("--num_gpus", type=int, default=None)
    p.add_argument("training_script", type=str)
    p.add_argument("training_script_args", nargs=argparse.REMAINDER)
    return p.parse_args()


def main():
    args = parse_args()
    if args.world_info:
        world_info = json.loads(args.world_info)
        hosts = list(world_info.keys())
        local_gpus = world_info[hosts[args.node_rank]]
        ranks_before = sum(len(world_info[h])
                           for h in hosts[:args.node_rank])
        world_size = sum(len(v) for v in world_info.values())
    else:
        import torch
        n = args.num_gpus or max(torch.cuda.device_count(), 1)
        local_gpus = list(range(n))
        ranks_before = 0
        world_size = n

    procs = []
    for local_rank, gpu in enumerate(local_gpus):
        env = os.environ.copy()
        env["RANK"] = str(ranks_before + local_rank)
        env["LOCAL_RANK"] = str(local_rank)
        env["WORLD_SIZE"] = str(world_size)
        env["MASTER_ADDR"] = args.master_addr
        env["MASTER_PORT"] = str(args.master_port)
        cmd = [sys.executable, args.training_script,
               f"--local_rank={local_rank}"] + args.training_script_args
        procs.append(subprocess.Popen(cmd, env=env))

    def kill_all(signum, frame):
        for p in procs:
            if p.poll() is None:
                p.terminate()
        sys.exit(1)

    signal.signal(signal.SIGINT, kill_all)
    signal.signal(signal.SIGTERM, kill_all)

    rc = 0
    for p in procs:
        p.wait()
        if p.returncode != 0:
            rc = p.returncode
            for q in procs:
                if q.poll() is None:
                    q.terminate()
    sys.exit(rc)


if __name__ == "__main__":
    main()
