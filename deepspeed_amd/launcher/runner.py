"""`deepspeed` CLI: hostfile parsing + local/multi-node launching.

Parity: reference `deepspeed/launcher/runner.py` (`parse_args:48`,
`fetch_hostfile:230`, `main:436`) and `multinode_runner.py` (PDSH/MPI
runners). Single-node launches spawn launch.py directly; multi-node uses
pdsh/ssh when available.
"""
import argparse
import json
import os
import shlex
import subprocess
import sys

from ..utils.logging import logger


def parse_args(args=None):
    p = argparse.ArgumentParser(
        description="deepspeed-amd launcher (MI355X)")
    p.add_argument("-H", "--hostfile", type=str, default="/job/hostfile")
    p.add_argument("-i", "--include", type=str, default="")
    p.add_argument("-e", "--exclude", type=str, default="")
    p.add_argument("--num_nodes", type=int, default=-1)
    p.add_argument("--num_gpus", type=int, default=-1)
    p.add_argument("--master_port", type=int, default=29500)
    p.add_argument("--master_addr", type=str, default="")
    p.add_argument("--launcher", type=str, default="pdsh",
                   choices=["pdsh", "ssh", "local", "openmpi", "mpich",
                            "impi", "slurm", "mvapich"])
    p.add_argument("--launcher_args", type=str, default="")
    p.add_argument("user_script", type=str)
    p.add_argument("user_args", nargs=argparse.REMAINDER)
    return p.parse_args(args)


def fetch_hostfile(path):
    """Parse 'hostname slots=N' lines -> {host: slots}."""
    if not os.path.exists(path):
        return None
    resources = {}
    with open(path) as f:
        for line in f:
            line = line.split("#")[0].strip()
            if not line:
                continue
            parts = line.split()
            host = parts[0]
            slots = 1
            for tok in parts[1:]:
                if tok.startswith("slots="):
                    slots = int(tok.split("=")[1])
            resources[host] = slots
    return resources


def _filter_resources(resources, include, exclude):
    if include:
        keep = {}
        for spec in include.split("@"):
            host = spec.split(":")[0]
            if host in resources:
                if ":" in spec:
                    keep[host] = [int(x) for x in
                                  spec.split(":")[1].split(",")]
                else:
                    keep[host] = list(range(resources[host]))
        return keep
    out = {h: list(range(s)) for h, s in resources.items()}
    for host in (exclude.split(",") if exclude else []):
        out.pop(host.split(":")[0], None)
    return out


def main(args=None):
    args = parse_args(args)
    resources = fetch_hostfile(args.hostfile)

    if resources is None or args.launcher == "local":
        # single node
        import torch
        n = args.num_gpus if args.num_gpus > 0 else \
            max(torch.cuda.device_count(), 1)
        cmd = [sys.executable, "-m", "deepspeed_amd.launcher.launch",
               f"--master_port={args.master_port}",
               f"--num_gpus={n}", args.user_script] + args.user_args
        os.execvpe(sys.executable, cmd, os.environ.copy())
        return

    world = _filter_resources(resources, args.include, args.exclude)
    if args.num_nodes > 0:
        world = dict(list(world.items())[:args.num_nodes])
    master_addr = args.master_addr or list(world.keys())[0]

    if args.launcher in ("openmpi", "mpich", "impi", "slurm", "mvapich",
                         "pdsh"):
        from .multinode_runner import get_runner
        runner = get_runner(args.launcher, args, world)
        if not runner.backend_exists():
            logger.error(f"{args.launcher} backend not found on PATH")
            sys.exit(1)
        runner.add_export("MASTER_ADDR", master_addr)
        runner.add_export("MASTER_PORT", str(args.master_port))
        cmd = runner.get_cmd()
        logger.info(f"launching: {' '.join(map(str, cmd))}")
        sys.exit(subprocess.call(cmd))

    # ssh fallback: fan launch.py out per node ourselves
    world_info = json.dumps({h: g for h, g in world.items()})
    procs = []
    for node_rank, host in enumerate(world):
        launch_cmd = (
            f"cd {os.getcwd()} && {sys.executable} -m "
            f"deepspeed_amd.launcher.launch "
            f"--node_rank={node_rank} --master_addr={master_addr} "
            f"--master_port={args.master_port} "
            f"--world_info={shlex.quote(world_info)} "
            f"{args.user_script} {' '.join(args.user_args)}")
        cmd = ["ssh", host, launch_cmd]
        procs.append(subprocess.Popen(cmd))
    rc = 0
    for p in procs:
        p.wait()
        rc = rc or p.returncode
    sys.exit(rc)


if __name__ == "__main__":
    main()
