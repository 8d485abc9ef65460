"""NUMA-aware CPU binding for the offload tier (host EPYC of an MI355X
node).

Parity: reference `deepspeed/utils/numa.py` (core binding for CPU-Adam
workers). The OpenMP cpu_adam_step is DRAM-bandwidth bound; binding each
rank's host threads to the NUMA node closest to its GPU avoids cross-socket
traffic.
"""
import os

from .logging import logger


def get_numa_node_count():
    base = "/sys/devices/system/node"
    try:
        return len([d for d in os.listdir(base) if d.startswith("node")
                    and d[4:].isdigit()])
    except OSError:
        return 1


def get_cores_for_node(node):
    path = f"/sys/devices/system/node/node{node}/cpulist"
    try:
        with open(path) as f:
            spec = f.read().strip()
    except OSError:
        return list(range(os.cpu_count() or 1))
    cores = []
    for part in spec.split(","):
        if "-" in part:
            a, b = part.split("-")
            cores.extend(range(int(a), int(b) + 1))
        elif part:
            cores.append(int(part))
    return cores


def bind_to_numa_node(local_rank, num_local_ranks=8):
    """Pin this process to the NUMA node serving its GPU (round-robin)."""
    n_nodes = get_numa_node_count()
    if n_nodes <= 1:
        return False
    node = local_rank * n_nodes // max(num_local_ranks, 1)
    cores = get_cores_for_node(node)
    if not cores:
        return False
    try:
        os.sched_setaffinity(0, cores)
        os.environ.setdefault("OMP_NUM_THREADS", str(len(cores)))
        logger.info(f"rank local:{local_rank} bound to NUMA node {node} "
                    f"({len(cores)} cores)")
        return True
    except OSError:
        return False
