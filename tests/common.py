"""Distributed-without-a-cluster test harness.

Modeled on the reference's DistributedTest mechanism
(`tests/unit/common.py:139,427`): N local processes, file-store rendezvous,
gloo on CPU / nccl(RCCL) on GPU.
"""
import os
import pickle
import tempfile
import traceback

import torch
import torch.distributed as dist
import torch.multiprocessing as mp

DEFAULT_TIMEOUT_S = 300


def _worker(rank, world_size, fn, args, kwargs, init_file, result_dir, backend):
    try:
        os.environ["RANK"] = str(rank)
        ndev = torch.cuda.device_count() if backend == "nccl" else 0
        os.environ["LOCAL_RANK"] = str(rank % ndev if ndev else rank)
        os.environ["WORLD_SIZE"] = str(world_size)
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        dist.init_process_group(backend=backend,
                                init_method=f"file://{init_file}",
                                rank=rank, world_size=world_size)
        if backend == "nccl":
            torch.cuda.set_device(rank % torch.cuda.device_count())
        result = fn(*args, **kwargs)
        with open(os.path.join(result_dir, f"result_{rank}.pkl"), "wb") as f:
            pickle.dump(("ok", result), f)
    except Exception as e:
        with open(os.path.join(result_dir, f"result_{rank}.pkl"), "wb") as f:
            pickle.dump(("error", f"{e}\n{traceback.format_exc()}"), f)
        raise
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def run_distributed(fn, world_size=2, backend=None, args=(), kwargs=None,
                    timeout=DEFAULT_TIMEOUT_S):
    """Run fn on world_size local ranks; returns list of per-rank results."""
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    kwargs = kwargs or {}
    with tempfile.TemporaryDirectory() as tmp:
        init_file = os.path.join(tmp, "init")
        ctx = mp.get_context("spawn")
        procs = []
        for rank in range(world_size):
            p = ctx.Process(target=_worker,
                            args=(rank, world_size, fn, args, kwargs,
                                  init_file, tmp, backend))
            p.start()
            procs.append(p)
        failed = []
        for rank, p in enumerate(procs):
            p.join(timeout)
            if p.is_alive():
                p.terminate()
                p.join(10)
                failed.append((rank, "timeout"))
            elif p.exitcode != 0:
                failed.append((rank, f"exit {p.exitcode}"))
        results = []
        for rank in range(world_size):
            path = os.path.join(tmp, f"result_{rank}.pkl")
            if os.path.exists(path):
                with open(path, "rb") as f:
                    status, payload = pickle.load(f)
                if status == "error":
                    raise AssertionError(
                        f"rank {rank} failed:\n{payload}")
                results.append(payload)
            else:
                results.append(None)
        if failed:
            raise AssertionError(f"distributed run failures: {failed}")
        return results
