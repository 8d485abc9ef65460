"""Elastic training config: valid (global batch, GPU count) families.

Parity: reference `elasticity/elasticity.py:238` (compute_elastic_config),
candidate search @27-195. Training can resume at any compatible GPU count
without changing convergence (paired with universal checkpoints).
"""
from .utils.logging import logger  # noqa: F401  (kept for parity surface)


class ElasticityError(Exception):
    pass


def get_valid_gpus(batch_size, micro_batches, min_valid_gpus, max_valid_gpus):
    valid = []
    for mb in micro_batches:
        if batch_size % mb:
            continue
        max_gpus = batch_size // mb
        for i in range(1, max_gpus + 1):
            if max_gpus % i == 0:
                g = max_gpus // i
                if min_valid_gpus <= g <= max_valid_gpus and g not in valid:
                    valid.append(g)
    return sorted(valid)


def get_best_candidates(candidate_batch_sizes, micro_batches, min_gpus,
                        max_gpus, prefer_larger):
    max_valid = 0
    best_batch = None
    best_gpus = None
    for batch in candidate_batch_sizes:
        gpus = get_valid_gpus(batch, micro_batches, min_gpus, max_gpus)
        if (len(gpus) > max_valid or
                (prefer_larger and len(gpus) == max_valid and
                 best_batch is not None and batch > best_batch)):
            max_valid = len(gpus)
            best_batch = batch
            best_gpus = gpus
    return best_batch, best_gpus


def _candidate_batches(base, max_acc_step, micro_batches):
    candidates = set()
    for mb in micro_batches:
        for acc in range(1, max_acc_step + 1):
            b = mb * acc
            if b <= base:
                candidates.add((base // b) * b)
    return sorted(candidates, reverse=True)


def compute_elastic_config(ds_config, target_deepspeed_version=None,
                           world_size=0, return_microbatch=False):
    """From an `elasticity` config section -> (final_batch, valid_gpus,
    micro_batch[!])."""
    ec = ds_config.get("elasticity", {})
    if not ec.get("enabled", False):
        raise ElasticityError("elasticity not enabled in config")
    max_batch = ec["max_train_batch_size"]
    micro_batches = ec["micro_batch_sizes"]
    min_gpus = ec.get("min_gpus", 1)
    max_gpus = ec.get("max_gpus", 10000)
    prefer_larger = ec.get("prefer_larger_batch", True)
    max_acc = ec.get("max_acc_step", 64)

    candidates = _candidate_batches(max_batch, max_acc, micro_batches)
    final_batch, valid_gpus = get_best_candidates(
        candidates, micro_batches, min_gpus, max_gpus, prefer_larger)
    if final_batch is None:
        raise ElasticityError("no compatible (batch, gpus) configuration")

    if world_size > 0:
        if world_size not in valid_gpus:
            raise ElasticityError(
                f"world size {world_size} not in valid gpus {valid_gpus}")
        # pick largest micro batch that divides the per-world batch
        per_gpu = final_batch // world_size
        micro = max((m for m in micro_batches if per_gpu % m == 0),
                    default=None)
        if micro is None:
            raise ElasticityError("no valid micro batch for world size")
        if return_microbatch:
            return final_batch, valid_gpus, micro
        return final_batch, valid_gpus
    if return_microbatch:
        return final_batch, valid_gpus, None
    return final_batch, valid_gpus


def _try_import_elastic_agent():
    try:
        from torch.distributed.elastic.agent.server.local_elastic_agent \
            import LocalElasticAgent
        return LocalElasticAgent
    except ImportError:
        return None


_Base = _try_import_elastic_agent()

if _Base is not None:

    class DSElasticAgent(_Base):
        """Elastic agent with DeepSpeed worker environment.

        Parity: reference `elasticity/elastic_agent.py:32` (DSElasticAgent
        overriding `_set_master_addr_port` / worker env so restarted
        workers re-enter `deepspeed.initialize` cleanly at the new scale).
        MI355X note: rendezvous over 127.0.0.1 on single-node pools (the
        container hostname may not resolve); on restart, workers should
        reload the latest checkpoint and call `compute_elastic_config`
        with the surviving GPU count to re-derive the batch triple.
        """

        def __init__(self, spec, logs_specs=None, start_method="spawn",
                     exit_barrier_timeout=300, log_line_prefix_template=None,
                     ds_env=None, **kw):
            self._ds_env = dict(ds_env or {})
            try:
                super().__init__(
                    spec, logs_specs=logs_specs, start_method=start_method,
                    exit_barrier_timeout=exit_barrier_timeout,
                    log_line_prefix_template=log_line_prefix_template, **kw)
            except TypeError:  # older torch signature
                super().__init__(spec, start_method=start_method,
                                 exit_barrier_timeout=exit_barrier_timeout)

        def _start_workers(self, worker_group):
            import os
            spec = worker_group.spec
            for k, v in self._ds_env.items():
                os.environ.setdefault(k, str(v))
            os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
            # one process per GPU: LOCAL_RANK -> HIP device is implicit
            os.environ.setdefault("NCCL_MIN_NCHANNELS", "32")
            worker_ids = super()._start_workers(worker_group)
            from .utils.logging import logger
            logger.info(
                f"DSElasticAgent: (re)started {spec.local_world_size} "
                f"workers for group role={spec.role}")
            return worker_ids
else:  # pragma: no cover - torch without elastic
    DSElasticAgent = None
