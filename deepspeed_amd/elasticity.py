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
