"""Variable batch size + LR scaling for length-heterogeneous data.

Parity: reference
`runtime/data_pipeline/data_sampling/variable_batch_size_and_lr.py`
(`batch_by_seqlens:23`, `scale_lr:149`,
`dataloader_for_variable_batch_size:165`, `VariableBatchSizeLR:226`).

Packs samples into micro-batches capped by a token budget (so a batch of
short sequences holds many samples, a batch of long ones few) and scales
the learning rate per batch by the realized batch size (linear or sqrt
rule), undoing the gradient-noise change that variable batch sizes cause.
"""
import math
import random

import torch
from torch.utils.data import DataLoader, DistributedSampler

from ..utils.logging import logger


def batch_by_seqlens(seqlens, max_tokens, min_batch_size=1,
                     max_batch_size=None, sequence_picking_order="dataloader",
                     seed=None):
    """Greedy token-budget packing.

    Returns (microbatch_ids, batch_sizes, batch_max_seqlens):
    microbatch_ids[i] = (batch_id, [sample indices]); batch_sizes[i] =
    number of samples (for LR scaling); batch_max_seqlens[i] = longest
    sample in micro-batch i (for optional padding).
    """
    assert sequence_picking_order in ("dataloader", "random", "seqlen")
    order = list(range(len(seqlens)))
    if sequence_picking_order == "random":
        random.Random(seed).shuffle(order)
    elif sequence_picking_order == "seqlen":
        order.sort(key=lambda i: seqlens[i])

    too_long = [i for i in order if seqlens[i] > max_tokens]
    if too_long:
        logger.warning(f"variable-batch: {len(too_long)} samples exceed "
                       f"max_tokens={max_tokens} and are skipped")
        order = [i for i in order if seqlens[i] <= max_tokens]

    microbatch_ids, batch_sizes, batch_max_seqlens = [], [], []
    cur, cur_tokens, cur_max = [], 0, 0
    for i in order:
        n = seqlens[i]
        over_tokens = cur_tokens + n > max_tokens
        over_count = max_batch_size is not None and \
            len(cur) + 1 > max_batch_size
        if cur and (over_tokens or over_count):
            if len(cur) >= min_batch_size:
                microbatch_ids.append((len(microbatch_ids), cur))
                batch_sizes.append(len(cur))
                batch_max_seqlens.append(cur_max)
            cur, cur_tokens, cur_max = [], 0, 0
        cur.append(i)
        cur_tokens += n
        cur_max = max(cur_max, n)
    if cur and len(cur) >= min_batch_size:
        microbatch_ids.append((len(microbatch_ids), cur))
        batch_sizes.append(len(cur))
        batch_max_seqlens.append(cur_max)
    return microbatch_ids, batch_sizes, batch_max_seqlens


def scale_lr(base_batch_size, batch_size, base_lr=1.0, method="linear"):
    """Linear rule (Goyal et al.) or sqrt rule (Krizhevsky)."""
    if method == "linear":
        return base_lr * batch_size / base_batch_size
    if method == "sqrt":
        return base_lr * math.sqrt(batch_size / base_batch_size)
    if method is None or str(method).upper() == "NONE":
        return base_lr
    raise ValueError(f"unknown LR scaling method {method}")


class VariableBatchSizeLR:
    """Scales the optimizer's (or an inner scheduler's) LR per batch by
    the realized batch size. Drop-in torch-scheduler-shaped object."""

    def __init__(self, optimizer, base_batch_size, batch_sizes,
                 lr_scaling_method="linear", inner_scheduler=None,
                 verbose=False):
        self.optimizer = optimizer
        self.base_batch_size = base_batch_size
        self.batch_sizes = list(batch_sizes)
        self.method = lr_scaling_method
        self.inner = inner_scheduler
        self.verbose = verbose
        self._batch = 0
        self.base_lrs = [g["lr"] for g in optimizer.param_groups]
        self._last_lr = list(self.base_lrs)
        self.step(0)

    def state_dict(self):
        return {"batch": self._batch,
                "base_lrs": self.base_lrs,
                "inner": self.inner.state_dict() if self.inner else None}

    def load_state_dict(self, sd):
        self._batch = sd["batch"]
        self.base_lrs = sd["base_lrs"]
        if self.inner and sd.get("inner"):
            self.inner.load_state_dict(sd["inner"])

    def get_last_lr(self):
        return self._last_lr

    def step(self, batch_id=None):
        if batch_id is None:
            batch_id = self._batch
        self._batch = batch_id + 1
        if self.inner is not None:
            self.inner.step()
            self.base_lrs = self.inner.get_last_lr()
        if not self.batch_sizes:
            return
        bs = self.batch_sizes[batch_id % len(self.batch_sizes)]
        self._last_lr = [scale_lr(self.base_batch_size, bs, lr, self.method)
                        for lr in self.base_lrs]
        for g, lr in zip(self.optimizer.param_groups, self._last_lr):
            g["lr"] = lr
        if self.verbose:
            logger.info(f"variable-batch LR: batch {batch_id} size {bs} "
                        f"-> lr {self._last_lr}")


def dataloader_for_variable_batch_size(
        dataset, microbatch_ids, batch_max_seqlens=None,
        dataloader_rank=0, dataloader_num_replicas=1, collate_fn=None,
        num_workers=0, pin_memory=False, sample_padding_fn=None):
    """DataLoader over packed micro-batches, interleaved across DP ranks."""
    sampler = DistributedSampler(dataset=microbatch_ids,
                                 num_replicas=dataloader_num_replicas,
                                 rank=dataloader_rank, shuffle=False,
                                 drop_last=False)

    def collate_wrapper(items):
        batch = []
        for batch_id, ids in items:
            data = [dataset[i] for i in ids]
            if sample_padding_fn is not None and batch_max_seqlens:
                data = [sample_padding_fn(s, batch_max_seqlens[batch_id])
                        for s in data]
            batch += data
        return collate_fn(batch) if collate_fn else batch

    return DataLoader(dataset=microbatch_ids, batch_size=1, sampler=sampler,
                      num_workers=num_workers, collate_fn=collate_wrapper,
                      pin_memory=pin_memory)


def get_variable_batch_dataloader_and_lr(
        dataset, seqlens, max_tokens, optimizer, base_batch_size,
        lr_scaling_method="linear", inner_scheduler=None,
        sequence_picking_order="dataloader", seed=None,
        dataloader_rank=0, dataloader_num_replicas=1, collate_fn=None,
        sample_padding_fn=None, **packing_kwargs):
    """One-call glue (ref
    get_dataloader_and_lr_scheduler_for_variable_batch_size:432)."""
    mb_ids, batch_sizes, max_lens = batch_by_seqlens(
        seqlens, max_tokens,
        sequence_picking_order=sequence_picking_order, seed=seed,
        **packing_kwargs)
    dl = dataloader_for_variable_batch_size(
        dataset, mb_ids, max_lens, dataloader_rank=dataloader_rank,
        dataloader_num_replicas=dataloader_num_replicas,
        collate_fn=collate_fn, sample_padding_fn=sample_padding_fn)
    sched = VariableBatchSizeLR(optimizer, base_batch_size, batch_sizes,
                                lr_scaling_method, inner_scheduler)
    return dl, sched
