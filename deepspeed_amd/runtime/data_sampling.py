"""Data sampling: mmap indexed dataset + curriculum-aware loader.

Parity: reference `runtime/data_pipeline/data_sampling/indexed_dataset.py`
(mmap .bin/.idx token datasets) and `data_sampler.py`
(DeepSpeedDataSampler). Lite implementations: binary token storage with an
index of document offsets, and a loader wrapper that applies the
curriculum difficulty (sequence length) per step.
"""
import os
import struct

import numpy as np
import torch

_MAGIC = b"DSAMDIDX"
_DTYPE_CODES = {np.uint16: 1, np.int32: 2, np.int64: 3}
_CODE_DTYPES = {1: np.uint16, 2: np.int32, 3: np.int64}


class IndexedDatasetBuilder:
    """Writes <path>.bin (tokens) and <path>.idx (doc offsets)."""

    def __init__(self, path, dtype=np.uint16):
        self.path = path
        self.dtype = dtype
        self._bin = open(path + ".bin", "wb")
        self.sizes = []

    def add_item(self, tokens):
        arr = np.asarray(tokens, dtype=self.dtype)
        self._bin.write(arr.tobytes())
        self.sizes.append(len(arr))

    def finalize(self):
        self._bin.close()
        with open(self.path + ".idx", "wb") as f:
            f.write(_MAGIC)
            code = next(c for d, c in _DTYPE_CODES.items()
                        if d == self.dtype)
            f.write(struct.pack("<BQ", code, len(self.sizes)))
            np.asarray(self.sizes, dtype=np.int64).tofile(f)


class IndexedDataset(torch.utils.data.Dataset):
    """mmap-backed read of an IndexedDatasetBuilder output."""

    def __init__(self, path):
        with open(path + ".idx", "rb") as f:
            assert f.read(8) == _MAGIC, "bad index file"
            code, n = struct.unpack("<BQ", f.read(9))
            self.dtype = _CODE_DTYPES[code]
            self.sizes = np.fromfile(f, dtype=np.int64, count=n)
        self.offsets = np.concatenate([[0], np.cumsum(self.sizes)])
        self.data = np.memmap(path + ".bin", dtype=self.dtype, mode="r")

    def __len__(self):
        return len(self.sizes)

    def __getitem__(self, idx):
        lo, hi = self.offsets[idx], self.offsets[idx + 1]
        return torch.from_numpy(self.data[lo:hi].astype(np.int64))


class DeepSpeedDataSampler:
    """Curriculum-aware batch loader wrapper (ref data_sampler.py:349).

    Wraps any iterable of (input_ids[, labels]) batches; truncates the
    sequence dimension to the current curriculum difficulty, stepping the
    schedule once per batch.
    """

    def __init__(self, dataloader, curriculum_scheduler, seq_dim=1):
        self.dataloader = dataloader
        self.scheduler = curriculum_scheduler
        self.seq_dim = seq_dim
        self.global_step = 0

    def _truncate(self, t, n):
        if torch.is_tensor(t) and t.dim() > self.seq_dim and \
                t.shape[self.seq_dim] > n:
            return t.narrow(self.seq_dim, 0, n)
        return t

    def __iter__(self):
        for batch in self.dataloader:
            n = self.scheduler.update_difficulty(self.global_step)
            self.global_step += 1
            if isinstance(batch, (tuple, list)):
                yield type(batch)(self._truncate(x, n) for x in batch)
            else:
                yield self._truncate(batch, n)

    def __len__(self):
        return len(self.dataloader)

    def state_dict(self):
        return {"global_step": self.global_step,
                "scheduler": self.scheduler.state_dict()}

    def load_state_dict(self, sd):
        self.global_step = sd["global_step"]
        self.scheduler.load_state_dict(sd["scheduler"])


class DataAnalyzer:
    """Offline map-reduce over a dataset computing per-sample difficulty
    metrics for curriculum learning.

    Parity: reference `runtime/data_pipeline/data_analyzer.py`
    (DataAnalyzer/DistributedDataAnalyzer): each worker scans its shard
    of sample indices, applies `metric_function` per batch, and the
    results merge into (a) `<metric>_sample_to_metric` — metric value per
    sample index — and (b) `<metric>_metric_to_sample` — sample indices
    bucketed by metric value — both stored as IndexedDatasets so the
    curriculum sampler can mmap them.

    When torch.distributed is initialized the sample range splits across
    ranks and rank 0 merges (gloo-friendly; the scan is CPU work).
    """

    def __init__(self, dataset, metric_names, metric_functions,
                 output_path, batch_size=64, metric_dtype=np.int64):
        assert len(metric_names) == len(metric_functions)
        self.dataset = dataset
        self.metric_names = metric_names
        self.metric_functions = metric_functions
        self.output_path = output_path
        self.batch_size = batch_size
        self.metric_dtype = metric_dtype

    def _my_range(self):
        import torch.distributed as dist
        n = len(self.dataset)
        if dist.is_initialized() and dist.get_world_size() > 1:
            world, rank = dist.get_world_size(), dist.get_rank()
            per = (n + world - 1) // world
            return range(rank * per, min(n, (rank + 1) * per))
        return range(n)

    def run_map_reduce(self):
        import torch.distributed as dist
        os.makedirs(self.output_path, exist_ok=True)
        local = {m: [] for m in self.metric_names}
        idxs = list(self._my_range())
        for b0 in range(0, len(idxs), self.batch_size):
            batch_idx = idxs[b0:b0 + self.batch_size]
            batch = [self.dataset[i] for i in batch_idx]
            for name, fn in zip(self.metric_names, self.metric_functions):
                vals = fn(batch)
                if torch.is_tensor(vals):
                    vals = vals.tolist()
                local[name].extend(int(v) for v in vals)
        world = dist.get_world_size() if dist.is_initialized() else 1
        if world > 1:
            gathered = [None] * world
            dist.all_gather_object(gathered, (list(self._my_range()),
                                              local))
            if dist.get_rank() != 0:
                dist.barrier()
                return self.output_path
            merged = {m: {} for m in self.metric_names}
            for rng, loc in gathered:
                for m in self.metric_names:
                    for i, v in zip(rng, loc[m]):
                        merged[m][i] = v
            final = {m: [merged[m][i] for i in range(len(self.dataset))]
                     for m in self.metric_names}
        else:
            final = local
        for m in self.metric_names:
            vals = final[m]
            b = IndexedDatasetBuilder(
                os.path.join(self.output_path, f"{m}_sample_to_metric"),
                dtype=np.int64)
            for v in vals:
                b.add_item([v])
            b.finalize()
            buckets = {}
            for i, v in enumerate(vals):
                buckets.setdefault(v, []).append(i)
            b = IndexedDatasetBuilder(
                os.path.join(self.output_path, f"{m}_metric_to_sample"),
                dtype=np.int64)
            self._bucket_keys = sorted(buckets)
            for k in sorted(buckets):
                b.add_item(buckets[k])
            b.finalize()
            with open(os.path.join(self.output_path,
                                   f"{m}_metric_values.txt"), "w") as f:
                f.write("\n".join(str(k) for k in sorted(buckets)))
        if world > 1:
            dist.barrier()
        return self.output_path


def load_metric_index(output_path, metric_name):
    """(sample_to_metric IndexedDataset, sorted metric values,
    metric_to_sample IndexedDataset) for a DataAnalyzer output."""
    s2m = IndexedDataset(
        os.path.join(output_path, f"{metric_name}_sample_to_metric"))
    m2s = IndexedDataset(
        os.path.join(output_path, f"{metric_name}_metric_to_sample"))
    with open(os.path.join(output_path,
                           f"{metric_name}_metric_values.txt")) as f:
        values = [int(x) for x in f.read().split()]
    return s2m, values, m2s


class CurriculumMetricSampler(torch.utils.data.Sampler):
    """Difficulty-gated index sampler driven by a DataAnalyzer metric
    index (ref data_sampler.py:349 DeepSpeedDataSampler's metric path).

    Each epoch step the eligible pool = all samples whose metric value is
    <= the curriculum scheduler's current difficulty; indices are drawn
    (shuffled, DP-sharded) from that pool, so training sees easy samples
    first and the pool grows with the schedule.
    """

    def __init__(self, metric_index_path, metric_name, scheduler,
                 total_samples=None, shuffle=True, seed=0,
                 dp_rank=0, dp_world=1):
        s2m, values, m2s = load_metric_index(metric_index_path, metric_name)
        self.values = values                      # sorted distinct metrics
        self.buckets = [m2s[i].tolist() for i in range(len(values))]
        self.scheduler = scheduler
        self.total = total_samples or sum(len(b) for b in self.buckets)
        self.shuffle = shuffle
        self.seed = seed
        self.dp_rank = dp_rank
        self.dp_world = dp_world
        self.epoch = 0
        self.global_step = 0

    def set_epoch(self, epoch):
        self.epoch = epoch

    def _eligible(self):
        diff = self.scheduler.update_difficulty(self.global_step)
        pool = []
        for v, b in zip(self.values, self.buckets):
            if v <= diff:
                pool.extend(b)
        return pool or list(self.buckets[0])

    def __iter__(self):
        g = torch.Generator().manual_seed(self.seed + self.epoch)
        produced = 0
        while produced < self.total:
            pool = self._eligible()
            order = torch.randperm(len(pool), generator=g).tolist() \
                if self.shuffle else range(len(pool))
            for j in order:
                if produced >= self.total:
                    return
                if produced % self.dp_world == self.dp_rank:
                    yield pool[j]
                produced += 1
                self.global_step += 1

    def __len__(self):
        return (self.total + self.dp_world - 1) // self.dp_world

    def state_dict(self):
        return {"epoch": self.epoch, "global_step": self.global_step,
                "scheduler": self.scheduler.state_dict()}

    def load_state_dict(self, sd):
        self.epoch = sd["epoch"]
        self.global_step = sd["global_step"]
        self.scheduler.load_state_dict(sd["scheduler"])
