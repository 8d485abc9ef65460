"""Sparse gradient support: compressed row sparse tensor + allreduce.

Parity: reference `runtime/sparse_tensor.py:13` (SparseTensor) and
`runtime/engine.py:3800-3860` (sparse_allreduce_* — sparse embedding
grads average by all-gathering (indices, values) rather than densifying
a vocab-sized tensor).
"""
import torch

from .. import comm as dist


class SparseTensor:
    """Row-compressed view of a (possibly torch-sparse) gradient."""

    def __init__(self, dense_tensor=None):
        self.orig_dense_tensor = dense_tensor
        if dense_tensor is not None:
            self.is_sparse = dense_tensor.is_sparse
            self.dtype = dense_tensor.dtype
            if self.is_sparse:
                dense_tensor = dense_tensor.coalesce()
                self.indices = dense_tensor.indices().flatten()
                self.values = dense_tensor.values()
            else:
                nz = torch.sum(dense_tensor.abs(), dim=1).nonzero().flatten()
                self.indices = nz
                self.values = dense_tensor[nz]
            self.dense_size = list(dense_tensor.size())
        else:
            self.indices = None
            self.values = None
            self.dense_size = None

    def to_coo_tensor(self):
        return torch.sparse_coo_tensor(self.indices.unsqueeze(0),
                                       self.values, self.dense_size)

    def to_dense(self):
        out = torch.zeros(self.dense_size, dtype=self.values.dtype,
                          device=self.values.device)
        out.index_add_(0, self.indices, self.values)
        return out

    @staticmethod
    def type():
        return "deepspeed.SparseTensor"

    def sparse_size(self):
        return self.indices.numel() + self.values.numel()


def sparse_allreduce(sparse, dp_group=None, dp_world_size=None):
    """Average a SparseTensor across the DP group by exchanging
    (indices, values) — comm bytes scale with touched rows, not vocab
    size. Duplicated rows across ranks sum (then divide by world)."""
    world = dp_world_size or dist.get_world_size(dp_group)
    if world <= 1:
        return sparse
    values = sparse.values / world
    idx_list = _allgather_uneven(sparse.indices, dp_group)
    val_list = _allgather_uneven(values, dp_group)
    sparse.indices = torch.cat(idx_list)
    sparse.values = torch.cat(val_list)
    return sparse


def _allgather_uneven(t, group):
    """All-gather tensors whose first dim differs per rank."""
    world = dist.get_world_size(group)
    n = torch.tensor([t.shape[0]], dtype=torch.long, device=t.device)
    sizes = [torch.zeros_like(n) for _ in range(world)]
    dist.all_gather(sizes, n, group=group)
    sizes = [int(s.item()) for s in sizes]
    mx = max(sizes) if sizes else 0
    pad_shape = (mx,) + tuple(t.shape[1:])
    padded = torch.zeros(pad_shape, dtype=t.dtype, device=t.device)
    if t.shape[0]:
        padded[:t.shape[0]] = t
    outs = [torch.zeros_like(padded) for _ in range(world)]
    dist.all_gather(outs, padded, group=group)
    return [o[:s] for o, s in zip(outs, sizes)]


def sparse_allreduce_bucket(bucket, dp_group=None):
    return [sparse_allreduce(s, dp_group) for s in bucket]
