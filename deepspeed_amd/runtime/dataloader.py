"""DeepSpeedDataLoader — DP-sharded loader (ref runtime/dataloader.py:41)."""
import torch
from torch.utils.data import DataLoader, DistributedSampler, RandomSampler


class RepeatingLoader:
    def __init__(self, loader):
        self.loader = loader
        self.data_iter = iter(self.loader)

    def __iter__(self):
        return self

    def __len__(self):
        return len(self.loader)

    def __next__(self):
        try:
            return next(self.data_iter)
        except StopIteration:
            self.data_iter = iter(self.loader)
            return next(self.data_iter)


def DeepSpeedDataLoader(dataset, batch_size, data_parallel_world_size=1,
                        data_parallel_rank=0, collate_fn=None,
                        num_local_io_workers=0, pin_memory=None):
    if data_parallel_world_size > 1:
        sampler = DistributedSampler(dataset,
                                     num_replicas=data_parallel_world_size,
                                     rank=data_parallel_rank)
    else:
        sampler = RandomSampler(dataset)
    if pin_memory is None:
        pin_memory = torch.cuda.is_available()
    return DataLoader(dataset, batch_size=batch_size, sampler=sampler,
                      collate_fn=collate_fn, num_workers=num_local_io_workers,
                      pin_memory=pin_memory, drop_last=True)
