"""Indexed dataset round-trip + curriculum data sampler."""
import tempfile
import os

import numpy as np
import torch

from deepspeed_amd.runtime.data_sampling import (DeepSpeedDataSampler,
                                                 IndexedDataset,
                                                 IndexedDatasetBuilder)
from deepspeed_amd.runtime.data_pipeline import CurriculumScheduler


def test_indexed_dataset_roundtrip():
    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "ds")
        b = IndexedDatasetBuilder(path, dtype=np.uint16)
        docs = [[1, 2, 3], [42], list(range(100))]
        for doc in docs:
            b.add_item(doc)
        b.finalize()
        ds = IndexedDataset(path)
        assert len(ds) == 3
        for i, doc in enumerate(docs):
            assert torch.equal(ds[i], torch.tensor(doc, dtype=torch.long))


def test_curriculum_sampler_truncates():
    cs = CurriculumScheduler({"curriculum_type": "fixed_linear",
                              "min_difficulty": 4, "max_difficulty": 16,
                              "schedule_config": {
                                  "total_curriculum_step": 4,
                                  "difficulty_step": 4}})
    batches = [(torch.arange(32).reshape(2, 16),) for _ in range(6)]
    sampler = DeepSpeedDataSampler(batches, cs)
    lens = [b[0].shape[1] for b in sampler]
    assert lens[0] == 4            # min difficulty at step 0
    assert lens[-1] == 16          # full length at the end
    assert all(a <= b for a, b in zip(lens, lens[1:]))  # monotone
