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


def test_data_analyzer_map_reduce(tmp_path):
    import numpy as np
    from deepspeed_amd.runtime.data_sampling import (DataAnalyzer,
                                                     load_metric_index)
    data = [list(range(3 + (i % 5))) for i in range(23)]  # var-length docs

    def seqlen_metric(batch):
        return [len(x) for x in batch]

    an = DataAnalyzer(data, ["seqlen"], [seqlen_metric], str(tmp_path),
                      batch_size=4)
    an.run_map_reduce()
    s2m, values, m2s = load_metric_index(str(tmp_path), "seqlen")
    assert len(s2m) == 23
    assert [int(s2m[i][0]) for i in range(23)] == [3 + (i % 5)
                                                   for i in range(23)]
    assert values == [3, 4, 5, 6, 7]
    # bucket for value 3 holds exactly the samples with len 3
    b3 = [int(x) for x in m2s[0]]
    assert b3 == [i for i in range(23) if (i % 5) == 0]


def test_engine_curriculum_wiring():
    import torch
    from tests.common import run_distributed

    run_distributed(_curriculum_engine, world_size=1)


def _curriculum_engine():
    import torch
    import deepspeed_amd as ds
    from tests.simple_model import SimpleModel
    cfg = {
        "train_micro_batch_size_per_gpu": 2,
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
        "curriculum_learning": {
            "enabled": True, "curriculum_type": "seqlen",
            "min_difficulty": 8, "max_difficulty": 32,
            "schedule_type": "fixed_linear",
            "schedule_config": {"total_curriculum_step": 4,
                                "difficulty_step": 8},
        },
    }
    engine, _, _, _ = ds.initialize(model=SimpleModel(32), config=cfg)
    assert engine.curriculum_scheduler is not None
    d0 = engine.curriculum_scheduler.get_current_difficulty()
    x = torch.randn(2, 32)
    y = torch.randn(2, 32)
    for _ in range(5):
        loss = engine(x, y)
        engine.backward(loss)
        engine.step()
    d1 = engine.curriculum_scheduler.get_current_difficulty()
    assert d1 > d0, (d0, d1)
    assert d1 == 32


def test_curriculum_metric_sampler_gates_by_difficulty(tmp_path):
    """Samples with metric above the current difficulty are excluded; the
    pool grows as the schedule advances."""
    import torch
    from deepspeed_amd.runtime.data_pipeline import CurriculumScheduler
    from deepspeed_amd.runtime.data_sampling import (CurriculumMetricSampler,
                                                     DataAnalyzer)
    data = [torch.arange(n) for n in
            (3, 3, 5, 5, 8, 8, 12, 12, 20, 20)]
    an = DataAnalyzer(data, ["seqlen"],
                      [lambda batch: [len(x) for x in batch]],
                      str(tmp_path))
    an.run_map_reduce()
    sched = CurriculumScheduler({
        "curriculum_type": "seqlen", "min_difficulty": 5,
        "max_difficulty": 20,
        "schedule_type": "fixed_linear",
        "schedule_config": {"total_curriculum_step": 20,
                            "difficulty_step": 1}})
    samp = CurriculumMetricSampler(str(tmp_path), "seqlen", sched,
                                   total_samples=40)
    idxs = list(samp)
    assert len(idxs) == 40
    # draws before the schedule reaches difficulty 8 (step 4 on this
    # fixed_linear curve) must come from the easy pool
    early = idxs[:4]
    assert all(len(data[i]) <= 5 for i in early), \
        [len(data[i]) for i in early]
    # late draws include hard samples once difficulty reaches 20
    assert any(len(data[i]) >= 12 for i in idxs[20:])
