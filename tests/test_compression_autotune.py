"""Compression (QAT/pruning), autotuner, quantized reduction collectives."""
import torch

from tests.common import run_distributed


def test_fake_quant_straight_through():
    from deepspeed_amd.compression import fake_quantize
    x = torch.randn(100, requires_grad=True)
    y = fake_quantize(x, 8)
    assert (y - x).abs().max() < x.abs().max() / 100
    y.sum().backward()
    assert torch.allclose(x.grad, torch.ones_like(x))


def test_init_compression_and_clean():
    from deepspeed_amd.compression import (LinearLayerCompress,
                                           init_compression,
                                           redundancy_clean)
    m = torch.nn.Sequential(torch.nn.Linear(16, 16), torch.nn.ReLU(),
                            torch.nn.Linear(16, 8))
    cfg = {"weight_quantization": {"different_groups": {
        "wq": {"params": {"target_bits": 8}, "modules": [".*"]}}},
        "sparse_pruning": {"different_groups": {
            "sp": {"params": {"dense_ratio": 0.5}, "modules": ["0"]}}}}
    init_compression(m, cfg)
    assert isinstance(m[0], LinearLayerCompress)
    assert m[0].prune_mask is not None
    x = torch.randn(4, 16)
    out = m(x)
    assert out.shape == (4, 8)
    redundancy_clean(m)
    # ~half of m[0] weights pruned to zero
    frac_zero = (m[0].weight == 0).float().mean().item()
    assert 0.3 < frac_zero < 0.7


def test_autotuner_cpu():
    from deepspeed_amd.autotuning import Autotuner
    from tests.simple_model import SimpleModel

    def model_fn():
        torch.manual_seed(0)
        return SimpleModel(32)

    def data_fn(cfg):
        mb = cfg["train_micro_batch_size_per_gpu"]
        return (torch.randn(mb, 32, dtype=torch.bfloat16),
                torch.randn(mb, 32, dtype=torch.bfloat16))

    base = {"optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
            "bf16": {"enabled": True}}
    tuner = Autotuner(model_fn, data_fn, base, steps=2, warmup=1)
    best, results = tuner.tune(micro_batches=(2, 4), stages=(1, 2))
    assert best["train_micro_batch_size_per_gpu"] in (2, 4)
    assert len([r for r in results if "error" not in r]) >= 1


def _quant_reduce():
    import torch.distributed as td
    from deepspeed_amd.runtime.comm.coalesced_collectives import \
        (all_to_all_quant_reduce, reduce_scatter_coalesced)
    rank = td.get_rank()
    world = td.get_world_size()
    torch.manual_seed(17 + rank)
    g = torch.randn(4096, dtype=torch.float32)
    exact = reduce_scatter_coalesced([g.clone()])[0]
    quant = all_to_all_quant_reduce([g.clone()])[0]
    rel = (quant[:exact.numel()] - exact).abs().mean() / \
        exact.abs().mean()
    assert rel < 0.05, rel.item()
    return True


def test_quantized_grad_reduce_2rank():
    assert all(run_distributed(_quant_reduce, world_size=2))


def test_structured_pruning_and_layer_reduction():
    from deepspeed_amd.compression import (LinearLayerCompress,
                                           apply_layer_reduction,
                                           init_compression)
    torch.manual_seed(0)

    class Tiny(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.layers = torch.nn.ModuleList(
                [torch.nn.Linear(16, 16) for _ in range(6)])

        def forward(self, x):
            for l in self.layers:
                x = torch.tanh(l(x))
            return x

    m = Tiny()
    cfg = {"row_pruning": {"different_groups": {
               "rp": {"params": {"dense_ratio": 0.5},
                      "modules": ["layers.0"]}}},
           "head_pruning": {"different_groups": {
               "hp": {"params": {"dense_ratio": 0.5, "num_heads": 4},
                      "modules": ["layers.1"]}}},
           "activation_quantization": {"different_groups": {
               "aq": {"params": {"bits": 8}, "modules": ["layers.2"]}}}}
    init_compression(m, cfg)
    assert isinstance(m.layers[0], LinearLayerCompress)
    # row pruning: ~half the rows fully zero after clean
    w0 = m.layers[0]
    rows_zero = (w0.prune_mask.sum(1) == 0).float().mean().item()
    assert 0.3 < rows_zero < 0.7
    # head pruning: mask zeroes whole 4-row head blocks
    w1 = m.layers[1]
    head_mask = w1.prune_mask.reshape(4, 4, 16)
    per_head = head_mask.amax(dim=(1, 2))
    assert set(per_head.tolist()) <= {0.0, 1.0}
    assert 0 < per_head.sum() < 4
    out = m(torch.randn(2, 16))
    assert out.shape == (2, 16)
    # layer reduction: 6 -> 3 layers picking teacher indices
    apply_layer_reduction(m, {"module_name_prefix": "layers",
                              "keep_number": 3,
                              "teacher_layer": [0, 2, 4]})
    assert len(m.layers) == 3
    assert m(torch.randn(2, 16)).shape == (2, 16)


def test_compression_scheduler_offsets():
    """Techniques activate at their schedule_offset, not step 0, and
    deactivate at schedule_offset_end (ref compression/scheduler.py)."""
    import torch
    from deepspeed_amd.compression import (CompressionScheduler,
                                           LinearLayerCompress)
    model = torch.nn.Sequential(torch.nn.Linear(8, 8),
                                torch.nn.Linear(8, 8))
    cfg = {
        "weight_quantization": {
            "shared_parameters": {"enabled": True, "schedule_offset": 3},
            "different_groups": {"wq": {"params": {"target_bits": 8},
                                        "modules": [".*"]}}},
        "sparse_pruning": {
            "shared_parameters": {"enabled": True, "schedule_offset": 5,
                                  "schedule_offset_end": 7},
            "different_groups": {"sp": {"params": {"dense_ratio": 0.5},
                                        "modules": [".*"]}}},
    }
    sched = CompressionScheduler(model, cfg)
    assert not any(isinstance(m, LinearLayerCompress)
                   for m in model.modules())
    sched.step(3)  # -> step 3: weight quantization turns on
    wq_mods = [m for m in model.modules()
               if isinstance(m, LinearLayerCompress)]
    assert wq_mods and all(m.weight_quant_enabled for m in wq_mods)
    assert all(m.prune_mask is None for m in wq_mods)
    sched.step(2)  # -> step 5: sparse pruning turns on
    assert all(m.prune_mask is not None
               for m in model.modules()
               if isinstance(m, LinearLayerCompress))
    sched.step(2)  # -> step 7: sparse pruning end
    assert all(m.prune_mask is None
               for m in model.modules()
               if isinstance(m, LinearLayerCompress))
    sd = sched.state_dict()
    sched2 = CompressionScheduler(model, cfg)
    sched2.load_state_dict(sd)
    assert sched2.training_steps == sched.training_steps


def test_autotuner_full_cost_model_and_pruning(tmp_path):
    """Cost model fits sec = a + b*mb; infeasible configs are pruned by
    the ZeRO memory model; experiment records land on disk."""
    from deepspeed_amd.autotuning import (AutotunerFull, ModelBasedTuner,
                                          estimate_memory_per_gpu)
    t = ModelBasedTuner()
    t.record(1, 0.1 + 0.02 * 1)
    t.record(8, 0.1 + 0.02 * 8)
    a, b = t.fit()
    assert abs(a - 0.1) < 1e-6 and abs(b - 0.02) < 1e-6
    assert t.propose([1, 2, 4, 8]) == 4   # best predicted untried tput
    # memory model: stage 3 shards everything; stage 0 replicates
    m0 = estimate_memory_per_gpu(7e9, 0, world=8)
    m3 = estimate_memory_per_gpu(7e9, 3, world=8)
    assert m3 < m0 / 3

    import torch
    from tests.simple_model import SimpleModel, make_batches

    def data_fn(cfg):
        mb = cfg["train_micro_batch_size_per_gpu"]
        x = torch.randn(mb, 32)
        return (x, x)

    tuner = AutotunerFull(
        lambda: SimpleModel(32), data_fn,
        {"optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
         "train_micro_batch_size_per_gpu": 1},
        steps=2, warmup=1, results_dir=str(tmp_path),
        model_info={"num_params": 3000, "hidden_size": 32,
                    "num_layers": 3, "seq_len": 1})
    best, results = tuner.tune(micro_batches=(1, 4), stages=(1,),
                               max_experiments=3)
    assert best["zero_optimization"]["stage"] == 1
    assert any("samples_per_sec" in r for r in results)
    import os
    assert len(os.listdir(tmp_path)) >= 1


def test_moq_quantizer_schedule():
    """MoQ bit widths anneal 16->8 per layer period; eigenvalue-sensitive
    layers anneal slower; params really quantize."""
    import torch
    from deepspeed_amd.runtime.quantize import Quantizer
    q = Quantizer(layer_num=2, start_bits=16, target_bits=8,
                  quantize_period=5, q_verbose=False)
    p0 = torch.randn(32, 32)
    p1 = torch.randn(32, 32)
    orig0 = p0.clone()
    for step in range(20):
        q.quantize([[p0], [p1]])
    assert q.bits == [8, 8]
    assert not q.any_precision_switch()
    assert not torch.equal(p0, orig0)  # fake-quantized in place
    # distinct values collapse to <= 2^8 levels per row-ish group
    assert p0.unique().numel() < orig0.unique().numel()

    # eigenvalue modulation: layer 1 twice as sensitive -> longer period
    q2 = Quantizer(layer_num=2, start_bits=16, target_bits=8,
                   quantize_period=5, q_eigenvalue=True)
    a, b = torch.randn(8, 8), torch.randn(8, 8)
    for step in range(7):
        q2.quantize([[a], [b]], eigenvalue_enabled=True,
                    block_eigenvalue={0: 0.1, 1: 10.0})
    assert q2.bits[0] == 8 and q2.bits[1] == 16, q2.bits
