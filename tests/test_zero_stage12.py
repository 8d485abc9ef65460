"""ZeRO stage 1/2 numerics vs plain torch AdamW (DDP-equivalent)."""
import torch

import deepspeed_amd
from tests.common import run_distributed
from tests.simple_model import SimpleModel, make_batches, reference_adamw_training

HIDDEN = 32
LR = 1e-3


def _zero_train(stage, dtype_str, grad_accum=1, steps=4):
    import torch.distributed as dist
    rank = dist.get_rank()
    world = dist.get_world_size()
    torch.manual_seed(11)
    model = SimpleModel(HIDDEN)
    config = {
        "train_micro_batch_size_per_gpu": 4,
        "gradient_accumulation_steps": grad_accum,
        "optimizer": {"type": "AdamW", "params": {"lr": LR}},
        "zero_optimization": {"stage": stage, "reduce_bucket_size": 2000},
        dtype_str: {"enabled": True},
    }
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config=config)
    batches = make_batches(steps * grad_accum * world, 4, HIDDEN,
                           dtype=engine.config.dtype)
    i = 0
    for s in range(steps):
        for g in range(grad_accum):
            # rank r takes sample (step*ga + g)*world + r → matches the
            # fp32 reference which consumes all samples with full batch
            x, y = batches[i * world + rank]
            loss = engine(x, y)
            engine.backward(loss)
            engine.step()
            i += 1
    return [p.detach().float().cpu() for p in model.parameters()]


def _reference_params(steps, grad_accum, world):
    batches = make_batches(steps * grad_accum * world, 4, HIDDEN)
    # full-batch reference: concat the per-rank micro batches
    merged = []
    for i in range(steps * grad_accum):
        xs = torch.cat([batches[i * world + r][0] for r in range(world)])
        ys = torch.cat([batches[i * world + r][1] for r in range(world)])
        merged.append((xs, ys))
    model = reference_adamw_training(lambda: SimpleModel(HIDDEN), merged,
                                     lr=LR, grad_accum=grad_accum)
    return [p.detach().float() for p in model.parameters()]


def _check(stage, dtype_str, grad_accum=1, tol=3e-2):
    steps, world = 4, 2
    results = run_distributed(_zero_train, world_size=world,
                              args=(stage, dtype_str, grad_accum, steps))
    ref = _reference_params(steps, grad_accum, world)
    for r in range(world):
        got = results[r]
        assert len(got) == len(ref)
        for g, e in zip(got, ref):
            assert torch.allclose(g, e, atol=tol, rtol=tol), \
                f"stage{stage} {dtype_str} mismatch: max diff " \
                f"{(g - e).abs().max()}"
    # ranks agree exactly
    for g0, g1 in zip(results[0], results[1]):
        assert torch.equal(g0, g1)


def test_zero1_bf16():
    _check(1, "bf16")


def test_zero2_bf16():
    _check(2, "bf16")


def test_zero2_bf16_grad_accum():
    _check(2, "bf16", grad_accum=2)


def test_zero1_fp16():
    _check(1, "fp16")


def _gas_equivalence():
    """gas=2/mb=2 must equal gas=1/mb=4 on the same sample stream."""
    import deepspeed_amd
    from deepspeed_amd.comm import groups

    def train(mb, gas):
        groups.reset_groups()
        torch.manual_seed(11)
        model = SimpleModel(HIDDEN)
        config = {
            "train_micro_batch_size_per_gpu": mb,
            "gradient_accumulation_steps": gas,
            "optimizer": {"type": "AdamW", "params": {"lr": LR}},
            "zero_optimization": {"stage": 2, "reduce_bucket_size": 2000},
            "bf16": {"enabled": True},
        }
        engine, _, _, _ = deepspeed_amd.initialize(model=model,
                                                   config=config)
        data = make_batches(12, 4, HIDDEN, dtype=torch.bfloat16)
        flat = [(x.reshape(-1, HIDDEN), y.reshape(-1, HIDDEN))
                for x, y in data]
        xs = torch.cat([x for x, _ in flat])
        ys = torch.cat([y for _, y in flat])
        i = 0
        for s in range(3):
            for g in range(gas):
                x = xs[i:i + mb]
                y = ys[i:i + mb]
                i += mb
                loss = engine(x, y)
                engine.backward(loss)
                engine.step()
        out = [p.detach().float().clone() for p in model.parameters()]
        engine.destroy()
        return out

    a = train(4, 1)
    b = train(2, 2)
    for pa, pb in zip(a, b):
        assert torch.allclose(pa, pb, atol=3e-3), (pa - pb).abs().max()
    return True


def test_gas_equivalence_single_rank():
    results = run_distributed(_gas_equivalence, world_size=1)
    assert all(results)


def _qgz12_body(stage, steps=4):
    """Stage 1/2 qgZ: int8 bucket reduction tracks exact reduce-scatter."""
    import torch
    import torch.distributed as tdist
    import deepspeed_amd
    from deepspeed_amd.comm import groups
    from tests.simple_model import SimpleModel, make_batches
    rank = tdist.get_rank()
    world = tdist.get_world_size()

    def run(qgz):
        groups.reset_groups()
        torch.manual_seed(11)
        model = SimpleModel(32)
        config = {
            "train_micro_batch_size_per_gpu": 4,
            "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
            "zero_optimization": {"stage": stage,
                                  "zero_quantized_gradients": qgz},
            "bf16": {"enabled": True},
        }
        engine, _, _, _ = deepspeed_amd.initialize(model=model,
                                                   config=config)
        assert engine.optimizer.quantized_gradients == qgz
        batches = make_batches(steps * world, 4, 32, seed=5,
                               dtype=torch.bfloat16)
        for i in range(steps):
            x, y = batches[i * world + rank]
            loss = engine(x, y)
            engine.backward(loss)
            engine.step()
        return [b.master32.detach().cpu().clone()
                for b in engine.optimizer.buckets]

    s_q = run(True)
    s_f = run(False)
    for a, b in zip(s_q, s_f):
        err = (a - b).abs().max().item()
        assert err < 2e-2, f"stage{stage} qgZ diverged: {err}"
    return True


def test_zero1_quantized_gradients():
    from tests.common import run_distributed
    run_distributed(_qgz12_body, world_size=2, args=(1,))


def test_zero2_quantized_gradients():
    from tests.common import run_distributed
    run_distributed(_qgz12_body, world_size=2, args=(2,))


def _predivide_body(steps=3):
    """gradient_predivide_factor + fp32 communication_data_type produce
    the same training result as the default on-wire averaging."""
    import torch
    import torch.distributed as tdist
    import deepspeed_amd
    from deepspeed_amd.comm import groups
    from tests.simple_model import SimpleModel, make_batches
    rank = tdist.get_rank()
    world = tdist.get_world_size()

    def run(extra):
        groups.reset_groups()
        torch.manual_seed(11)
        model = SimpleModel(32)
        config = {
            "train_micro_batch_size_per_gpu": 4,
            "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
            "zero_optimization": {"stage": 2},
            "bf16": {"enabled": True},
        }
        config.update(extra)
        engine, _, _, _ = deepspeed_amd.initialize(model=model,
                                                   config=config)
        batches = make_batches(steps * world, 4, 32, seed=5,
                               dtype=torch.bfloat16)
        for i in range(steps):
            x, y = batches[i * world + rank]
            loss = engine(x, y)
            engine.backward(loss)
            engine.step()
        return [b.master32.detach().cpu().clone()
                for b in engine.optimizer.buckets]

    base = run({})
    pre = run({"gradient_predivide_factor": 2.0,
               "communication_data_type": "fp32"})
    for a, b in zip(base, pre):
        err = (a - b).abs().max().item()
        assert err < 5e-3, f"predivide path diverged: {err}"
    return True


def test_zero2_predivide_and_comm_dtype():
    from tests.common import run_distributed
    run_distributed(_predivide_body, world_size=2)
