"""Variable batch size + LR scaling (ref variable_batch_size_and_lr.py)."""
import torch

from deepspeed_amd.runtime.variable_batch import (
    VariableBatchSizeLR, batch_by_seqlens,
    dataloader_for_variable_batch_size, get_variable_batch_dataloader_and_lr,
    scale_lr)


def test_packing_respects_token_budget():
    seqlens = [100, 200, 300, 50, 900, 400, 150, 820, 10, 30]
    mbs, sizes, maxlens = batch_by_seqlens(seqlens, max_tokens=1000)
    all_ids = [i for _, ids in mbs for i in ids]
    assert sorted(all_ids) == list(range(len(seqlens)))  # every sample once
    for (bid, ids), mx in zip(mbs, maxlens):
        total = sum(seqlens[i] for i in ids)
        assert total <= 1000, f"batch {bid} holds {total} tokens"
        assert mx == max(seqlens[i] for i in ids)
    assert sizes == [len(ids) for _, ids in mbs]


def test_packing_skips_oversized_and_caps_count():
    seqlens = [10, 2000, 20, 30, 40]
    mbs, sizes, _ = batch_by_seqlens(seqlens, max_tokens=100,
                                     max_batch_size=2)
    ids = [i for _, s in mbs for i in s]
    assert 1 not in ids  # the 2000-token sample is skipped
    assert max(sizes) <= 2


def test_packing_seqlen_order_groups_similar_lengths():
    seqlens = [512, 16, 500, 20, 480, 24]
    mbs, _, maxlens = batch_by_seqlens(seqlens, max_tokens=1024,
                                       sequence_picking_order="seqlen")
    # short ones packed together first (greedy fill: 16+20+24+480 <= 1024)
    assert {1, 3, 5} <= set(mbs[0][1])
    assert len(mbs[0][1]) > len(mbs[-1][1])


def test_scale_lr_rules():
    assert scale_lr(32, 64, 0.1, "linear") == 0.2
    assert abs(scale_lr(32, 64, 0.1, "sqrt") - 0.1 * 2 ** 0.5) < 1e-9
    assert scale_lr(32, 64, 0.1, None) == 0.1


def test_variable_lr_scheduler_scales_per_batch():
    p = torch.nn.Parameter(torch.zeros(3))
    opt = torch.optim.SGD([p], lr=0.1)
    sched = VariableBatchSizeLR(opt, base_batch_size=4,
                                batch_sizes=[4, 8, 2],
                                lr_scaling_method="linear")
    assert abs(opt.param_groups[0]["lr"] - 0.1) < 1e-9        # bs 4
    sched.step(1)
    assert abs(opt.param_groups[0]["lr"] - 0.2) < 1e-9        # bs 8
    sched.step(2)
    assert abs(opt.param_groups[0]["lr"] - 0.05) < 1e-9       # bs 2
    sd = sched.state_dict()
    sched2 = VariableBatchSizeLR(opt, 4, [4, 8, 2], "linear")
    sched2.load_state_dict(sd)
    assert sched2._batch == sched._batch


def test_dataloader_yields_packed_batches_with_padding():
    data = [torch.arange(n) for n in (5, 7, 3, 9, 2, 4)]
    seqlens = [len(d) for d in data]
    mbs, sizes, maxlens = batch_by_seqlens(seqlens, max_tokens=12)

    def pad(sample, to_len):
        return torch.nn.functional.pad(sample, (0, to_len - len(sample)))

    dl = dataloader_for_variable_batch_size(
        data, mbs, maxlens, sample_padding_fn=pad,
        collate_fn=lambda xs: torch.stack(xs))
    seen = 0
    for batch in dl:
        assert batch.dim() == 2
        seen += batch.shape[0]
    assert seen == sum(sizes)


def test_glue_end_to_end_lr_follows_batches():
    data = [torch.arange(n).float() for n in (5, 7, 3, 9, 2, 4, 6, 8)]
    seqlens = [len(d) for d in data]
    p = torch.nn.Parameter(torch.zeros(3))
    opt = torch.optim.SGD([p], lr=0.1)
    dl, sched = get_variable_batch_dataloader_and_lr(
        data, seqlens, max_tokens=16, optimizer=opt, base_batch_size=2,
        collate_fn=lambda xs: xs)
    lrs = []
    for _i, batch in enumerate(dl):
        lrs.append(opt.param_groups[0]["lr"])
        sched.step()
    assert len(set(lrs)) > 1, "LR never adapted to batch size"
