"""BERT family on the native op set (BASELINE BERT rows)."""
import pytest
import torch

from tests.common import run_distributed


def _bert_train_body(steps=4):
    import torch.distributed as tdist
    import deepspeed_amd
    from deepspeed_amd.comm import groups
    from deepspeed_amd.models.bert import BERT_CONFIGS, BertForPreTraining
    groups.reset_groups()
    rank = tdist.get_rank()
    torch.manual_seed(0)
    cfg = BERT_CONFIGS["bert-tiny"]
    model = BertForPreTraining(cfg)
    config = {
        "train_micro_batch_size_per_gpu": 2,
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
        "zero_optimization": {"stage": 2},
        "bf16": {"enabled": True},
    }
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config=config)
    g = torch.Generator().manual_seed(5 + rank)
    losses = []
    for _ in range(steps):
        ids = torch.randint(0, cfg.vocab_size, (2, 64), generator=g)
        mask = torch.ones(2, 64, dtype=torch.long)
        mask[:, 50:] = 0  # padded tail
        labels = ids.clone()
        labels[mask == 0] = -100
        loss = engine(ids, attention_mask=mask, labels=labels)
        engine.backward(loss)
        engine.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0], losses
    return losses


def test_bert_mlm_zero2_world2():
    run_distributed(_bert_train_body, world_size=2)


def test_bert_padding_mask_matches_unpadded():
    """Padded positions must not affect unpadded tokens' outputs."""
    from deepspeed_amd.models.bert import BERT_CONFIGS, BertModel
    torch.manual_seed(0)
    cfg = BERT_CONFIGS["bert-tiny"]
    model = BertModel(cfg).eval()
    ids = torch.randint(0, cfg.vocab_size, (1, 32))
    with torch.no_grad():
        full = model(ids)                       # no mask, S=32
        padded_ids = torch.cat([ids, torch.zeros(1, 16,
                                                 dtype=torch.long)], 1)
        mask = torch.cat([torch.ones(1, 32, dtype=torch.long),
                          torch.zeros(1, 16, dtype=torch.long)], 1)
        masked = model(padded_ids, attention_mask=mask)
    err = (masked[:, :32] - full).abs().max().item()
    assert err < 1e-4, f"padding leaked into real tokens: {err}"


def test_bert_qa_head():
    from deepspeed_amd.models.bert import (BERT_CONFIGS,
                                           BertForQuestionAnswering)
    torch.manual_seed(0)
    cfg = BERT_CONFIGS["bert-tiny"]
    model = BertForQuestionAnswering(cfg)
    ids = torch.randint(0, cfg.vocab_size, (2, 48))
    loss = model(ids, start_positions=torch.tensor([3, 7]),
                 end_positions=torch.tensor([5, 9]))
    loss.backward()
    assert torch.isfinite(loss)


@pytest.mark.gpu
def test_bert_trains_gpu():
    """bert-tiny... at head_dim 64 on GPU so the D=64 + kv-padding-mask
    flash kernel path carries a real model end-to-end."""
    import os
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    os.environ.setdefault("LOCAL_RANK", "0")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29519")
    import deepspeed_amd
    from deepspeed_amd.models.bert import BertConfig, BertForPreTraining
    torch.manual_seed(0)
    cfg = BertConfig(vocab_size=2048, hidden_size=256,
                     num_hidden_layers=2, num_attention_heads=4,
                     intermediate_size=512,
                     max_position_embeddings=256)  # head_dim 64
    with torch.device("cuda:0"):
        model = BertForPreTraining(cfg)
    config = {
        "train_micro_batch_size_per_gpu": 2,
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
        "zero_optimization": {"stage": 2},
        "bf16": {"enabled": True},
    }
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config=config)
    ids = torch.randint(0, cfg.vocab_size, (2, 128), device="cuda:0")
    mask = torch.ones(2, 128, dtype=torch.long, device="cuda:0")
    mask[:, 100:] = 0
    labels = ids.clone()
    labels[mask == 0] = -100
    losses = []
    for _ in range(6):
        loss = engine(ids, attention_mask=mask, labels=labels)
        engine.backward(loss)
        engine.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0] * 0.9, losses
    engine.destroy()
