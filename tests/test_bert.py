"""BERT encoder tests (reference contract: the fused-encoder training path
tests in tests/unit/ops/transformer/). MLM+NSP pretraining through the
engine with ZeRO-2, padding-mask semantics."""

import torch

from .common import run_local
from deepspeed_amd.models import BertForPreTraining, bert_tiny


def test_bert_forward_masking():
    torch.manual_seed(0)
    model = BertForPreTraining(bert_tiny())
    model.eval()
    ids = torch.randint(0, 512, (2, 16))
    mask = torch.ones(2, 16, dtype=torch.long)
    mask[1, 8:] = 0  # pad second half of sample 1
    with torch.no_grad():
        mlm_full, _ = model(ids, attention_mask=mask)
        ids2 = ids.clone()
        ids2[1, 8:] = 7  # change PADDED tokens only
        mlm_pad, _ = model(ids2, attention_mask=mask)
    # visible positions of sample 1 ignore padded-token changes
    torch.testing.assert_close(mlm_full[1, :8], mlm_pad[1, :8],
                               rtol=1e-4, atol=1e-5)
    assert not torch.allclose(mlm_full[1, 8:], mlm_pad[1, 8:], atol=1e-3)


def _bert_train_worker(rank, world):
    import deepspeed_amd
    torch.manual_seed(3)
    model = BertForPreTraining(bert_tiny())
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config={
        "train_micro_batch_size_per_gpu": 4,
        "zero_optimization": {"stage": 2, "overlap_comm": False},
        "optimizer": {"type": "AdamW", "params": {"lr": 5e-4}},
    })
    torch.manual_seed(9)
    ids = torch.randint(0, 512, (4, 32))
    labels = ids.clone()
    masked = torch.rand(4, 32) < 0.15
    labels[~masked] = -100
    nsp = torch.randint(0, 2, (4,))
    losses = []
    for _ in range(8):
        loss = engine(ids, labels=labels, next_sentence_label=nsp)
        engine.backward(loss)
        engine.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0]
    # tied MLM head: decoder weight IS the embedding weight
    assert engine.module.mlm_head.weight.data_ptr() == \
        engine.module.bert.embeddings.word_embeddings.weight.data_ptr()


def test_bert_pretraining_engine():
    run_local(_bert_train_worker)
