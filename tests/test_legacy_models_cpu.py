"""Forward/backward smoke + shape checks for the legacy model families
(BERT / T5 / biencoder / classification), mirroring the reference's coverage
of megatron/model/{bert_model,t5_model,biencoder_model,classification}.py."""

import pytest
import torch


def _cfg(**kw):
    from megatron_amd.config import TrainingConfig, set_config

    base = dict(
        num_layers=2, hidden_size=64, num_attention_heads=4,
        num_attention_heads_kv=4, seq_length=32, max_position_embeddings=64,
        hidden_dropout=0.0, attention_dropout=0.0,
        use_cpu_initialization=True, position_embedding_type="absolute",
        use_rms_norm=False, glu_activation=None, use_bias=True,
        use_flash_attn=False,
    )
    base.update(kw)
    cfg = TrainingConfig(**base)
    cfg.finalize()
    cfg.pad_vocab_size(128)
    set_config(cfg)
    return cfg


def test_bert_forward_backward(dist_single):
    from megatron_amd.models.bert_model import BertModel

    cfg = _cfg(bert_binary_head=True)
    model = BertModel(cfg, num_tokentypes=2, add_binary_head=True,
                      parallel_output=True)
    b, s = 2, 32
    tokens = torch.randint(0, 128, (b, s))
    padding_mask = torch.ones(b, s, dtype=torch.long)
    padding_mask[:, -4:] = 0
    types = torch.zeros(b, s, dtype=torch.long)
    types[:, s // 2:] = 1
    labels = torch.randint(0, 128, (b, s))

    lm_loss, sop_logits = model(tokens, padding_mask, tokentype_ids=types,
                                lm_labels=labels)
    assert lm_loss.shape == (b, s)
    assert sop_logits.shape == (b, 2)
    loss = lm_loss.mean() + sop_logits.sum() * 0
    loss.backward()
    assert torch.isfinite(loss)

    # no labels -> logits [b, s, v] (reference bert_model.py transposes)
    model.zero_grad()
    logits, _ = model(tokens, padding_mask, tokentype_ids=types)
    assert logits.shape == (b, s, 128)


def test_t5_forward_backward(dist_single):
    from megatron_amd.models.t5_model import T5Model

    cfg = _cfg()
    model = T5Model(cfg, parallel_output=True)
    b, se, sd = 2, 32, 16
    enc_tokens = torch.randint(0, 128, (b, se))
    dec_tokens = torch.randint(0, 128, (b, sd))
    enc_mask = torch.ones(b, se, dtype=torch.long)
    enc_mask[:, -3:] = 0
    dec_mask = torch.ones(b, sd, dtype=torch.long)
    enc_mask_2d = enc_mask.unsqueeze(1) * enc_mask.unsqueeze(2)
    dec_mask_2d = dec_mask.unsqueeze(1) * dec_mask.unsqueeze(2)
    enc_dec_mask = dec_mask.unsqueeze(2) * enc_mask.unsqueeze(1)
    labels = torch.randint(0, 128, (b, sd))

    loss = model(enc_tokens, dec_tokens, enc_mask_2d, dec_mask_2d,
                 enc_dec_mask, lm_labels=labels)
    assert loss.shape == (b, sd)
    loss.mean().backward()
    assert torch.isfinite(loss).all()

    # logits path + cached encoder hidden states (inference pattern)
    model.zero_grad()
    with torch.no_grad():
        logits = model(enc_tokens, dec_tokens, enc_mask_2d, dec_mask_2d,
                       enc_dec_mask)
        assert logits.shape == (b, sd, 128)


def test_biencoder_forward_backward(dist_single):
    from megatron_amd.models.biencoder_model import BiEncoderModel

    cfg = _cfg(bert_binary_head=False)
    model = BiEncoderModel(cfg, num_tokentypes=2, projection_dim=16)
    b, s = 2, 32
    q = torch.randint(0, 128, (b, s))
    c = torch.randint(0, 128, (b, s))
    qm = torch.ones(b, s, dtype=torch.long)
    cm = torch.ones(b, s, dtype=torch.long)
    q_emb = model.embed_query(q, qm)
    c_emb = model.embed_context(c, cm)
    assert q_emb.shape == (b, 16)
    assert c_emb.shape == (b, 16)
    scores = model(q, qm, c, cm)  # in-batch negatives retrieval scores
    assert scores.shape == (b, b)
    torch.nn.functional.cross_entropy(scores, torch.arange(b)).backward()


def test_classification_and_multichoice(dist_single):
    from megatron_amd.models.classification import (
        Classification, MultipleChoice,
    )

    cfg = _cfg(bert_binary_head=False)
    model = Classification(cfg, num_classes=3, num_tokentypes=2)
    b, s = 2, 32
    tokens = torch.randint(0, 128, (b, s))
    mask = torch.ones(b, s, dtype=torch.long)
    types = torch.zeros(b, s, dtype=torch.long)
    logits = model(tokens, mask, types)
    assert logits.shape == (b, 3)
    torch.nn.functional.cross_entropy(
        logits, torch.tensor([0, 2])
    ).backward()

    mc = MultipleChoice(cfg, num_tokentypes=2)
    n_choices = 4
    tokens = torch.randint(0, 128, (b, n_choices, s))
    mask = torch.ones(b, n_choices, s, dtype=torch.long)
    types = torch.zeros(b, n_choices, s, dtype=torch.long)
    logits = mc(tokens, mask, types)
    assert logits.shape == (b, n_choices)
