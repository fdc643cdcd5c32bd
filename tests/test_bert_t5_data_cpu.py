"""BERT masked-LM and T5 span-corruption datasets + the pretrain_bert /
pretrain_t5 entry forward paths, end to end on CPU (reference
megatron/data/bert_dataset.py, t5_dataset.py, pretrain_bert.py,
pretrain_t5.py)."""

import os
import sys

import numpy as np
import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

VOCAB = 128
EXTRA_IDS = 8


@pytest.fixture()
def indexed_docs(tmp_path):
    from megatron_amd.data import indexed_dataset

    prefix = str(tmp_path / "docs")
    builder = indexed_dataset.make_builder(prefix + ".bin", dtype=np.int32)
    rng = np.random.RandomState(0)
    for _ in range(8):
        # body tokens avoid special ids 0-5 and the sentinel range
        doc = rng.randint(6, VOCAB - EXTRA_IDS,
                          size=rng.randint(20, 60)).astype(np.int32)
        builder.add_item(doc)
        builder.end_document()
    builder.finalize(prefix + ".idx")
    return indexed_dataset.make_dataset(prefix, "infer")


@pytest.fixture()
def fake_tokenizer():
    from megatron_amd import global_state
    from megatron_amd.tokenizer.tokenizers import FakeTokenizer

    tok = FakeTokenizer(VOCAB, vocab_extra_ids=EXTRA_IDS)
    global_state.set_tokenizer(tok)
    return tok


def test_bert_dataset_masking(indexed_docs, fake_tokenizer):
    from megatron_amd.data.bert_dataset import BertDataset

    ds = BertDataset("train", indexed_docs, np.arange(8), 16,
                     max_seq_length=64, masked_lm_prob=0.15,
                     short_seq_prob=0.1, seed=3, binary_head=True)
    tok = fake_tokenizer
    saw_random = saw_next = False
    for i in range(16):
        s = ds[i]
        assert s["text"].shape == (64,)
        assert s["text"][0] == tok.cls
        n_masked = int(s["loss_mask"].sum())
        assert n_masked >= 1
        # every predicted position has its original token as the label
        for j in np.nonzero(s["loss_mask"])[0]:
            assert 0 <= s["labels"][j] < VOCAB
        # unmasked positions carry -1 labels
        assert (s["labels"][s["loss_mask"] == 0] == -1).all()
        # padding consistency
        real = int(s["padding_mask"].sum())
        assert (s["text"][real:] == 0).all()
        saw_random |= s["is_random"] == 1
        saw_next |= s["is_random"] == 0
    assert saw_random and saw_next  # NSP produces both classes


def test_t5_dataset_span_corruption(indexed_docs, fake_tokenizer):
    from megatron_amd.data.t5_dataset import T5Dataset

    ds = T5Dataset("train", indexed_docs, np.arange(8), 16,
                   max_seq_length=64, max_seq_length_dec=32,
                   masked_lm_prob=0.15, seed=5)
    tok = fake_tokenizer
    sentinels = set(tok.additional_special_tokens_ids)
    for i in range(16):
        s = ds[i]
        assert s["text_enc"].shape == (64,)
        assert s["text_dec"].shape == (32,)
        # decoder starts with BOS, labels are decoder input shifted left
        assert s["text_dec"][0] == tok.bos_token_id
        n = int(s["loss_mask"].sum())
        assert n >= 2
        assert np.array_equal(s["labels"][: n - 1], s["text_dec"][1:n])
        # encoder contains sentinel tokens where spans were removed, and
        # the same sentinels appear in the decoder sequence in order
        enc_sent = [t for t in s["text_enc"] if t in sentinels]
        dec_sent = [t for t in s["text_dec"] if t in sentinels]
        assert len(enc_sent) >= 1
        assert enc_sent == dec_sent


def _loader_iter(ds, batch_size):
    dl = torch.utils.data.DataLoader(ds, batch_size=batch_size, shuffle=False)
    return iter(dl)


def test_pretrain_bert_forward_step(indexed_docs, fake_tokenizer, dist_single):
    import pretrain_bert
    from megatron_amd.config import TrainingConfig, set_config
    from megatron_amd.data.bert_dataset import BertDataset

    cfg = TrainingConfig(
        model_name="bert", num_layers=2, hidden_size=64,
        num_attention_heads=4, num_attention_heads_kv=4, seq_length=64,
        max_position_embeddings=64, micro_batch_size=2, global_batch_size=2,
        hidden_dropout=0.0, attention_dropout=0.0,
        use_cpu_initialization=True, position_embedding_type="absolute",
        use_rms_norm=False, glu_activation=None, use_bias=True,
        use_flash_attn=False, bert_binary_head=True,
    )
    cfg.finalize()
    cfg.pad_vocab_size(VOCAB)
    set_config(cfg)
    from megatron_amd import global_state
    global_state.init_timers()

    model = pretrain_bert.model_provider()
    ds = BertDataset("train", indexed_docs, np.arange(8), 4, 64, 0.15, 0.1, 3)
    out, loss_closure = pretrain_bert.forward_step(_loader_iter(ds, 2), model)
    loss, stats = loss_closure(out)
    assert torch.isfinite(loss)
    assert "lm loss" in stats and "sop loss" in stats
    loss.backward()


def test_pretrain_t5_forward_step(indexed_docs, fake_tokenizer, dist_single):
    import pretrain_t5
    from megatron_amd.config import TrainingConfig, set_config
    from megatron_amd.data.t5_dataset import T5Dataset

    cfg = TrainingConfig(
        model_name="t5", num_layers=2, hidden_size=64,
        num_attention_heads=4, num_attention_heads_kv=4, seq_length=64,
        max_position_embeddings=64, micro_batch_size=2, global_batch_size=2,
        decoder_seq_length=32, hidden_dropout=0.0, attention_dropout=0.0,
        use_cpu_initialization=True, position_embedding_type="absolute",
        use_rms_norm=False, glu_activation=None, use_bias=True,
        use_flash_attn=False,
    )
    cfg.finalize()
    cfg.pad_vocab_size(VOCAB)
    set_config(cfg)
    from megatron_amd import global_state
    global_state.init_timers()

    model = pretrain_t5.model_provider()
    ds = T5Dataset("train", indexed_docs, np.arange(8), 4, 64, 32, 0.15, 5)
    out, loss_closure = pretrain_t5.forward_step(_loader_iter(ds, 2), model)
    loss, stats = loss_closure(out)
    assert torch.isfinite(loss)
    loss.backward()


def test_pretrain_ict_forward_step(indexed_docs, fake_tokenizer, dist_single,
                                   tmp_path, monkeypatch):
    """ICT retriever pretraining: dataset provider + biencoder forward/loss
    (pretrain_ict.py end to end minus the driver loop)."""
    import numpy as np

    import pretrain_ict
    from megatron_amd import global_state
    from megatron_amd.config import TrainingConfig, set_config

    # re-save the docs as files pretrain_ict's provider can open by prefix
    prefix = indexed_docs  # fixture returns the dataset; rebuild prefix
    from megatron_amd.data import indexed_dataset as idx

    p = str(tmp_path / "ictdocs")
    builder = idx.make_builder(p + ".bin", dtype=np.int32)
    rng = np.random.RandomState(1)
    for _ in range(8):
        builder.add_item(rng.randint(6, 100, size=40).astype(np.int32))
        builder.end_document()
    builder.finalize(p + ".idx")

    cfg = TrainingConfig(
        model_name="bert", num_layers=2, hidden_size=64,
        num_attention_heads=4, num_attention_heads_kv=4, seq_length=32,
        max_position_embeddings=64, micro_batch_size=4, global_batch_size=4,
        hidden_dropout=0.0, attention_dropout=0.0,
        use_cpu_initialization=True, position_embedding_type="absolute",
        use_rms_norm=False, glu_activation=None, use_bias=True,
        use_flash_attn=False, bert_binary_head=False,
        data_path=[p], split="8,1,1",
    )
    cfg.finalize()
    cfg.pad_vocab_size(VOCAB)
    set_config(cfg)
    global_state.init_timers()

    train, valid, test = pretrain_ict.train_valid_test_datasets_provider(
        [8, 2, 2]
    )
    assert len(train) == 8
    batch = torch.utils.data.default_collate([train[i] for i in range(4)])
    model = pretrain_ict.model_provider()
    out, loss_fn = pretrain_ict.forward_step(iter([batch]), model)
    assert out.shape == (4, 4)
    loss, stats = loss_fn(out)
    assert torch.isfinite(loss)
    loss.backward()
    assert "retrieval loss" in stats


def test_pretrain_bert_full_driver(tmp_path, dist_single, fake_tokenizer):
    """pretrain() end to end for BERT: the entry's dataset provider over a
    real indexed corpus, 2 train iters, eval, checkpoint save."""
    import numpy as np

    import pretrain_bert
    from megatron_amd import global_state
    from megatron_amd.config import TrainingConfig, set_config
    from megatron_amd.models import ModelType
    from megatron_amd.training import pretrain

    p = str(tmp_path / "bertdocs")
    builder = __import__(
        "megatron_amd.data.indexed_dataset", fromlist=["make_builder"]
    ).make_builder(p + ".bin", dtype=np.int32)
    rng = np.random.RandomState(2)
    for _ in range(12):
        builder.add_item(rng.randint(6, 120, size=40).astype(np.int32))
        builder.end_document()
    builder.finalize(p + ".idx")

    cfg = TrainingConfig(
        model_name="bert", num_layers=2, hidden_size=64,
        num_attention_heads=4, num_attention_heads_kv=4, seq_length=64,
        max_position_embeddings=64, micro_batch_size=2, global_batch_size=2,
        train_iters=2, lr=1e-3, min_lr=1e-4, lr_decay_style="constant",
        lr_warmup_iters=0, eval_interval=10, eval_iters=1, log_interval=1,
        hidden_dropout=0.0, attention_dropout=0.0,
        use_cpu_initialization=True, position_embedding_type="absolute",
        use_rms_norm=False, glu_activation=None, use_bias=True,
        use_flash_attn=False, bert_binary_head=True, clip_grad=1.0,
        data_path=[p], split="10,1,1", save=str(tmp_path / "ckpt"),
        save_interval=100, world_size=1, rank=0,
    )
    cfg.finalize()
    cfg.pad_vocab_size(128)
    set_config(cfg)

    pretrain(
        pretrain_bert.train_valid_test_datasets_provider,
        pretrain_bert.model_provider,
        ModelType.encoder_or_decoder,
        pretrain_bert.forward_step,
        cfg=cfg,
    )
    import os as _os

    assert _os.path.isfile(_os.path.join(
        str(tmp_path / "ckpt"), "latest_checkpointed_iteration.txt"
    ))


def test_pretrain_t5_full_driver(tmp_path, dist_single, fake_tokenizer):
    """pretrain() end to end for T5 (ModelType.encoder_and_decoder)."""
    import numpy as np

    import pretrain_t5
    from megatron_amd.config import TrainingConfig, set_config
    from megatron_amd.models import ModelType
    from megatron_amd.training import pretrain

    p = str(tmp_path / "t5docs")
    from megatron_amd.data import indexed_dataset as idx

    builder = idx.make_builder(p + ".bin", dtype=np.int32)
    rng = np.random.RandomState(3)
    for _ in range(12):
        builder.add_item(
            rng.randint(6, 120 - EXTRA_IDS, size=40).astype(np.int32)
        )
        builder.end_document()
    builder.finalize(p + ".idx")

    cfg = TrainingConfig(
        model_name="t5", num_layers=2, hidden_size=64,
        num_attention_heads=4, num_attention_heads_kv=4, seq_length=64,
        encoder_seq_length=64, decoder_seq_length=32,
        max_position_embeddings=64, micro_batch_size=2, global_batch_size=2,
        train_iters=2, lr=1e-3, min_lr=1e-4, lr_decay_style="constant",
        lr_warmup_iters=0, eval_interval=10, eval_iters=1, log_interval=1,
        hidden_dropout=0.0, attention_dropout=0.0,
        use_cpu_initialization=True, position_embedding_type="absolute",
        use_rms_norm=False, glu_activation=None, use_bias=True,
        use_flash_attn=False, clip_grad=1.0,
        data_path=[p], split="10,1,1", world_size=1, rank=0,
    )
    cfg.finalize()
    cfg.pad_vocab_size(128)
    set_config(cfg)

    pretrain(
        pretrain_t5.train_valid_test_datasets_provider,
        pretrain_t5.model_provider,
        ModelType.encoder_and_decoder,
        pretrain_t5.forward_step,
        cfg=cfg,
    )


def test_dataloader_workers_fork(tmp_path, dist_single, fake_tokenizer):
    """MMapIndexedDataset survives DataLoader worker forks (pickling via
    __getstate__/path re-open)."""
    import numpy as np

    from megatron_amd.config import TrainingConfig, set_config
    from megatron_amd.data import indexed_dataset as idx
    from megatron_amd.data.gpt_dataset import build_train_valid_test_datasets

    cfg = TrainingConfig(seq_length=8)
    cfg.finalize()
    set_config(cfg)
    p = str(tmp_path / "wdocs")
    b = idx.make_builder(p + ".bin", dtype=np.int32)
    for i in range(20):
        b.add_item(np.full(30, 5 + i % 3, dtype=np.int32))
        b.end_document()
    b.finalize(p + ".idx")

    train, _, _ = build_train_valid_test_datasets(
        data_prefix=[p], data_impl="mmap", splits_string="100,0,0",
        train_valid_test_num_samples=[32, 0, 0], seq_length=8, seed=5,
        skip_warmup=True,
    )
    loader = torch.utils.data.DataLoader(train, batch_size=4, num_workers=2)
    n = 0
    for batch in loader:
        assert batch["text"].shape[1] == 9  # last batch may be partial
        n += 1
    assert n >= 8
