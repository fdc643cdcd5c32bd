"""CPU tests for the tools: indexed dataset build/read, preprocessing,
checkpoint resharding roundtrip, GPT dataset index mappings."""

import json
import os
import subprocess
import sys

import numpy as np
import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_indexed_dataset_roundtrip(tmp_path):
    from megatron_amd.data import indexed_dataset

    prefix = str(tmp_path / "ds")
    builder = indexed_dataset.make_builder(prefix + ".bin", dtype=np.int32)
    docs = [np.array([1, 2, 3, 4], dtype=np.int32),
            np.array([9, 8], dtype=np.int32),
            np.array([5] * 100, dtype=np.int32)]
    for d in docs:
        builder.add_item(d)
        builder.end_document()
    builder.finalize(prefix + ".idx")

    ds = indexed_dataset.make_dataset(prefix, "infer")
    assert len(ds) == 3
    for i, d in enumerate(docs):
        assert np.array_equal(ds[i], d)
    assert np.array_equal(ds.get(2, offset=10, length=5),
                          np.array([5] * 5, dtype=np.int32))
    assert list(ds.doc_idx) == [0, 1, 2, 3]


def test_merge_datasets(tmp_path):
    from megatron_amd.data import indexed_dataset

    for name in ("a", "b"):
        builder = indexed_dataset.make_builder(
            str(tmp_path / f"{name}.bin"), dtype=np.int32
        )
        builder.add_item(np.array([1, 2], dtype=np.int32))
        builder.end_document()
        builder.finalize(str(tmp_path / f"{name}.idx"))
    out = str(tmp_path / "merged")
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "merge_datasets.py"),
         "--input", str(tmp_path), "--output_prefix", out],
        capture_output=True, text=True,
    )
    assert r.returncode == 0, r.stderr
    ds = indexed_dataset.make_dataset(out, "infer")
    assert len(ds) == 2


def test_gpt_dataset_mapping(tmp_path, dist_single):
    from megatron_amd.data import indexed_dataset
    from megatron_amd.data.gpt_dataset import GPTDataset

    prefix = str(tmp_path / "gpt")
    builder = indexed_dataset.make_builder(prefix + ".bin", dtype=np.int32)
    rng = np.random.RandomState(0)
    all_tokens = []
    for _ in range(10):
        doc = rng.randint(0, 1000, size=rng.randint(5, 50)).astype(np.int32)
        builder.add_item(doc)
        builder.end_document()
        all_tokens.append(doc)
    builder.finalize(prefix + ".idx")

    ds = indexed_dataset.make_dataset(prefix, "infer")
    documents = np.arange(10, dtype=np.int32)
    gpt = GPTDataset("train", prefix, documents, ds, num_samples=20,
                     seq_length=16, seed=5)
    assert len(gpt) >= 20
    for i in range(5):
        sample = gpt[i]["text"]
        assert sample.shape == (17,)  # seq_length + 1
        assert sample.dtype == np.int64


def test_sample_idx_matches_naive():
    from megatron_amd.data.gpt_dataset import _build_sample_idx

    rng = np.random.RandomState(1)
    sizes = rng.randint(3, 30, size=50).astype(np.int32)
    doc_idx = rng.permutation(np.arange(50)).astype(np.int32)
    seq = 16
    tot = sizes[doc_idx].sum()
    num_samples = (tot - 1) // seq
    sample_idx = _build_sample_idx(sizes, doc_idx, seq, num_samples)

    # naive reference walk
    flat_doc = []
    for d in doc_idx:
        flat_doc.extend([d] * sizes[d])
    for i in range(0, num_samples + 1):
        b = i * seq
        doc_pos, offset = sample_idx[i]
        # position b is `offset` tokens into document at doc_idx[doc_pos]
        start = sum(sizes[doc_idx[j]] for j in range(doc_pos))
        assert start + offset == b, (i, doc_pos, offset)


def test_preprocess_and_instruction_pipeline(tmp_path, dist_single):
    # build a tiny instruct jsonl, preprocess with the Fake tokenizer path?
    # preprocess needs a real tokenizer; use GPT2BPE with a tiny vocab
    vocab = {chr(ord('a') + i): i for i in range(26)}
    vocab.update({"<|endoftext|>": 26, "Ġ": 27})
    # minimal byte-level vocab won't round-trip real text; instead test the
    # instruction dataset + collator directly with synthetic indexed data
    from megatron_amd.data import indexed_dataset
    from megatron_amd.data.instruction_dataset import (
        ROLE_ASSISTANT, ROLE_PROMPTER, InstructionDataset,
    )
    from megatron_amd.config import TrainingConfig, set_config

    text_b = indexed_dataset.make_builder(
        str(tmp_path / "x-text.bin"), dtype=np.int32
    )
    role_b = indexed_dataset.make_builder(
        str(tmp_path / "x-role.bin"), dtype=np.int8
    )
    rng = np.random.RandomState(0)
    for _ in range(6):
        n = rng.randint(10, 30)
        toks = rng.randint(0, 100, n).astype(np.int32)
        roles = np.array(
            [ROLE_PROMPTER] * (n // 2) + [ROLE_ASSISTANT] * (n - n // 2),
            dtype=np.int8,
        )
        text_b.add_item(toks)
        text_b.end_document()
        role_b.add_item(roles)
        role_b.end_document()
    text_b.finalize(str(tmp_path / "x-text.idx"))
    role_b.finalize(str(tmp_path / "x-role.idx"))

    cfg = TrainingConfig(seq_length=32)
    cfg.finalize()
    set_config(cfg)

    text_ds = indexed_dataset.make_dataset(str(tmp_path / "x-text"), "infer")
    role_ds = indexed_dataset.make_dataset(str(tmp_path / "x-role"), "infer")
    ds = InstructionDataset("train", str(tmp_path / "x"),
                            np.arange(6, dtype=np.int32), text_ds, role_ds,
                            num_samples=6, seq_length=32, seed=0)
    batch = ds.collate_fn([ds[0], ds[1]])
    assert batch["text"].shape == (2, 32)
    assert batch["assistant_mask"].shape == (2, 32)
    # loss mask covers only assistant tokens
    assert batch["assistant_mask"].sum() > 0
    assert (batch["assistant_mask"] <= batch["pad_mask"]).all()


def test_checkpoint_resharding_roundtrip(tmp_path, dist_single):
    """release tp1 -> tp2 -> tp1 preserves every tensor (analog of the
    reference's incremental shard/unshard pipeline test)."""
    from megatron_amd.config import TrainingConfig, set_config
    from megatron_amd.models import LlamaModel

    cfg = TrainingConfig(
        num_layers=2, hidden_size=64, num_attention_heads=4,
        num_attention_heads_kv=2, seq_length=16, max_position_embeddings=32,
        use_cpu_initialization=True, hidden_dropout=0.0,
        attention_dropout=0.0,
    )
    cfg.finalize()
    cfg.pad_vocab_size(128)
    set_config(cfg)
    m = LlamaModel(cfg)
    sd = {k: v.detach().clone()
          for k, v in m.language_model.state_dict().items()}

    import argparse as ap

    margs = ap.Namespace(
        num_layers=2, hidden_size=64, num_attention_heads=4,
        num_attention_heads_kv=2, ffn_hidden_size=cfg.ffn_hidden_size,
        padded_vocab_size=128, tensor_model_parallel_size=1,
        pipeline_model_parallel_size=1, glu_activation="swiglu",
    )
    src = tmp_path / "src"
    (src / "release" / "mp_rank_00").mkdir(parents=True)
    torch.save(
        {"args": margs, "checkpoint_version": 3.0, "iteration": 0,
         "model": {"language_model": sd}},
        src / "release" / "mp_rank_00" / "model_optim_rng.pt",
    )
    (src / "latest_checkpointed_iteration.txt").write_text("release")

    mid = tmp_path / "tp2"
    out = tmp_path / "back"
    script = os.path.join(REPO, "tools", "checkpoint_util.py")
    r1 = subprocess.run(
        [sys.executable, script, "--model_type", "llama2",
         "--load_dir", str(src), "--save_dir", str(mid),
         "--target_tensor_parallel_size", "2"],
        capture_output=True, text=True,
    )
    assert r1.returncode == 0, r1.stderr
    r2 = subprocess.run(
        [sys.executable, script, "--model_type", "llama2",
         "--load_dir", str(mid), "--save_dir", str(out),
         "--target_tensor_parallel_size", "1"],
        capture_output=True, text=True,
    )
    assert r2.returncode == 0, r2.stderr

    back = torch.load(out / "release" / "mp_rank_00" / "model_optim_rng.pt",
                      map_location="cpu", weights_only=False)
    back_sd = back["model"]["language_model"]
    assert set(back_sd.keys()) == set(sd.keys())
    for key in sd:
        assert torch.equal(back_sd[key], sd[key]), key


def test_bench_decode_cpu_smoke(dist_single):
    """tools/bench_decode.py runs end to end on CPU with the tiny model
    (eager path; the hipGraph path is GPU-only)."""
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "bench_decode.py"),
         "--model", "llama2-tiny", "--tokens", "4", "--prompt", "4"],
        capture_output=True, text=True, timeout=300,
        env={**os.environ, "MASTER_PORT": "29681"},
    )
    assert r.returncode == 0, r.stderr[-800:]
    assert "speedup" in r.stdout


def test_blended_datasets(tmp_path, dist_single):
    """Weighted blending of two tokenized corpora: sample mix tracks the
    weights and every sample comes from the right source (native
    build_blending_indices path)."""
    import numpy as np

    from megatron_amd.config import TrainingConfig, set_config
    from megatron_amd.data import indexed_dataset
    from megatron_amd.data.gpt_dataset import build_train_valid_test_datasets

    cfg = TrainingConfig(seq_length=8)
    cfg.finalize()
    set_config(cfg)

    prefixes = []
    for tag, token in (("a", 7), ("b", 9)):
        prefix = str(tmp_path / f"ds_{tag}")
        b = indexed_dataset.make_builder(prefix + ".bin", dtype=np.int32)
        for _ in range(40):
            b.add_item(np.full(50, token, dtype=np.int32))
            b.end_document()
        b.finalize(prefix + ".idx")
        prefixes.append(prefix)

    train, valid, test = build_train_valid_test_datasets(
        data_prefix=["0.75", prefixes[0], "0.25", prefixes[1]],
        data_impl="mmap", splits_string="100,0,0",
        train_valid_test_num_samples=[400, 0, 0],
        seq_length=8, seed=123, skip_warmup=True,
    )
    counts = {7: 0, 9: 0}
    n = 400
    for i in range(n):
        tok = int(train[i]["text"][0])
        counts[tok] += 1
    frac_a = counts[7] / n
    assert 0.70 < frac_a < 0.80, counts  # tracks the 0.75 weight
    assert counts[9] > 0


def test_per_split_data_paths(tmp_path, dist_single):
    """--train_data_path / --valid_data_path build each split from its own
    corpus (split string ignored), matching the reference's behavior."""
    import numpy as np

    from megatron_amd.config import TrainingConfig, set_config
    from megatron_amd.data import indexed_dataset
    from megatron_amd.data.gpt_dataset import build_train_valid_test_datasets

    cfg = TrainingConfig(seq_length=8)
    cfg.finalize()
    set_config(cfg)

    def make(tag, token):
        prefix = str(tmp_path / f"sp_{tag}")
        b = indexed_dataset.make_builder(prefix + ".bin", dtype=np.int32)
        for _ in range(10):
            b.add_item(np.full(40, token, dtype=np.int32))
            b.end_document()
        b.finalize(prefix + ".idx")
        return prefix

    ptrain, pvalid = make("train", 3), make("valid", 5)
    train, valid, test = build_train_valid_test_datasets(
        data_prefix=None, data_impl="mmap", splits_string="969,30,1",
        train_valid_test_num_samples=[50, 20, 0], seq_length=8, seed=1,
        skip_warmup=True,
        train_data_prefix=[ptrain], valid_data_prefix=[pvalid],
    )
    assert test is None
    assert int(train[0]["text"][0]) == 3
    assert int(valid[0]["text"][0]) == 5
    assert len(train) >= 50 and len(valid) >= 20


def test_verify_correctness_cli(dist_single):
    """verify_correctness.py runs offline (--random_init) on a tiny llama
    and reports per-iteration logit errors."""
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "verify_correctness.py"),
         "--model_name", "llama2", "--random_init",
         "--num_layers", "2", "--hidden_size", "64",
         "--num_attention_heads", "4", "--num_attention_heads_kv", "2",
         "--seq_length", "32", "--max_position_embeddings", "64",
         "--micro_batch_size", "1", "--iters", "2",
         "--use_cpu_initialization", "--make_vocab_size_divisible_by", "16"],
        capture_output=True, text=True, timeout=600,
        env={**os.environ, "MASTER_ADDR": "127.0.0.1",
             "MASTER_PORT": "29691", "RANK": "0", "WORLD_SIZE": "1"},
    )
    assert r.returncode == 0, (r.stdout[-500:], r.stderr[-800:])
    assert "max abs error" in r.stdout or "error" in r.stdout.lower()


def test_pretraining_sampler_resume(dist_single):
    """MegatronPretrainingSampler fast-forwards by consumed_samples, so a
    resumed run continues at the right batch (reference data_samplers.py)."""
    from megatron_amd.data.samplers import MegatronPretrainingSampler

    s = MegatronPretrainingSampler(
        total_samples=20, consumed_samples=0, micro_batch_size=2,
        data_parallel_rank=0, data_parallel_size=1,
    )
    order = [i for batch in iter(s) for i in [batch]]
    flat = [i for b in order for i in b]
    assert flat[:4] == [0, 1, 2, 3]

    s2 = MegatronPretrainingSampler(
        total_samples=20, consumed_samples=6, micro_batch_size=2,
        data_parallel_rank=0, data_parallel_size=1,
    )
    flat2 = [i for b in iter(s2) for i in b]
    assert flat2[0] == 6  # resumes exactly after the consumed samples
    assert len(flat2) == 14

    # dp sharding: rank 1 of 2 sees the second half of each global batch
    s3 = MegatronPretrainingSampler(
        total_samples=8, consumed_samples=0, micro_batch_size=2,
        data_parallel_rank=1, data_parallel_size=2,
    )
    flat3 = [i for b in iter(s3) for i in b]
    assert flat3 == [2, 3, 6, 7]


def test_finetune_instruction_loss_mask(dist_single):
    """finetune.get_batch (instruction mode): loss covers assistant tokens
    at weight 1 and the rest of the conversation at scalar_loss_mask."""
    import finetune
    from megatron_amd.config import TrainingConfig, set_config

    cfg = TrainingConfig(model_type="instruction", scalar_loss_mask=0.25,
                         seq_length=7)
    cfg.finalize()
    set_config(cfg)

    batch = {
        "text": torch.tensor([[5, 6, 7, 8, 9, 10, 11, 12]]),
        # roles per TOKEN: prompter x4, assistant x3, pad x1
        "assistant_mask": torch.tensor([[0, 0, 0, 0, 1, 1, 1, 0]]),
        "pad_mask": torch.tensor([[1, 1, 1, 1, 1, 1, 1, 0]]),
    }
    tokens, labels, loss_mask, am, pids = finetune.get_batch(iter([batch]))
    assert tokens.shape == (1, 7)
    assert torch.equal(labels, batch["text"][:, 1:])
    # positions 0..2 predict prompter tokens -> 0.25; 3..5 assistant -> 1;
    # 6 predicts the pad token -> 0
    expect = torch.tensor([[0.25, 0.25, 0.25, 1.0, 1.0, 1.0, 0.0]])
    assert torch.allclose(loss_mask, expect), loss_mask


def test_stream_reshard_bounded_and_correct(tmp_path):
    """Streaming reshard (tp1,pp2 -> tp2,pp1): results equal the full-merge
    reference, and the shard cache holds only ONE source PP stage at a time
    (the bounded-memory property replacing the reference's loader/saver
    subprocess streaming, reference tools/checkpoint_util.py:13-86)."""
    import argparse as ap
    import sys
    sys.path.insert(0, os.path.join(REPO, "tools"))
    import checkpoint_util as cu

    torch.manual_seed(5)
    H, V = 32, 64
    margs = ap.Namespace(
        num_layers=4, hidden_size=H, num_attention_heads=4,
        num_attention_heads_kv=2, ffn_hidden_size=3 * H,
        padded_vocab_size=V, tensor_model_parallel_size=1,
        pipeline_model_parallel_size=2, glu_activation="swiglu",
    )

    def layer_sd(prefix):
        return {
            f"{prefix}.input_layernorm.weight": torch.randn(H),
            f"{prefix}.self_attention.query_key_value.weight":
                torch.randn(2 * H, H),
            f"{prefix}.self_attention.dense.weight": torch.randn(H, H),
            f"{prefix}.post_attention_layernorm.weight": torch.randn(H),
            f"{prefix}.mlp.dense_h_to_4h.weight": torch.randn(6 * H, H),
            f"{prefix}.mlp.dense_4h_to_h.weight": torch.randn(H, 3 * H),
        }

    stage0 = {"embedding.word_embeddings.weight": torch.randn(V, H)}
    for i in range(2):
        stage0.update(layer_sd(f"encoder.layers.{i}"))
    stage1 = {}
    for i in range(2):
        stage1.update(layer_sd(f"encoder.layers.{i}"))
    stage1["encoder.final_layernorm.weight"] = torch.randn(H)
    stage1["lm_head"] = torch.randn(V, H)

    src = tmp_path / "src"
    for pp, sd in [(0, stage0), (1, stage1)]:
        d = src / "release" / f"mp_rank_00_{pp:03d}"
        d.mkdir(parents=True)
        torch.save({"args": margs, "checkpoint_version": 3.0, "iteration": 0,
                    "model": {"language_model": sd}},
                   d / "model_optim_rng.pt")
    (src / "latest_checkpointed_iteration.txt").write_text("release")

    out = tmp_path / "out"
    cache = cu.stream_reshard(str(src), str(out), tp=2, pp=1, glu=True,
                              progress=lambda *_: None)
    assert cache.max_loaded == 1  # one tp shard per stage, one stage live

    # reference: full merge then split
    shards = {(0, 0): {"model": {"language_model": stage0}},
              (0, 1): {"model": {"language_model": stage1}}}
    full = cu.merge_full_state(shards, 1, 2, 4, glu=True)
    ref_split = cu.split_full_state(full, 2, 1, 4, glu=True)

    for tpr in range(2):
        got = torch.load(
            out / "release" / f"mp_rank_{tpr:02d}" / "model_optim_rng.pt",
            map_location="cpu", weights_only=False,
        )["model"]["language_model"]
        ref = ref_split[(tpr, 0)]
        assert set(got.keys()) == set(ref.keys())
        for k in ref:
            assert torch.equal(got[k], ref[k]), k


def test_stream_reshard_tp2pp2_to_tp4pp1(tmp_path):
    """Streaming reshard with BOTH source axes sharded (tp2 x pp2 ->
    tp4 x pp1): equality vs the full-merge reference on every target."""
    import argparse as ap
    import sys
    sys.path.insert(0, os.path.join(REPO, "tools"))
    import checkpoint_util as cu

    torch.manual_seed(9)
    H, V = 32, 64
    margs = ap.Namespace(
        num_layers=4, hidden_size=H, num_attention_heads=8,
        num_attention_heads_kv=4, ffn_hidden_size=3 * H,
        padded_vocab_size=V, tensor_model_parallel_size=2,
        pipeline_model_parallel_size=2, glu_activation="swiglu",
    )

    def layer_sd(prefix, tp):
        # per-TP-shard shapes (dim0 keys halved, dim1 keys halved on dim1)
        return {
            f"{prefix}.input_layernorm.weight": torch.randn(H),
            f"{prefix}.self_attention.query_key_value.weight":
                torch.randn(H, H),
            f"{prefix}.self_attention.dense.weight": torch.randn(H, H // 2),
            f"{prefix}.mlp.dense_h_to_4h.weight": torch.randn(3 * H, H),
            f"{prefix}.mlp.dense_4h_to_h.weight":
                torch.randn(H, (3 * H) // 4),
        }

    shards = {}
    for pp in range(2):
        for tp in range(2):
            sd = {}
            if pp == 0:
                sd["embedding.word_embeddings.weight"] = (
                    torch.randn(V // 2, H)
                )
            for i in range(2):
                sd.update(layer_sd(f"encoder.layers.{i}", tp))
            if pp == 1:
                sd["encoder.final_layernorm.weight"] = torch.randn(H)
                sd["lm_head"] = torch.randn(V // 2, H)
            shards[(tp, pp)] = sd

    src = tmp_path / "src"
    for (tp, pp), sd in shards.items():
        d = src / "release" / f"mp_rank_{tp:02d}_{pp:03d}"
        d.mkdir(parents=True)
        torch.save({"args": margs, "checkpoint_version": 3.0, "iteration": 0,
                    "model": {"language_model": sd}},
                   d / "model_optim_rng.pt")
    (src / "latest_checkpointed_iteration.txt").write_text("release")

    out = tmp_path / "out"
    cache = cu.stream_reshard(str(src), str(out), tp=4, pp=1, glu=True,
                              progress=lambda *_: None)
    assert cache.max_loaded == 2  # one stage of 2 TP shards resident

    wrapped = {k: {"model": {"language_model": v}}
               for k, v in shards.items()}
    full = cu.merge_full_state(wrapped, 2, 2, 4, glu=True)
    ref_split = cu.split_full_state(full, 4, 1, 4, glu=True)
    for tpr in range(4):
        got = torch.load(
            out / "release" / f"mp_rank_{tpr:02d}" / "model_optim_rng.pt",
            map_location="cpu", weights_only=False,
        )["model"]["language_model"]
        ref = ref_split[(tpr, 0)]
        assert set(got.keys()) == set(ref.keys())
        for k in ref:
            assert torch.equal(got[k], ref[k]), k
