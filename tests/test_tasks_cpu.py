"""Downstream tasks on CPU: GLUE/RACE dataset parsing + packing, a real
one-epoch GLUE finetune through tasks.finetune_utils (with the accuracy
callback), and zero-shot LAMBADA/WikiText evaluation — all on synthetic
files with the FakeTokenizer."""

import json
import os
import sys

import numpy as np
import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

VOCAB = 128


class WordTokenizer:
    """Maps each word to a stable id — enough for task-data tests."""

    def __init__(self):
        self.eod = 0
        self.pad = 0
        self.cls = 3
        self.sep = 4
        self.mask = 5
        self.vocab_size = VOCAB

    def tokenize(self, text):
        return [6 + (hash(w) % (VOCAB - 6)) for w in text.split()]

    def detokenize(self, ids):
        return " ".join(str(i) for i in ids)


@pytest.fixture()
def qqp_file(tmp_path):
    p = tmp_path / "qqp_train.tsv"
    rows = ["id\tqid1\tqid2\tquestion1\tquestion2\tis_duplicate"]
    for i in range(12):
        rows.append(
            f"{i}\ta{i}\tb{i}\tis this question {i} real\t"
            f"is question {i} a duplicate\t{i % 2}"
        )
    p.write_text("\n".join(rows) + "\n")
    return str(p)


@pytest.fixture()
def mnli_file(tmp_path):
    p = tmp_path / "mnli_dev.tsv"
    header = "\t".join(f"c{j}" for j in range(12))
    rows = [header]
    labels = ["contradiction", "entailment", "neutral"]
    for i in range(9):
        cols = [str(i)] + ["x"] * 7 + [
            f"premise sentence {i}", f"hypothesis sentence {i}", "x",
            labels[i % 3],
        ]
        rows.append("\t".join(cols))
    p.write_text("\n".join(rows) + "\n")
    return str(p)


def test_qqp_dataset(qqp_file):
    from tasks.glue.qqp import QQPDataset

    tok = WordTokenizer()
    ds = QQPDataset("train", [qqp_file], tok, max_seq_length=32)
    assert len(ds) == 12
    s = ds[0]
    assert s["text"].shape == (32,)
    assert s["text"][0] == tok.cls
    assert s["label"] in (0, 1)
    # types flip to 1 on the B segment
    assert s["types"].max() == 1
    # padding mask covers the real tokens only
    n_real = int(s["padding_mask"].sum())
    assert (s["text"][n_real:] == tok.pad).all()


def test_mnli_dataset(mnli_file):
    from tasks.glue.mnli import MNLIDataset

    ds = MNLIDataset("dev", [mnli_file], WordTokenizer(), max_seq_length=32)
    assert len(ds) == 9
    assert sorted({ds[i]["label"] for i in range(9)}) == [0, 1, 2]


@pytest.fixture()
def race_dir(tmp_path):
    d = tmp_path / "race"
    d.mkdir()
    docs = []
    for i in range(3):
        docs.append(json.dumps({
            "article": f"some long article text number {i} " * 5,
            "questions": [f"what is the answer to question _ {i}"],
            "options": [[f"choice {c}" for c in range(4)]],
            "answers": ["B"],
        }))
    (d / "docs.txt").write_text("\n".join(docs) + "\n")
    return str(d)


def test_race_dataset(race_dir):
    from tasks.race.data import RaceDataset

    ds = RaceDataset("test", [race_dir], WordTokenizer(), max_seq_length=64)
    assert len(ds) == 3
    assert ds.sample_multiplier == 4
    s = ds[0]
    assert s["text"].shape == (4, 64)
    assert s["label"] == 1  # "B"


def test_glue_finetune_end_to_end(qqp_file, tmp_path, dist_single):
    """One real epoch of QQP classification through the shared finetune
    loop + accuracy callback, on a tiny BERT."""
    from megatron_amd import global_state
    from megatron_amd.config import TrainingConfig, set_config
    from megatron_amd.microbatches import setup_microbatch_calculator

    cfg = TrainingConfig(
        model_name="bert", task="QQP", num_layers=2, hidden_size=64,
        num_attention_heads=4, num_attention_heads_kv=4, seq_length=32,
        max_position_embeddings=64, micro_batch_size=4, global_batch_size=4,
        lr=1e-3, min_lr=1e-4, epochs=1, train_data=[qqp_file],
        valid_data=[qqp_file], hidden_dropout=0.0, attention_dropout=0.0,
        use_cpu_initialization=True, position_embedding_type="absolute",
        use_rms_norm=False, glu_activation=None, use_bias=True,
        use_flash_attn=False, bert_binary_head=False, clip_grad=1.0,
        lr_decay_style="constant", lr_warmup_iters=0, train_iters=3,
        eval_interval=1000, save=None, num_workers=0,
    )
    cfg.finalize()
    cfg.pad_vocab_size(VOCAB)
    set_config(cfg)
    global_state.init_timers()
    global_state.set_tokenizer(WordTokenizer())
    setup_microbatch_calculator(cfg)

    from tasks.glue.finetune import main as glue_main

    glue_main("QQP")  # runs 1 epoch + end-of-epoch accuracy callback


def test_zeroshot_lambada_and_wikitext(tmp_path, dist_single):
    from megatron_amd import global_state
    from megatron_amd.config import TrainingConfig, set_config
    from megatron_amd.microbatches import setup_microbatch_calculator

    lam = tmp_path / "lambada.jsonl"
    lam.write_text("\n".join(
        json.dumps({"text": f"a story about thing {i} ends with word"})
        for i in range(4)
    ) + "\n")
    wiki = tmp_path / "wiki.test.tokens"
    wiki.write_text("some text = = heading = = more text @-@ here " * 20)

    cfg = TrainingConfig(
        model_name="gpt", num_layers=2, hidden_size=64,
        num_attention_heads=4, num_attention_heads_kv=4, seq_length=32,
        max_position_embeddings=64, micro_batch_size=2, global_batch_size=2,
        hidden_dropout=0.0, attention_dropout=0.0,
        use_cpu_initialization=True, position_embedding_type="absolute",
        use_rms_norm=False, glu_activation=None, use_bias=True,
        use_flash_attn=False, valid_data=[str(lam)], overlapping_eval=16,
        num_workers=0,
    )
    cfg.finalize()
    cfg.pad_vocab_size(VOCAB)
    set_config(cfg)
    global_state.init_timers()
    global_state.set_tokenizer(WordTokenizer())
    setup_microbatch_calculator(cfg)

    from tasks.zeroshot_gpt.evaluate import main as zmain

    zmain("LAMBADA")
    cfg.valid_data = [str(wiki)]
    zmain("WIKITEXT103")


def test_detokenizers():
    from tasks.zeroshot_gpt.detokenizer import (
        get_detokenizer, wikitext_detokenizer,
    )

    assert wikitext_detokenizer("1 @-@ 2 @,@ 3") == "1-2,3"
    assert wikitext_detokenizer("= = h = =") == "== h =="
    assert get_detokenizer("/data/wiki.test.tokens")("a @-@ b") == "a-b"
    assert get_detokenizer("/data/lambada.jsonl")("x  y") == "x  y"


def test_msdp_f1_metric_and_eval(tmp_path):
    from tasks.msdp.evaluate import evaluate_f1
    from tasks.msdp.metrics import F1Metric, normalize_answer

    assert normalize_answer("The Cat, sat!") == "cat sat"
    p, r, f1 = F1Metric.compute_each_pair("the cat sat", "a cat sat down")
    assert 0 < f1 < 1
    assert F1Metric.compute_each_pair("", "gold")[2] == 0
    assert F1Metric.compute_each_pair("x", "")[2] is None

    g = tmp_path / "guess.txt"
    a = tmp_path / "answer.txt"
    g.write_text("the cat sat\nhello world<|endoftext|>\n")
    a.write_text("a cat sat down\nhello world\n")
    precision, recall, f1 = evaluate_f1(str(g), str(a))
    assert f1 > 0.5


def test_msdp_prompts_and_preprocessing(tmp_path):
    from tasks.msdp.preprocessing import (
        build_knowledge_prompts, process_wow_dataset,
    )
    from tasks.msdp.prompt import build_input, read_prompts

    raw = tmp_path / "wow.json"
    raw.write_text(json.dumps([{
        "chosen_topic": "Cats",
        "dialog": [
            {"speaker": "0_apprentice", "text": "tell me about cats"},
            {"speaker": "1_wizard", "text": "cats are felines",
             "checked_sentence": {"k": "Cats are small felines."},
             "checked_passage": {"p": "Cats"}},
        ],
    }]))
    tsv = tmp_path / "wow.tsv"
    kref = tmp_path / "k.txt"
    rref = tmp_path / "r.txt"
    process_wow_dataset(str(raw), str(tsv), str(kref), str(rref))
    line = tsv.read_text().strip()
    topic, context, knowledge, response = line.split("\t")
    assert topic == "Cats"
    assert knowledge == "Cats are small felines."

    pfile = tmp_path / "prompts.jsonl"
    build_knowledge_prompts(str(tsv), str(pfile), n_examples=5)
    prompts = read_prompts(str(pfile), "knowledge", 5)
    key = f"{topic} {context.split(' [SEP] ')[-1]}"
    assert key in prompts
    inp = build_input(line, "knowledge", prompts, None)
    assert inp.endswith("=>")

    rprompt = "Topic: X. User says: hi We know that: y System replies: z \n"
    (tmp_path / "rp.txt").write_text(rprompt)
    rp = read_prompts(str(tmp_path / "rp.txt"), "response", 10)
    inp2 = build_input(line, "response", None, rp)
    assert inp2.endswith("System replies:")


def test_orqa_qa_utils():
    from tasks.orqa.unsupervised.qa_utils import (
        calculate_matches, exact_match_score, has_answer,
    )

    assert has_answer(["Barack Obama"], "mr barack obama was president")
    assert not has_answer(["Obama"], "nothing here")
    assert has_answer([r"ob\w+"], "barack obama", match_type="regex")
    assert exact_match_score("The Answer!", "answer")

    all_docs = {1: ("the sky is blue", "t1"), 2: ("grass is green", "t2")}
    stats = calculate_matches(
        all_docs, [["blue"], ["purple"]],
        [([2, 1], [0.9, 0.8]), ([1, 2], [0.9, 0.8])],
    )
    assert stats.top_k_hits == [0, 1]  # first q hits at rank 2, second never
    assert stats.questions_doc_hits[0] == [False, True]


def test_orqa_retrieval_end_to_end(tmp_path, dist_single):
    """Evidence embedding -> MIPS retrieval -> top-k answer matching with a
    tiny random biencoder on CPU."""
    from megatron_amd import global_state
    from megatron_amd.config import TrainingConfig, set_config

    evidence = tmp_path / "evidence.tsv"
    evidence.write_text(
        "id\ttext\ttitle\n"
        "1\tthe sky is blue today\tweather\n"
        "2\tcats are small felines\tcats\n"
        "3\tparis is the capital of france\tfrance\n"
    )
    nq = tmp_path / "nq_dev.jsonl"
    nq.write_text("\n".join(json.dumps(d) for d in [
        {"question": "what color is the sky?", "answers": ["blue"]},
        {"question": "what is the capital of france?", "answers": ["paris"]},
    ]) + "\n")

    cfg = TrainingConfig(
        model_name="bert", num_layers=2, hidden_size=64,
        num_attention_heads=4, num_attention_heads_kv=4, seq_length=32,
        max_position_embeddings=64, micro_batch_size=2, global_batch_size=2,
        hidden_dropout=0.0, attention_dropout=0.0,
        use_cpu_initialization=True, position_embedding_type="absolute",
        use_rms_norm=False, glu_activation=None, use_bias=True,
        use_flash_attn=False, bert_binary_head=False,
        evidence_data_path=str(evidence), qa_data_dev=str(nq),
        num_workers=0, report_topk_accuracies=[1, 2],
    )
    cfg.finalize()
    cfg.pad_vocab_size(VOCAB)
    set_config(cfg)
    global_state.init_timers()
    global_state.set_tokenizer(WordTokenizer())

    from tasks.orqa.evaluate_utils import ORQAEvaluator

    evaluator = ORQAEvaluator()
    stats = evaluator.evaluate(str(nq), "DEV")
    # random model: hits are whatever they are, but shapes must line up
    assert len(stats.questions_doc_hits) == 2
    assert all(len(h) == 2 for h in stats.questions_doc_hits)


def test_orqa_supervised_forward(dist_single):
    from megatron_amd import global_state
    from megatron_amd.config import TrainingConfig, set_config
    from tasks.orqa.supervised import finetune as ret

    cfg = TrainingConfig(
        model_name="bert", num_layers=2, hidden_size=64,
        num_attention_heads=4, num_attention_heads_kv=4, seq_length=32,
        max_position_embeddings=64, micro_batch_size=4, global_batch_size=4,
        hidden_dropout=0.0, attention_dropout=0.0,
        use_cpu_initialization=True, position_embedding_type="absolute",
        use_rms_norm=False, glu_activation=None, use_bias=True,
        use_flash_attn=False, bert_binary_head=False, num_workers=0,
    )
    cfg.finalize()
    cfg.pad_vocab_size(VOCAB)
    set_config(cfg)
    global_state.init_timers()
    global_state.set_tokenizer(WordTokenizer())

    model = ret.model_provider()
    batch = {
        "query": torch.randint(6, VOCAB, (4, 32)),
        "query_pad_mask": torch.ones(4, 32, dtype=torch.long),
        "context": torch.randint(6, VOCAB, (4, 32)),
        "context_pad_mask": torch.ones(4, 32, dtype=torch.long),
    }
    scores, loss_closure = ret._forward_step(batch, model)
    assert scores.shape == (4, 4)
    loss, stats = loss_closure(scores)
    assert torch.isfinite(loss)
    loss.backward()
    assert "in-batch acc" in stats


def test_index_builder_and_blockdata_roundtrip(tmp_path, dist_single):
    """IndexBuilder embeds blocks with the biencoder context tower, shards
    save/merge/load, and the MIPS index retrieves the nearest block."""
    import numpy as np

    from megatron_amd import global_state
    from megatron_amd.config import TrainingConfig, set_config
    from megatron_amd.inference.indexer import IndexBuilder
    from megatron_amd.inference.realm_index import (
        BlockData, FaissMIPSIndex,
    )
    from megatron_amd.models.biencoder_model import BiEncoderModel

    cfg = TrainingConfig(
        model_name="bert", num_layers=2, hidden_size=64,
        num_attention_heads=4, num_attention_heads_kv=4, seq_length=16,
        max_position_embeddings=32, micro_batch_size=2,
        hidden_dropout=0.0, attention_dropout=0.0,
        use_cpu_initialization=True, position_embedding_type="absolute",
        use_rms_norm=False, glu_activation=None, use_bias=True,
        use_flash_attn=False, bert_binary_head=False,
    )
    cfg.finalize()
    cfg.pad_vocab_size(VOCAB)
    set_config(cfg)
    global_state.set_tokenizer(WordTokenizer())
    model = BiEncoderModel(cfg, projection_dim=16)

    class Blocks(torch.utils.data.Dataset):
        def __len__(self):
            return 6

        def __getitem__(self, i):
            t = torch.full((16,), 6 + i, dtype=torch.long)
            return {"context_tokens": t,
                    "context_mask": torch.ones(16, dtype=torch.long),
                    "block_id": i}

    path = str(tmp_path / "embeds.pkl")
    builder = IndexBuilder(model, Blocks(), batch_size=3,
                           embedding_path=path)
    bd = builder.build_and_save_index()
    assert len(bd.embed_data) == 6

    merged = BlockData(path)
    merged.merge_shards_and_save([path + ".rank0"])
    loaded = BlockData(path, load_from_path=True)
    assert len(loaded.embed_data) == 6

    index = FaissMIPSIndex(embed_size=16, embed_data=loaded, use_gpu=False)
    q = model.embed_query(
        torch.full((1, 16), 8, dtype=torch.long),
        torch.ones(1, 16, dtype=torch.long),
    ).detach()
    scores, ids = index.search_mips_index(q, top_k=3)
    assert scores.shape == (1, 3) and ids.shape == (1, 3)
    # the index must agree with brute-force inner products
    import numpy as _np

    mat = _np.stack([loaded.embed_data[i] for i in range(6)]).astype(
        _np.float32
    )
    brute = (q.numpy() @ mat.T)[0]
    assert list(ids[0]) == list(_np.argsort(-brute)[:3])
