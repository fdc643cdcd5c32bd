"""Text-generation loop end-to-end on CPU: greedy generation with KV cache
must match repeated full forwards (reference generation.py:89-285 semantics)."""

import pytest
import torch

from megatron_amd import global_state
from megatron_amd.config import TrainingConfig, set_config


@pytest.fixture()
def small_model(dist_single):
    from megatron_amd.models import LlamaModel
    from megatron_amd.tokenizer.tokenizers import FakeTokenizer

    cfg = TrainingConfig(
        num_layers=2, hidden_size=64, num_attention_heads=4,
        num_attention_heads_kv=2, seq_length=64, max_position_embeddings=64,
        micro_batch_size=1, hidden_dropout=0.0, attention_dropout=0.0,
        use_cpu_initialization=True, use_flash_attn=True,
        inference_batch_times_seqlen_threshold=10_000,
    )
    cfg.finalize()
    cfg.pad_vocab_size(100)
    set_config(cfg)
    global_state.set_tokenizer(FakeTokenizer(100))
    torch.manual_seed(9)
    m = LlamaModel(cfg, parallel_output=False)
    m.eval()
    return m, cfg


def test_greedy_generation_matches_full_forward(small_model):
    from megatron_amd.inference.generation import (
        generate_tokens_probs_and_return_on_first_stage,
    )
    from megatron_amd.utils import get_ltor_masks_and_position_ids

    m, cfg = small_model
    prompt_len = 8
    total_len = 16
    torch.manual_seed(3)
    prompt = torch.randint(1, 99, (1, prompt_len))
    tokens = torch.zeros(1, total_len, dtype=torch.long)
    tokens[:, :prompt_len] = prompt
    lengths = torch.tensor([prompt_len])

    out_tokens, gen_lengths, logprobs = (
        generate_tokens_probs_and_return_on_first_stage(
            m, tokens.clone(), lengths, return_output_log_probs=True,
            top_k=1, use_eod_token_for_early_termination=False,
        )
    )

    # reference: greedy decode with full forward each step
    ref = tokens.clone()
    with torch.no_grad():
        for pos in range(prompt_len, total_len):
            inp = ref[:, :pos]
            am, _, pids = get_ltor_masks_and_position_ids(
                inp, 0, False, False, False
            )
            logits = m(inp, pids, am)
            # generation clamps samples to the true vocab size (the padded
            # dummy logits can win the argmax under random init)
            ref[0, pos] = logits[0, -1].argmax().clamp(max=99)

    assert torch.equal(out_tokens[0, :total_len], ref[0, :total_len]), (
        out_tokens, ref
    )
    assert logprobs is not None


def test_scoring_mode(small_model):
    from megatron_amd.inference.generation import (
        score_and_return_on_first_stage,
    )

    m, cfg = small_model
    tokens = torch.randint(1, 99, (2, 12))
    lengths = torch.tensor([12, 12])
    out_tokens, out_lengths, logprobs = score_and_return_on_first_stage(
        m, tokens, lengths
    )
    assert logprobs.shape == (2, 11)
    assert torch.isfinite(logprobs).all()


def test_sampling_top_k_top_p():
    from megatron_amd.inference.sampling import sample

    torch.manual_seed(0)
    logits = torch.randn(4, 50)
    greedy = sample(logits, top_k=1)
    assert torch.equal(greedy, logits.argmax(-1))
    s_k = sample(logits, top_k=5, temperature=0.7)
    assert s_k.shape == (4,)
    # top-k sampling can only return top-5 ids
    topk_ids = logits.topk(5, dim=-1).indices
    for i in range(4):
        assert s_k[i] in topk_ids[i]
    s_p = sample(logits, top_p=0.9)
    assert s_p.shape == (4,)


def test_beam_search_runs(small_model):
    from megatron_amd.inference.generation import (
        beam_search_and_return_on_first_stage,
    )

    m, cfg = small_model
    tokens = torch.zeros(1, 12, dtype=torch.long)
    tokens[:, :4] = torch.randint(1, 99, (1, 4))
    lengths = torch.tensor([4])
    out_tokens, scores = beam_search_and_return_on_first_stage(
        m, tokens, lengths, beam_size=2, stop_token=0, num_return_gen=2,
        length_penalty=1.0,
    )
    assert out_tokens.shape[0] == 2
    assert scores.shape[0] == 2


def test_rest_server_contract(small_model):
    """PUT /api with prompts returns text/segments (reference
    text_generation_server.py wire format) — exercised in-process via the
    FastAPI test client on the single-rank path."""
    httpx = pytest.importorskip("httpx")
    from fastapi.testclient import TestClient

    from megatron_amd.inference.server import MegatronServer

    m, cfg = small_model
    server = MegatronServer(m)
    app = server._build_app()
    client = TestClient(app)

    r = client.put("/api", json={"prompts": ["hello world"],
                                 "tokens_to_generate": 4, "top_k": 1})
    assert r.status_code == 200, r.text
    body = r.json()
    assert "text" in body and len(body["text"]) == 1
    assert "segments" in body

    r = client.put("/api", json={})
    assert r.status_code == 400


def test_stop_on_eol_and_colon_guard(small_model):
    """stop_on_eol terminates at the newline token (GPT-2 id 198, matching
    the reference's hard-coded convention) and prevent_newline_after_colon
    bans a newline directly after ':'."""
    import types

    from megatron_amd import global_state
    from megatron_amd.inference.generation import (
        generate_tokens_probs_and_return_on_first_stage,
    )

    model, cfg = small_model

    class Tok:
        eod = 0
        pad = 0
        vocab_size = 100

        def tokenize(self, text):
            return {":": [25], "\n": [198 % 100]}.get(text, [1])

        def detokenize(self, ids):
            return " ".join(map(str, ids))

    global_state.set_tokenizer(Tok())

    tokens = torch.zeros(1, 24, dtype=torch.long)
    tokens[:, :4] = torch.tensor([5, 6, 7, 8])
    lengths = torch.tensor([4])
    out, glen, _ = generate_tokens_probs_and_return_on_first_stage(
        model, tokens.clone(), lengths, top_k=1, stop_on_eol=True,
        prevent_newline_after_colon=True,
    )
    # newline id never follows a colon id
    seq = out[0].tolist()
    for a, b in zip(seq, seq[1:]):
        assert not (a == 25 and b == 198 % 100)
    # if a newline/eod was generated, the length reflects early stop
    assert glen[0] <= 24


def test_beam_search_scores_sorted_and_deterministic(small_model):
    from megatron_amd.inference.generation import (
        beam_search_and_return_on_first_stage,
    )

    m, cfg = small_model
    tokens = torch.zeros(1, 12, dtype=torch.long)
    tokens[:, :4] = torch.tensor([3, 5, 7, 9])
    lengths = torch.tensor([4])

    def run():
        return beam_search_and_return_on_first_stage(
            m, tokens.clone(), lengths, beam_size=3, stop_token=0,
            num_return_gen=3, length_penalty=1.0,
        )

    out1, scores1 = run()
    out2, scores2 = run()
    # deterministic (no sampling in beam search)
    assert torch.equal(out1, out2)
    assert torch.allclose(scores1, scores2)
    # returned hypotheses are best-first
    s = scores1.tolist()
    assert all(a >= b for a, b in zip(s, s[1:])), s


def test_sampling_modes():
    """sample(): greedy argmax, top-k restriction, top-p nucleus mass,
    temperature effect, vocab clamp (reference sampling.py:45-98)."""
    from megatron_amd.inference.sampling import sample

    torch.manual_seed(0)
    logits = torch.tensor([[1.0, 5.0, 3.0, 0.5, -2.0]])

    # greedy
    assert sample(logits, top_k=1).item() == 1

    # top-k=2: only the two best ids can ever be drawn
    draws = {sample(logits.clone(), top_k=2).item() for _ in range(50)}
    assert draws <= {1, 2}

    # top-p: with p just over the top token's mass, only id 1 survives
    probs = torch.softmax(logits, -1)
    draws = {
        sample(logits.clone(), top_p=probs[0, 1].item() - 0.01).item()
        for _ in range(20)
    }
    assert draws == {1}

    # near-zero temperature sharpens to greedy
    draws = {
        sample(logits.clone(), temperature=1e-4).item() for _ in range(20)
    }
    assert draws == {1}

    # vocab_size clamps out-of-vocab padding logits
    padded = torch.cat([logits, torch.full((1, 3), 100.0)], dim=1)
    assert sample(padded, top_k=1, vocab_size=5).item() < 5
