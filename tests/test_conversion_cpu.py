"""HF <-> megatron_amd conversion parity against transformers' Llama/Mistral
implementations (random-init, CPU, fp32) — the offline analog of the
reference's verify_correctness.py + tests/test_llama_weights.py pipeline."""

import os
import sys

import pytest
import torch

sys.path.append(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from megatron_amd.config import TrainingConfig, set_config

transformers = pytest.importorskip("transformers")


def _hf_tiny_llama(n_kv_heads=4):
    from transformers import LlamaConfig, LlamaForCausalLM

    cfg = LlamaConfig(
        vocab_size=128, hidden_size=64, intermediate_size=176,
        num_hidden_layers=2, num_attention_heads=4,
        num_key_value_heads=n_kv_heads, max_position_embeddings=64,
        rms_norm_eps=1e-5, tie_word_embeddings=False, rope_theta=10000.0,
        attention_bias=False,
    )
    torch.manual_seed(42)
    return LlamaForCausalLM(cfg).eval(), cfg


def _convert_to_ours(hf_model, hf_cfg):
    from weights_conversion.hf_to_megatron import (
        llama_like_to_megatron, pad_embeddings,
    )

    weights = hf_model.state_dict()
    sd = llama_like_to_megatron(
        weights, hf_cfg.num_hidden_layers, hf_cfg.hidden_size,
        hf_cfg.num_attention_heads, hf_cfg.num_key_value_heads,
    )
    sd = pad_embeddings(sd, make_vocab_size_divisible_by=128)
    return sd


def _our_model(hf_cfg, sd):
    from megatron_amd.models import LlamaModel

    cfg = TrainingConfig(
        num_layers=hf_cfg.num_hidden_layers,
        hidden_size=hf_cfg.hidden_size,
        ffn_hidden_size=hf_cfg.intermediate_size,
        num_attention_heads=hf_cfg.num_attention_heads,
        num_attention_heads_kv=hf_cfg.num_key_value_heads,
        seq_length=32, max_position_embeddings=64,
        micro_batch_size=1, hidden_dropout=0.0, attention_dropout=0.0,
        use_cpu_initialization=True, use_flash_attn=False,
        perform_initialization=False,
        layernorm_epsilon=hf_cfg.rms_norm_eps,
    )
    cfg.finalize()
    cfg.padded_vocab_size = sd["embedding.word_embeddings.weight"].shape[0]
    set_config(cfg)
    m = LlamaModel(cfg, parallel_output=False)
    missing, unexpected = m.language_model.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert not missing, missing
    return m.eval(), cfg


@pytest.mark.parametrize("n_kv", [4, 2])
def test_llama_conversion_logit_parity(dist_single, n_kv):
    hf_model, hf_cfg = _hf_tiny_llama(n_kv)
    sd = _convert_to_ours(hf_model, hf_cfg)
    ours, cfg = _our_model(hf_cfg, sd)

    from megatron_amd.utils import get_ltor_masks_and_position_ids

    torch.manual_seed(0)
    tokens = torch.randint(0, 128, (2, 32))
    am, _, pids = get_ltor_masks_and_position_ids(tokens, 0, False, False,
                                                  False)
    with torch.no_grad():
        hf_logits = hf_model(tokens).logits
        our_logits = ours(tokens, pids, am)[:, :, :128]

    max_err = (hf_logits - our_logits).abs().max().item()
    # reference gate: <=1e-3 avg max error at fp32 (test_llama_weights.py:117)
    assert max_err < 1e-3, f"logit mismatch {max_err}"


def test_roundtrip_megatron_to_hf(dist_single):
    from weights_conversion.megatron_to_hf import megatron_to_hf_llama

    hf_model, hf_cfg = _hf_tiny_llama(2)
    sd = _convert_to_ours(hf_model, hf_cfg)
    back = megatron_to_hf_llama(
        sd, hf_cfg.num_hidden_layers, hf_cfg.hidden_size,
        hf_cfg.num_attention_heads, hf_cfg.num_key_value_heads,
        hf_cfg.intermediate_size, hf_cfg.vocab_size,
    )
    orig = hf_model.state_dict()
    for k, v in back.items():
        ok = k if k in orig else k.replace("model.", "", 1)
        assert torch.equal(v, orig[k]), k


def test_permute_qkv_roundtrip():
    from weights_conversion.permute_qkv import permute_qkv

    dim, heads, kv = 64, 4, 2
    head_dim = dim // heads
    w = torch.randn((heads + 2 * kv) * head_dim, dim)
    assert torch.equal(
        permute_qkv(permute_qkv(w, dim, heads, kv), dim, heads, kv,
                    revert=True),
        w,
    )


def test_falcon_conversion_logit_parity(dist_single):
    """Tiny HF Falcon (MQA + parallel attention) -> FalconModel: fp32 logits
    must agree (reference weights_conversion covers falcon too)."""
    transformers = pytest.importorskip("transformers")
    from transformers import FalconConfig, FalconForCausalLM

    from megatron_amd.models import FalconModel
    from weights_conversion.hf_to_megatron import (
        falcon_to_megatron, pad_embeddings,
    )

    hf_cfg = FalconConfig(
        vocab_size=128, hidden_size=64, num_hidden_layers=2,
        num_attention_heads=4, multi_query=True, parallel_attn=True,
        bias=False, new_decoder_architecture=False, alibi=False,
    )
    torch.manual_seed(11)
    hf_model = FalconForCausalLM(hf_cfg).eval()

    sd = falcon_to_megatron(hf_model.state_dict(), size=7, n_layers=2,
                            n_heads=4, n_kv=1)
    sd = pad_embeddings(sd, make_vocab_size_divisible_by=128)

    cfg = TrainingConfig(
        model_name="falcon",
        num_layers=2, hidden_size=64, ffn_hidden_size=256,
        num_attention_heads=4, num_attention_heads_kv=1,
        seq_length=32, max_position_embeddings=64, micro_batch_size=1,
        hidden_dropout=0.0, attention_dropout=0.0,
        use_cpu_initialization=True, use_flash_attn=False,
        perform_initialization=False,
        layernorm_epsilon=hf_cfg.layer_norm_epsilon,
    )
    cfg.finalize()
    cfg.padded_vocab_size = sd["embedding.word_embeddings.weight"].shape[0]
    set_config(cfg)
    ours = FalconModel(cfg, parallel_output=False)
    missing, unexpected = ours.language_model.load_state_dict(sd,
                                                              strict=False)
    assert not unexpected, unexpected
    assert not [m for m in missing if "rope" not in m], missing
    ours.eval()

    from megatron_amd.utils import get_ltor_masks_and_position_ids

    tokens = torch.randint(0, 128, (1, 16))
    am, _, pids = get_ltor_masks_and_position_ids(tokens, 0, False, False,
                                                  False)
    with torch.no_grad():
        ours_logits = ours(tokens, pids, am).float()[:, :, :128]
        hf_logits = hf_model(tokens).logits.float()
    err = (ours_logits - hf_logits).abs().max().item()
    assert err < 1e-3, f"falcon logit error {err}"


def test_falcon_roundtrip_megatron_to_hf(dist_single):
    """HF falcon -> megatron -> HF must restore every tensor bitwise."""
    transformers = pytest.importorskip("transformers")
    from transformers import FalconConfig, FalconForCausalLM

    from weights_conversion.hf_to_megatron import falcon_to_megatron
    from weights_conversion.megatron_to_hf import megatron_to_hf_falcon

    hf_cfg = FalconConfig(
        vocab_size=128, hidden_size=64, num_hidden_layers=2,
        num_attention_heads=4, multi_query=True, parallel_attn=True,
        bias=False, new_decoder_architecture=False, alibi=False,
    )
    torch.manual_seed(5)
    hf_model = FalconForCausalLM(hf_cfg)
    orig = hf_model.state_dict()
    sd = falcon_to_megatron(orig, size=7, n_layers=2, n_heads=4, n_kv=1)
    back = megatron_to_hf_falcon(sd, 2, 64, 4, 1, 128)
    for k, v in back.items():
        assert torch.equal(v, orig[k]), k


def test_merge_meta_llama(tmp_path):
    """Meta consolidated.XX.pth shards -> merged HF-style dict: column-
    parallel weights stitch on dim 0, row-parallel on dim 1, norms are
    replicated, rope.freqs dropped, and q/k rotary rows convert from Meta's
    interleaved pairing to HF's half-split so the standard pipeline
    (rearrange_qkv + permute_qkv) applies unchanged."""
    from weights_conversion.hf_to_megatron import _permute_rotary_rows
    from weights_conversion.merge_llama import merge_meta_llama

    n_heads, head_dim, hidden, ffn, vocab = 4, 8, 32, 48, 64
    torch.manual_seed(11)

    # Build the FULL model in HF convention first, then shard it the way
    # Meta does (2 shards), converting q/k to Meta's interleaved rows.
    full = {
        "tok_embeddings.weight": torch.randn(vocab, hidden),
        "norm.weight": torch.randn(hidden),
        "output.weight": torch.randn(vocab, hidden),
        "layers.0.attention.wq.weight": torch.randn(hidden, hidden),
        "layers.0.attention.wk.weight": torch.randn(hidden, hidden),
        "layers.0.attention.wv.weight": torch.randn(hidden, hidden),
        "layers.0.attention.wo.weight": torch.randn(hidden, hidden),
        "layers.0.feed_forward.w1.weight": torch.randn(ffn, hidden),
        "layers.0.feed_forward.w2.weight": torch.randn(hidden, ffn),
        "layers.0.feed_forward.w3.weight": torch.randn(ffn, hidden),
        "layers.0.attention_norm.weight": torch.randn(hidden),
        "layers.0.ffn_norm.weight": torch.randn(hidden),
        "rope.freqs": torch.randn(head_dim // 2),
    }
    # q/k in the Meta files use interleaved rotary rows = what our kernel
    # uses = _permute_rotary_rows applied to the HF layout
    meta_full = dict(full)
    for key in ("layers.0.attention.wq.weight",
                "layers.0.attention.wk.weight"):
        meta_full[key] = _permute_rotary_rows(full[key], n_heads)

    dim0 = {"wq", "wk", "wv", "w1", "w3", "output"}
    dim1 = {"tok_embeddings", "wo", "w2"}
    shards = [{}, {}]
    for k, w in meta_full.items():
        short = k.split(".")[-2]
        if short in dim0:
            halves = torch.chunk(w, 2, dim=0)
        elif short in dim1:
            halves = torch.chunk(w, 2, dim=1)
        else:
            halves = (w, w)
        shards[0][k], shards[1][k] = halves[0], halves[1]
    torch.save(shards[0], tmp_path / "consolidated.00.pth")
    torch.save(shards[1], tmp_path / "consolidated.01.pth")

    hf = merge_meta_llama(str(tmp_path), n_heads=n_heads)

    assert "rope.freqs" not in hf and not any("rope" in k for k in hf)
    assert torch.equal(hf["model.embed_tokens.weight"],
                       full["tok_embeddings.weight"])
    assert torch.equal(hf["lm_head.weight"], full["output.weight"])
    assert torch.equal(hf["model.norm.weight"], full["norm.weight"])
    assert torch.equal(hf["model.layers.0.input_layernorm.weight"],
                       full["layers.0.attention_norm.weight"])
    assert torch.equal(hf["model.layers.0.mlp.gate_proj.weight"],
                       full["layers.0.feed_forward.w1.weight"])
    assert torch.equal(hf["model.layers.0.mlp.down_proj.weight"],
                       full["layers.0.feed_forward.w2.weight"])
    assert torch.equal(hf["model.layers.0.self_attn.o_proj.weight"],
                       full["layers.0.attention.wo.weight"])
    # rotary rows restored to HF half-split convention exactly: sharded
    # interleaved Meta q/k merge back to the original HF-layout tensors.
    # NOTE: with 2 shards of 4 heads, each shard holds 2 whole heads, so
    # per-head row permutations commute with the sharding.
    assert torch.equal(hf["model.layers.0.self_attn.q_proj.weight"],
                       full["layers.0.attention.wq.weight"])
    assert torch.equal(hf["model.layers.0.self_attn.k_proj.weight"],
                       full["layers.0.attention.wk.weight"])
    assert torch.equal(hf["model.layers.0.self_attn.v_proj.weight"],
                       full["layers.0.attention.wv.weight"])
