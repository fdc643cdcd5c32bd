"""Convert HuggingFace (or Meta-format) Llama / Llama-2 / Code-Llama /
Mistral / Falcon weights into a megatron_amd 'release' checkpoint.

Reference behavior: weights_conversion/hf_to_megatron.py:60-449. Output layout:

  <out>/release/mp_rank_00/model_optim_rng.pt  (tp=1, pp=1, unsharded)
  <out>/latest_checkpointed_iteration.txt      containing 'release'

The model state dict uses megatron_amd's flat key scheme under
'language_model':
  embedding.word_embeddings.weight
  encoder.layers.<i>.input_layernorm.weight
  encoder.layers.<i>.self_attention.query_key_value.weight   ([q...,k,v] per
      KV group, rotary dims interleaved via permute_qkv)
  encoder.layers.<i>.self_attention.dense.weight
  encoder.layers.<i>.post_attention_layernorm.weight
  encoder.layers.<i>.mlp.dense_h_to_4h.weight  (= concat [up(w3); gate(w1)])
  encoder.layers.<i>.mlp.dense_4h_to_h.weight
  encoder.final_layernorm.weight
  lm_head                                       (untied families)

Usage:
  python weights_conversion/hf_to_megatron.py llama2 \
      --size 7 --cache-dir /path/hf_model --out /path/megatron_ckpt
"""

from __future__ import annotations

import argparse
import os
import sys
from pathlib import Path

import torch

sys.path.append(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from weights_conversion.permute_qkv import permute_qkv  # noqa: E402

LLAMA_SPECS = {
    # size -> (layers, hidden, heads, kv heads, ffn)
    ("llama", 7): (32, 4096, 32, 32, 11008),
    ("llama", 13): (40, 5120, 40, 40, 13824),
    ("llama", 30): (60, 6656, 52, 52, 17920),
    ("llama", 65): (80, 8192, 64, 64, 22016),
    ("llama2", 7): (32, 4096, 32, 32, 11008),
    ("llama2", 13): (40, 5120, 40, 40, 13824),
    ("llama2", 70): (80, 8192, 64, 8, 28672),
    ("codellama", 7): (32, 4096, 32, 32, 11008),
    ("codellama", 13): (40, 5120, 40, 40, 13824),
    ("codellama", 34): (48, 8192, 64, 8, 22016),
    ("mistral", 7): (32, 4096, 32, 8, 14336),
}


def load_hf_state_dict(cache_dir: str) -> dict:
    """Load a HF checkpoint directory (safetensors or bin shards)."""
    import glob
    
    weights = {}
    st_files = sorted(glob.glob(os.path.join(cache_dir, "*.safetensors")))
    if st_files:
        from safetensors.torch import load_file

        for f in st_files:
            weights.update(load_file(f))
        return weights
    bin_files = sorted(glob.glob(os.path.join(cache_dir, "*.bin")))
    if bin_files:
        for f in bin_files:
            weights.update(torch.load(f, map_location="cpu",
                                      weights_only=True))
        return weights
    raise FileNotFoundError(f"no HF weight files found in {cache_dir}")


def llama_like_to_megatron(weights: dict, n_layers: int, hidden: int,
                           n_heads: int, n_kv_heads: int) -> dict:
    """HF LlamaForCausalLM / MistralForCausalLM state dict -> megatron_amd."""
    head_dim = hidden // n_heads
    sd = {}

    def hf(key):
        for prefix in ("model.", ""):
            if prefix + key in weights:
                return weights[prefix + key]
        raise KeyError(key)

    sd["embedding.word_embeddings.weight"] = hf("embed_tokens.weight")
    sd["lm_head"] = weights.get("lm_head.weight", hf("embed_tokens.weight"))
    sd["encoder.final_layernorm.weight"] = hf("norm.weight")

    for i in range(n_layers):
        p = f"layers.{i}"
        o = f"encoder.layers.{i}"
        sd[f"{o}.input_layernorm.weight"] = hf(f"{p}.input_layernorm.weight")
        sd[f"{o}.post_attention_layernorm.weight"] = hf(
            f"{p}.post_attention_layernorm.weight"
        )
        # MLP: dense_h_to_4h = [up (w3); gate (w1)]
        # (reference hf_to_megatron.py:237-239)
        sd[f"{o}.mlp.dense_h_to_4h.weight"] = torch.cat(
            [
                hf(f"{p}.mlp.up_proj.weight"),
                hf(f"{p}.mlp.gate_proj.weight"),
            ], dim=0,
        )
        sd[f"{o}.mlp.dense_4h_to_h.weight"] = hf(f"{p}.mlp.down_proj.weight")

        wq = hf(f"{p}.self_attn.q_proj.weight")
        wk = hf(f"{p}.self_attn.k_proj.weight")
        wv = hf(f"{p}.self_attn.v_proj.weight")
        sd[f"{o}.self_attention.query_key_value.weight"] = rearrange_qkv(
            wq, wk, wv, n_heads, n_kv_heads, head_dim, hidden
        )
        sd[f"{o}.self_attention.dense.weight"] = hf(
            f"{p}.self_attn.o_proj.weight"
        )
    return sd


def rearrange_qkv(wq, wk, wv, n_heads, n_kv_heads, head_dim, hidden):
    """Interleave [q_1..q_nq, k, v] per KV group then fix rotary ordering
    (reference hf_to_megatron.py:185-235)."""
    wq = torch.split(wq, head_dim, dim=0)
    wk = torch.split(wk, head_dim, dim=0)
    wv = torch.split(wv, head_dim, dim=0)
    assert len(wq) == n_heads
    assert len(wk) == n_kv_heads
    assert len(wv) == n_kv_heads
    n_qs_per_kv = n_heads // n_kv_heads
    w_qkv = []
    for g in range(n_kv_heads):
        w_qkv += [wq[g * n_qs_per_kv + i] for i in range(n_qs_per_kv)]
        w_qkv += [wk[g], wv[g]]
    return permute_qkv(torch.cat(w_qkv, dim=0), hidden, n_heads, n_kv_heads)


def _permute_rotary_rows(w: torch.Tensor, n_heads_in_w: int) -> torch.Tensor:
    """Reorder each head's rows from HF's half-split rotary pairing
    ((j, j+d/2) rotated together by rotate_half) to the interleaved pairing
    ((2j, 2j+1)) our RoPE kernel applies — the falcon analog of permute_qkv."""
    total, cols = w.shape
    d = total // n_heads_in_w
    w = w.view(n_heads_in_w, d, cols)
    half = d // 2
    idx = torch.arange(d)
    interleaved = torch.empty_like(idx)
    interleaved[0::2] = idx[:half]
    interleaved[1::2] = idx[half:]
    return w[:, interleaved, :].reshape(total, cols)


def falcon_to_megatron(weights: dict, size: int, n_layers: int = None,
                       n_heads: int = None, n_kv: int = None) -> dict:
    """HF FalconForCausalLM -> megatron_amd (reference hf_to_megatron.py:60-114).

    The fused query_key_value layout ([q heads..., k, v] per kv group)
    matches ours directly; only the q/k rotary row pairing is converted from
    HF's rotate_half convention to our interleaved-pair kernel."""
    sd = {}
    prefix1 = "transformer."
    if n_layers is None:
        n_layers = 32 if size == 7 else 60
    if n_heads is None:
        n_heads = 71 if size == 7 else 128
    if n_kv is None:
        n_kv = 1 if size == 7 else 8

    sd["embedding.word_embeddings.weight"] = weights[
        f"{prefix1}word_embeddings.weight"
    ]
    sd["encoder.final_layernorm.weight"] = weights[f"{prefix1}ln_f.weight"]
    sd["encoder.final_layernorm.bias"] = weights[f"{prefix1}ln_f.bias"]
    for i in range(n_layers):
        p = f"{prefix1}h.{i}"
        o = f"encoder.layers.{i}"
        if size == 7:
            sd[f"{o}.input_layernorm.weight"] = weights[
                f"{p}.input_layernorm.weight"
            ]
            sd[f"{o}.input_layernorm.bias"] = weights[
                f"{p}.input_layernorm.bias"
            ]
        else:
            sd[f"{o}.input_layernorm.weight"] = weights[f"{p}.ln_attn.weight"]
            sd[f"{o}.input_layernorm.bias"] = weights[f"{p}.ln_attn.bias"]
            sd[f"{o}.mlp_layernorm.weight"] = weights[f"{p}.ln_mlp.weight"]
            sd[f"{o}.mlp_layernorm.bias"] = weights[f"{p}.ln_mlp.bias"]
        qkv = weights[f"{p}.self_attention.query_key_value.weight"]
        # per kv group the fused rows are [nq q-heads, k, v] * head_dim; q
        # and k rows get the rotary-pairing permutation, v rows stay
        n_rows, hidden = qkv.shape
        hd = hidden // n_heads
        groups = n_rows // ((n_heads // n_kv + 2) * hd)
        nq = n_heads // n_kv
        qkv = qkv.view(groups, (nq + 2) * hd, hidden)
        q = qkv[:, : nq * hd, :].reshape(groups * nq * hd, hidden)
        k = qkv[:, nq * hd : (nq + 1) * hd, :].reshape(groups * hd, hidden)
        v = qkv[:, (nq + 1) * hd :, :].reshape(groups * hd, hidden)
        q = _permute_rotary_rows(q, groups * nq)
        k = _permute_rotary_rows(k, groups)
        sd[f"{o}.self_attention.query_key_value.weight"] = torch.cat(
            [
                q.view(groups, nq * hd, hidden),
                k.view(groups, hd, hidden),
                v.view(groups, hd, hidden),
            ],
            dim=1,
        ).reshape(n_rows, hidden)
        sd[f"{o}.self_attention.dense.weight"] = weights[
            f"{p}.self_attention.dense.weight"
        ]
        sd[f"{o}.mlp.dense_h_to_4h.weight"] = weights[
            f"{p}.mlp.dense_h_to_4h.weight"
        ]
        sd[f"{o}.mlp.dense_4h_to_h.weight"] = weights[
            f"{p}.mlp.dense_4h_to_h.weight"
        ]
    return sd


def pad_embeddings(sd: dict, make_vocab_size_divisible_by: int = 128):
    for key in ("embedding.word_embeddings.weight", "lm_head"):
        if key not in sd:
            continue
        w = sd[key]
        vocab = w.shape[0]
        padded = vocab
        while padded % make_vocab_size_divisible_by != 0:
            padded += 1
        if padded != vocab:
            pad = torch.zeros(padded - vocab, w.shape[1], dtype=w.dtype)
            sd[key] = torch.cat([w, pad], dim=0)
    return sd


def write_megatron_checkpoint(sd: dict, out_dir: str, model_name: str,
                              spec: tuple, dtype=torch.bfloat16):
    out = Path(out_dir)
    ckpt_dir = out / "release" / "mp_rank_00"
    ckpt_dir.mkdir(parents=True, exist_ok=True)

    n_layers, hidden, n_heads, n_kv_heads, ffn = spec
    sd = {k: v.to(dtype) for k, v in sd.items()}

    import argparse as ap

    args = ap.Namespace(
        num_layers=n_layers, hidden_size=hidden, num_attention_heads=n_heads,
        num_attention_heads_kv=n_kv_heads, ffn_hidden_size=ffn,
        padded_vocab_size=sd["embedding.word_embeddings.weight"].shape[0],
        tensor_model_parallel_size=1, pipeline_model_parallel_size=1,
        model_name=model_name, iteration="release",
    )
    state = {
        "args": args,
        "checkpoint_version": 3.0,
        "iteration": 0,
        "model": {"language_model": sd},
    }
    torch.save(state, ckpt_dir / "model_optim_rng.pt")
    with open(out / "latest_checkpointed_iteration.txt", "w") as f:
        f.write("release")
    print(f"saved megatron checkpoint to {out}")


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("model", choices=["llama", "llama2", "codellama",
                                          "mistral", "falcon"])
    parser.add_argument("--size", type=int, required=True)
    parser.add_argument("--cache-dir", "--cache_dir", dest="cache_dir",
                        required=True, help="HF checkpoint directory")
    parser.add_argument("--out", required=True)
    parser.add_argument("--dtype", default="bf16", choices=["bf16", "fp16",
                                                            "fp32"])
    args = parser.parse_args()

    dtype = {"bf16": torch.bfloat16, "fp16": torch.float16,
             "fp32": torch.float32}[args.dtype]

    import glob as _glob

    meta_shards = _glob.glob(os.path.join(args.cache_dir,
                                          "consolidated.*.pth"))
    if meta_shards and args.model != "falcon":
        # Meta-format checkpoint dir: merge shards -> HF-style dict
        # (reference hf_to_megatron.py:284 via utils/merge_llama.py)
        from weights_conversion.merge_llama import merge_meta_llama

        spec0 = LLAMA_SPECS[(args.model, args.size)]
        weights = merge_meta_llama(args.cache_dir, n_heads=spec0[2],
                                   n_kv_heads=spec0[3])
    else:
        weights = load_hf_state_dict(args.cache_dir)
    if args.model == "falcon":
        n_layers = 32 if args.size == 7 else 60
        hidden = 4544 if args.size == 7 else 8192
        n_heads = 71 if args.size == 7 else 128
        n_kv = 1 if args.size == 7 else 8
        spec = (n_layers, hidden, n_heads, n_kv, 4 * hidden)
        sd = falcon_to_megatron(weights, args.size)
    else:
        spec = LLAMA_SPECS[(args.model, args.size)]
        n_layers, hidden, n_heads, n_kv_heads, ffn = spec
        sd = llama_like_to_megatron(weights, n_layers, hidden, n_heads,
                                    n_kv_heads)
    sd = pad_embeddings(sd)
    write_megatron_checkpoint(sd, args.out, args.model, spec, dtype)


if __name__ == "__main__":
    main()
