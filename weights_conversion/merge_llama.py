"""Merge Meta-format Llama checkpoints (consolidated.XX.pth shards) into one
HF-style state dict that feeds `llama_like_to_megatron` unchanged.

Reference behavior: weights_conversion/utils/merge_llama.py:22-100 — Meta
shards a model column-parallel (dim 0: wq, wk, wv, w1, w3, output) or
row-parallel (dim 1: wo, w2, tok_embeddings); norms and rope.freqs are
replicated. This implementation concatenates along the right dim, then
additionally remaps Meta key names to HF names and converts q/k rotary row
ordering from Meta's interleaved-pair convention to HF's half-split
(rotate_half) convention, so the output is byte-for-byte an HF LlamaForCausalLM
state dict and the rest of the pipeline (rearrange_qkv + permute_qkv) needs no
source-specific branches.

Usage:
  from weights_conversion.merge_llama import merge_meta_llama
  hf_weights = merge_meta_llama("/path/to/llama-2-7b", n_heads=32)
"""

from __future__ import annotations

import glob
import os
import re

import torch

# Meta short name -> concat dim (None = replicated, take shard 0)
_CAT_DIM = {
    "wq": 0, "wk": 0, "wv": 0, "w1": 0, "w3": 0, "output": 0,
    "tok_embeddings": 1, "wo": 1, "w2": 1,
    "attention_norm": None, "ffn_norm": None, "norm": None, "rope": None,
}

# Meta key fragment -> HF key fragment (per layer)
_LAYER_MAP = {
    "attention.wq": "self_attn.q_proj",
    "attention.wk": "self_attn.k_proj",
    "attention.wv": "self_attn.v_proj",
    "attention.wo": "self_attn.o_proj",
    "feed_forward.w1": "mlp.gate_proj",
    "feed_forward.w2": "mlp.down_proj",
    "feed_forward.w3": "mlp.up_proj",
    "attention_norm": "input_layernorm",
    "ffn_norm": "post_attention_layernorm",
}


def _rows_interleaved_to_half_split(w: torch.Tensor,
                                    n_heads: int) -> torch.Tensor:
    """Per head, reorder rows from Meta's interleaved rotary pairing
    ((2j, 2j+1) rotated together) to HF's half-split pairing ((j, j+d/2)) —
    the inverse of hf_to_megatron._permute_rotary_rows."""
    total, cols = w.shape
    d = total // n_heads
    half = d // 2
    idx = torch.arange(d)
    interleaved = torch.empty_like(idx)
    interleaved[0::2] = idx[:half]
    interleaved[1::2] = idx[half:]
    inverse = torch.argsort(interleaved)
    return w.view(n_heads, d, cols)[:, inverse, :].reshape(total, cols)


def merge_shards(shards: list[dict]) -> dict:
    """Concatenate Meta shard state dicts along each weight's parallel dim."""
    merged = {}
    for key in shards[0]:
        short = key.split(".")[-2]
        dim = _CAT_DIM[short]
        if dim is None or len(shards) == 1:
            merged[key] = shards[0][key]
        else:
            merged[key] = torch.cat([s[key] for s in shards], dim=dim)
    return merged


def meta_to_hf_keys(meta: dict, n_heads: int,
                    n_kv_heads: int | None = None) -> dict:
    """Rename Meta keys to HF LlamaForCausalLM keys and fix q/k rotary row
    ordering (Meta interleaved -> HF half-split)."""
    if n_kv_heads is None:
        n_kv_heads = n_heads
    hf = {}
    for key, w in meta.items():
        if key.startswith("rope."):
            continue  # precomputed freqs; recomputed at load time
        if key == "tok_embeddings.weight":
            hf["model.embed_tokens.weight"] = w
        elif key == "norm.weight":
            hf["model.norm.weight"] = w
        elif key == "output.weight":
            hf["lm_head.weight"] = w
        else:
            m = re.match(r"layers\.(\d+)\.(.+)\.weight$", key)
            if m is None:
                raise KeyError(f"unrecognized Meta checkpoint key: {key}")
            i, frag = m.group(1), m.group(2)
            if frag not in _LAYER_MAP:
                raise KeyError(f"unrecognized Meta layer fragment: {frag}")
            if frag == "attention.wq":
                w = _rows_interleaved_to_half_split(w, n_heads)
            elif frag == "attention.wk":
                w = _rows_interleaved_to_half_split(w, n_kv_heads)
            hf[f"model.layers.{i}.{_LAYER_MAP[frag]}.weight"] = w
    return hf


def merge_meta_llama(root_dir: str, n_heads: int,
                     n_kv_heads: int | None = None) -> dict:
    """Load + merge all consolidated.XX.pth shards under root_dir and return
    an HF-style state dict."""
    paths = sorted(
        p for p in glob.glob(os.path.join(root_dir, "consolidated.*.pth"))
        if re.match(r"^consolidated\.\d+\.pth$", os.path.basename(p))
    )
    if not paths:
        raise FileNotFoundError(
            f"no consolidated.*.pth shards found in {root_dir}"
        )
    shards = [torch.load(p, map_location="cpu", weights_only=True)
              for p in paths]
    return meta_to_hf_keys(merge_shards(shards), n_heads, n_kv_heads)
