"""Convert a megatron_amd checkpoint back to HuggingFace format.

Reference behavior: weights_conversion/megatron_to_hf.py:47-476
(convert_wqkv / convert_ffn inverses, safetensors output + config).
"""

from __future__ import annotations

import argparse
import json
import os
import sys
from pathlib import Path

import torch

sys.path.append(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from weights_conversion.permute_qkv import permute_qkv  # noqa: E402


def load_megatron_checkpoint(input_dir: str):
    inp = Path(input_dir)
    tracker = inp / "latest_checkpointed_iteration.txt"
    with open(tracker) as f:
        meta = f.read().strip()
    sub = "release" if meta == "release" else f"iter_{int(meta):07d}"
    ckpt = inp / sub / "mp_rank_00" / "model_optim_rng.pt"
    state = torch.load(ckpt, map_location="cpu", weights_only=False)
    model = state["model"]
    if "language_model" in model:
        model = model["language_model"]
    return state, model


def convert_wqkv(qkv_w, n_heads, n_heads_kv, hidden):
    """Split the fused [q...,k,v]-per-group QKV back to separate HF q/k/v
    (reference megatron_to_hf.py:47-66)."""
    head_dim = hidden // n_heads
    qkv_w = permute_qkv(qkv_w, hidden, n_heads, n_heads_kv, revert=True)
    n_qs_per_kv = n_heads // n_heads_kv
    n_groups = qkv_w.size(0) // head_dim // (n_qs_per_kv + 2)
    qkv_w = list(torch.split(qkv_w, head_dim, dim=0))

    wq, wk, wv = [], [], []
    for _ in range(n_groups):
        for qs in range(n_qs_per_kv):
            wq.append(qkv_w[0])
            del qkv_w[0]
        wk.append(qkv_w[0])
        del qkv_w[0]
        wv.append(qkv_w[0])
        del qkv_w[0]
    assert len(qkv_w) == 0
    return torch.cat(wq, dim=0), torch.cat(wk, dim=0), torch.cat(wv, dim=0)


def convert_ffn(ffn_w, ffn_hidden):
    """Split dense_h_to_4h back into up (w3) and gate (w1)
    (reference megatron_to_hf.py:68-78)."""
    up, gate = torch.split(ffn_w, ffn_hidden, dim=0)
    return up, gate


def megatron_to_hf_llama(model_sd: dict, num_layers, hidden, n_heads,
                         n_heads_kv, ffn_hidden, vocab_size):
    out = {}
    out["model.embed_tokens.weight"] = model_sd[
        "embedding.word_embeddings.weight"
    ][:vocab_size]
    out["model.norm.weight"] = model_sd["encoder.final_layernorm.weight"]
    out["lm_head.weight"] = model_sd["lm_head"][:vocab_size]
    for i in range(num_layers):
        p = f"encoder.layers.{i}"
        o = f"model.layers.{i}"
        out[f"{o}.input_layernorm.weight"] = model_sd[
            f"{p}.input_layernorm.weight"
        ]
        out[f"{o}.post_attention_layernorm.weight"] = model_sd[
            f"{p}.post_attention_layernorm.weight"
        ]
        wq, wk, wv = convert_wqkv(
            model_sd[f"{p}.self_attention.query_key_value.weight"],
            n_heads, n_heads_kv, hidden,
        )
        out[f"{o}.self_attn.q_proj.weight"] = wq
        out[f"{o}.self_attn.k_proj.weight"] = wk
        out[f"{o}.self_attn.v_proj.weight"] = wv
        out[f"{o}.self_attn.o_proj.weight"] = model_sd[
            f"{p}.self_attention.dense.weight"
        ]
        up, gate = convert_ffn(
            model_sd[f"{p}.mlp.dense_h_to_4h.weight"], ffn_hidden
        )
        out[f"{o}.mlp.up_proj.weight"] = up
        out[f"{o}.mlp.gate_proj.weight"] = gate
        out[f"{o}.mlp.down_proj.weight"] = model_sd[
            f"{p}.mlp.dense_4h_to_h.weight"
        ]
    return out


def megatron_to_hf_falcon(model_sd: dict, num_layers, hidden, n_heads,
                          n_heads_kv, vocab_size):
    """Inverse of hf_to_megatron.falcon_to_megatron (reference
    megatron_to_hf.py:333-434): undo the interleaved->rotate_half rotary row
    permutation on q/k and restore the HF key names."""
    from weights_conversion.hf_to_megatron import _permute_rotary_rows

    def unpermute(w, heads):
        total, cols = w.shape
        d = total // heads
        wv = w.view(heads, d, cols)
        half = d // 2
        inv = torch.empty(d, dtype=torch.long)
        inv[:half] = torch.arange(0, d, 2)
        inv[half:] = torch.arange(1, d, 2)
        return wv[:, inv, :].reshape(total, cols)

    out = {}
    out["transformer.word_embeddings.weight"] = model_sd[
        "embedding.word_embeddings.weight"
    ][:vocab_size]
    out["transformer.ln_f.weight"] = model_sd["encoder.final_layernorm.weight"]
    out["transformer.ln_f.bias"] = model_sd["encoder.final_layernorm.bias"]
    out["lm_head.weight"] = out["transformer.word_embeddings.weight"]
    hd = hidden // n_heads
    nq = n_heads // n_heads_kv
    for i in range(num_layers):
        p = f"encoder.layers.{i}"
        o = f"transformer.h.{i}"
        if f"{p}.mlp_layernorm.weight" in model_sd:  # 40B parallel-LN
            out[f"{o}.ln_attn.weight"] = model_sd[f"{p}.input_layernorm.weight"]
            out[f"{o}.ln_attn.bias"] = model_sd[f"{p}.input_layernorm.bias"]
            out[f"{o}.ln_mlp.weight"] = model_sd[f"{p}.mlp_layernorm.weight"]
            out[f"{o}.ln_mlp.bias"] = model_sd[f"{p}.mlp_layernorm.bias"]
        else:
            out[f"{o}.input_layernorm.weight"] = model_sd[
                f"{p}.input_layernorm.weight"
            ]
            out[f"{o}.input_layernorm.bias"] = model_sd[
                f"{p}.input_layernorm.bias"
            ]
        qkv = model_sd[f"{p}.self_attention.query_key_value.weight"]
        n_rows, cols = qkv.shape
        groups = n_heads_kv
        qkv = qkv.view(groups, (nq + 2) * hd, cols)
        q = unpermute(qkv[:, : nq * hd, :].reshape(groups * nq * hd, cols),
                      groups * nq)
        k = unpermute(
            qkv[:, nq * hd : (nq + 1) * hd, :].reshape(groups * hd, cols),
            groups,
        )
        v = qkv[:, (nq + 1) * hd :, :].reshape(groups * hd, cols)
        out[f"{o}.self_attention.query_key_value.weight"] = torch.cat(
            [
                q.view(groups, nq * hd, cols),
                k.view(groups, hd, cols),
                v.view(groups, hd, cols),
            ],
            dim=1,
        ).reshape(n_rows, cols)
        out[f"{o}.self_attention.dense.weight"] = model_sd[
            f"{p}.self_attention.dense.weight"
        ]
        out[f"{o}.mlp.dense_h_to_4h.weight"] = model_sd[
            f"{p}.mlp.dense_h_to_4h.weight"
        ]
        out[f"{o}.mlp.dense_4h_to_h.weight"] = model_sd[
            f"{p}.mlp.dense_4h_to_h.weight"
        ]
    return out


def write_hf_checkpoint(hf_sd: dict, out_dir: str, config: dict,
                        use_safetensors=True):
    out = Path(out_dir)
    out.mkdir(parents=True, exist_ok=True)
    if use_safetensors:
        try:
            from safetensors.torch import save_file

            hf_sd = {k: v.contiguous() for k, v in hf_sd.items()}
            save_file(hf_sd, str(out / "model.safetensors"),
                      metadata={"format": "pt"})
        except ImportError:
            torch.save(hf_sd, out / "pytorch_model.bin")
    else:
        torch.save(hf_sd, out / "pytorch_model.bin")
    with open(out / "config.json", "w") as f:
        json.dump(config, f, indent=2)
    print(f"saved HF checkpoint to {out}")


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--input_dir", required=True)
    parser.add_argument("--output_dir", required=True)
    parser.add_argument("--model", default="llama2",
                        choices=["llama", "llama2", "codellama", "mistral"])
    parser.add_argument("--vocab_size", type=int, default=32000)
    args = parser.parse_args()

    state, model_sd = load_megatron_checkpoint(args.input_dir)
    margs = state.get("args")
    num_layers = getattr(margs, "num_layers")
    hidden = getattr(margs, "hidden_size")
    n_heads = getattr(margs, "num_attention_heads")
    n_heads_kv = getattr(margs, "num_attention_heads_kv", n_heads) or n_heads
    ffn_hidden = getattr(margs, "ffn_hidden_size")

    hf_sd = megatron_to_hf_llama(
        model_sd, num_layers, hidden, n_heads, n_heads_kv, ffn_hidden,
        args.vocab_size,
    )
    arch = (
        "MistralForCausalLM" if args.model == "mistral" else "LlamaForCausalLM"
    )
    config = {
        "architectures": [arch],
        "hidden_size": hidden,
        "intermediate_size": ffn_hidden,
        "num_attention_heads": n_heads,
        "num_key_value_heads": n_heads_kv,
        "num_hidden_layers": num_layers,
        "vocab_size": args.vocab_size,
        "rms_norm_eps": 1e-5,
        "torch_dtype": "bfloat16",
        "model_type": "mistral" if args.model == "mistral" else "llama",
    }
    if args.model == "mistral":
        config["sliding_window"] = 4096
    write_hf_checkpoint(hf_sd, args.output_dir, config)


if __name__ == "__main__":
    main()
