"""Run the megatron_amd model and a HuggingFace reference side-by-side on
identical batches, reporting per-iteration max/avg abs logit error and loss
delta (reference verify_correctness.py:105-190).

Offline-friendly: --hf_cache_dir points at a local HF checkpoint directory;
with --random_init a random-weight HF model of the same config is used (the
conversion path is then exercised via in-memory conversion).

  torchrun --nproc_per_node 1 verify_correctness.py --model_name llama2 \
      --load ckpt_dir --hf_cache_dir hf_dir --tokenizer_type ... \
      --micro_batch_size 1 --seq_length 512 ...

Documented tolerances (reference docs/guide/getting_started.md:153-154):
max abs logit error <= 0.01 avg at fp32, <= 0.1 bf16.
"""

from __future__ import annotations


import torch

from megatron_amd.checkpointing import load_checkpoint
from megatron_amd.initialize import initialize_megatron
from megatron_amd.models import MODEL_CLASSES, ModelType
from megatron_amd.training import get_model
from megatron_amd.utils import get_ltor_masks_and_position_ids, print_rank_0


def extra_args(parser):
    group = parser.add_argument_group("verify")
    group.add_argument("--hf_cache_dir", type=str, default=None)
    group.add_argument("--random_init", action="store_true")
    group.add_argument("--iters", type=int, default=10)
    return parser


def build_hf_model(cfg, hf_cache_dir, random_init):
    import transformers

    if random_init or hf_cache_dir is None:
        config_cls = (
            transformers.MistralConfig
            if cfg.model_name == "mistral" else transformers.LlamaConfig
        )
        hf_cfg = config_cls(
            vocab_size=cfg.padded_vocab_size,
            hidden_size=cfg.hidden_size,
            intermediate_size=cfg.ffn_hidden_size,
            num_hidden_layers=cfg.num_layers,
            num_attention_heads=cfg.num_attention_heads,
            num_key_value_heads=cfg.num_attention_heads_kv,
            max_position_embeddings=cfg.max_position_embeddings,
            rms_norm_eps=cfg.layernorm_epsilon,
            tie_word_embeddings=False,
            rope_theta=cfg.rope_theta,
        )
        model_cls = (
            transformers.MistralForCausalLM
            if cfg.model_name == "mistral"
            else transformers.LlamaForCausalLM
        )
        return model_cls(hf_cfg)
    return transformers.AutoModelForCausalLM.from_pretrained(
        hf_cache_dir, torch_dtype=torch.float32
    )


def main():
    cfg = initialize_megatron(extra_args_provider=extra_args,
                              args_defaults={"tokenizer_type": "FakeTokenizer"})
    device = "cuda" if torch.cuda.is_available() else "cpu"

    def model_provider(pre_process=True, post_process=True):
        model_cls = MODEL_CLASSES[cfg.model_name or "llama2"]
        return model_cls(cfg, parallel_output=False,
                         pre_process=pre_process, post_process=post_process)

    our = get_model(model_provider, ModelType.encoder_or_decoder,
                    wrap_with_ddp=False)
    if cfg.load:
        load_checkpoint(our, None, None, cfg)
    our = our[0]
    our.eval()

    hf = build_hf_model(cfg, getattr(cfg, "hf_cache_dir", None),
                        getattr(cfg, "random_init", False))
    hf = hf.to(device).eval()
    if getattr(cfg, "random_init", False) and not cfg.load:
        # convert hf weights into our model so both match
        from weights_conversion.hf_to_megatron import (
            llama_like_to_megatron, pad_embeddings,
        )

        sd = llama_like_to_megatron(
            hf.state_dict(), cfg.num_layers, cfg.hidden_size,
            cfg.num_attention_heads, cfg.num_attention_heads_kv,
        )
        sd = pad_embeddings(sd, cfg.make_vocab_size_divisible_by)
        sd = {k: v.to(cfg.params_dtype) for k, v in sd.items()}
        our_unwrapped = our
        from megatron_amd.utils import unwrap_model

        our_unwrapped = unwrap_model(our)
        our_unwrapped.language_model.load_state_dict(sd, strict=False)

    torch.manual_seed(4321)
    seq = cfg.seq_length or 512
    iters = getattr(cfg, "iters", 10)
    total_max, total_abs = 0.0, 0.0
    for it in range(iters):
        tokens = torch.randint(
            0, min(32000, cfg.padded_vocab_size), (cfg.micro_batch_size,
                                                   seq), device=device
        )
        am, _, pids = get_ltor_masks_and_position_ids(tokens, 0, False,
                                                      False, False)
        with torch.no_grad():
            ours_logits = our(tokens, pids, am).float()
            hf_logits = hf(tokens).logits.float()
        v = min(ours_logits.shape[-1], hf_logits.shape[-1])
        diff = (ours_logits[..., :v] - hf_logits[..., :v]).abs()
        max_err = diff.max().item()
        abs_err = diff.mean().item()
        total_max += max_err
        total_abs += abs_err
        print_rank_0(
            f"iter {it}: max abs logit error {max_err:.2e}, "
            f"avg abs {abs_err:.2e}"
        )
    print_rank_0(
        f"AVG over {iters} iters: max {total_max / iters:.2e}, "
        f"abs {total_abs / iters:.2e}"
    )


if __name__ == "__main__":
    main()
