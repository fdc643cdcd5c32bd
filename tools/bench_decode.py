"""Serving decode benchmark: tokens/s of single-stream greedy decode on a
random-init model, eager KV-cache path vs hipGraph-captured replay
(megatron_amd/inference/forward_step.py).

  python tools/bench_decode.py [--model llama2-7b] [--tokens 128] [--batch 1]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama2-7b")
    p.add_argument("--tokens", type=int, default=128)
    p.add_argument("--batch", type=int, default=1)
    p.add_argument("--prompt", type=int, default=32)
    args = p.parse_args()

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29671")
    import torch.distributed as dist

    dist.init_process_group(
        "nccl" if torch.cuda.is_available() else "gloo", rank=0, world_size=1
    )
    from megatron_amd import global_state, parallel as mpu
    from megatron_amd.config import TrainingConfig, set_config
    from megatron_amd.inference.generation import (
        generate_tokens_probs_and_return_on_first_stage,
    )
    from megatron_amd.models import MODEL_CLASSES
    from megatron_amd.tokenizer.tokenizers import FakeTokenizer

    mpu.initialize_model_parallel(1, 1)
    mpu.model_parallel_cuda_manual_seed(1234)

    from bench import MODEL_SPECS

    spec = MODEL_SPECS[args.model]
    total = args.prompt + args.tokens
    cfg = TrainingConfig(
        num_layers=spec["num_layers"], hidden_size=spec["hidden_size"],
        ffn_hidden_size=spec["ffn_hidden_size"],
        num_attention_heads=spec["num_attention_heads"],
        num_attention_heads_kv=spec["num_attention_heads_kv"],
        seq_length=total, max_position_embeddings=max(total, 4096),
        micro_batch_size=args.batch, hidden_dropout=0.0,
        attention_dropout=0.0, bf16=torch.cuda.is_available(),
        use_flash_attn=True, model_name=spec["model_name"],
    )
    cfg.finalize()
    cfg.pad_vocab_size(spec["vocab"])
    set_config(cfg)
    global_state.init_timers()
    global_state.set_tokenizer(FakeTokenizer(spec["vocab"]))

    model_cls = MODEL_CLASSES[cfg.model_name]
    model = model_cls(cfg, parallel_output=False)
    if torch.cuda.is_available():
        model = model.cuda().bfloat16()
    model.eval()

    def run(use_graph):
        cfg.use_hip_graph_decode = use_graph
        torch.manual_seed(3)
        tokens = torch.zeros(args.batch, total, dtype=torch.long,
                             device="cuda" if torch.cuda.is_available()
                             else "cpu")
        tokens[:, : args.prompt] = torch.randint(
            1, spec["vocab"] - 1, (args.batch, args.prompt),
            device=tokens.device,
        )
        lengths = torch.full((args.batch,), args.prompt,
                             device=tokens.device)
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        t0 = time.time()
        generate_tokens_probs_and_return_on_first_stage(
            model, tokens, lengths, top_k=1,
            use_eod_token_for_early_termination=False,
        )
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        dt = time.time() - t0
        tps = args.tokens * args.batch / dt
        print(f"{'graph' if use_graph else 'eager'}: {dt:.3f}s "
              f"{tps:.1f} tokens/s", flush=True)
        return tps

    eager = run(False)
    graph = run(True)
    graph2 = run(True)  # second run: capture cost amortized away
    print(f"speedup (captured): {graph2 / eager:.2f}x", flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
