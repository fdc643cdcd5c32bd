"""Benchmark driver — the contract entry point.

`python bench.py --gpus N --steps K --warmup W` runs the flagship training
step (Llama-2-7B bf16 seq4096, synthetic data, random-init weights) on N GPUs
of one node and prints ONE JSON line from rank 0 with the whole-job aggregate
tokens/s (the BASELINE.json metric).

Launched either directly (N=1) or via torch.distributed.run with one rank per
GPU over RCCL. Parallelism by N: 1 GPU -> TP1/DP1; N<=8 -> DP=N (TP1) unless
--tp/--pp given.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

# avoid caching-allocator fragmentation at the 288 GB capacity edge (the
# 32k-seq configs sit within ~1% of it)
os.environ.setdefault("PYTORCH_ALLOC_CONF", "expandable_segments:True")

# hipBLASLt TunableOp: use the pre-tuned GEMM algo selection committed under
# profiles/ when present (tools/tune_gemms.py produces it); read-only mode.
_TUNED = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                      "profiles", "tunableop_results0.csv")
if os.path.exists(_TUNED) and "PYTORCH_TUNABLEOP_ENABLED" not in os.environ:
    os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
    os.environ["PYTORCH_TUNABLEOP_TUNING"] = "0"
    os.environ["PYTORCH_TUNABLEOP_FILENAME"] = os.path.join(
        os.path.dirname(_TUNED), "tunableop_results.csv"
    )

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=8)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--model", type=str, default="llama2-7b",
                   choices=["llama2-7b", "llama2-70b", "llama2-70b-shard8",
                            "mistral-7b", "falcon-7b", "gpt-125m",
                            "llama2-tiny"])
    p.add_argument("--seq-len", type=int, default=None)
    p.add_argument("--micro-batch-size", type=int, default=None)
    p.add_argument("--global-batch", type=int, default=None)
    p.add_argument("--tp", type=int, default=None)
    p.add_argument("--pp", type=int, default=1)
    p.add_argument("--layers", type=int, default=None,
                   help="override layer count (shape validation only; "
                   "results with this flag are not official numbers)")
    p.add_argument("--recompute", action="store_true")
    p.add_argument("--no-grad-accum-fusion", action="store_true",
                   help="autograd bf16 wgrad + fp32 hook accumulate (the "
                   "reference's non-apex path) instead of the fused "
                   "fp32-accum wgrad GEMM")
    p.add_argument("--dtype", type=str, default="bf16",
                   choices=["bf16", "fp16", "fp8"])
    p.add_argument("--fp8-wgrad", action="store_true",
                   help="with --dtype fp8: run the wgrad GEMM in e4m3 too")
    p.add_argument("--loss-chunk-size", type=int, default=None,
                   help="override the chunked LM-head loss size (0 = off)")
    return p.parse_args()


MODEL_SPECS = {
    # Llama-2-7B (HF config): 32 layers, h=4096, ffn=11008, 32 heads
    "llama2-7b": dict(num_layers=32, hidden_size=4096, ffn_hidden_size=11008,
                      num_attention_heads=32, num_attention_heads_kv=32,
                      vocab=32000, seq=4096, model_name="llama2"),
    # Llama-2-70B: 80 layers, h=8192, ffn=28672, 64 heads, 8 kv heads (GQA)
    "llama2-70b": dict(num_layers=80, hidden_size=8192, ffn_hidden_size=28672,
                       num_attention_heads=64, num_attention_heads_kv=8,
                       vocab=32000, seq=4096, model_name="llama2"),
    # Mistral-7B: 32 layers, h=4096, ffn=14336, 32 heads, 8 kv, SWA 4096.
    # chunked LM-head loss (loss_chunk_size) bounds the fp32 logits peak at
    # seq 32k, reopening micro-batch 2 on 288 GB (round-1 ran mbs1)
    "mistral-7b": dict(num_layers=32, hidden_size=4096, ffn_hidden_size=14336,
                       num_attention_heads=32, num_attention_heads_kv=8,
                       vocab=32000, seq=32768, model_name="mistral",
                       sliding_window_size=4096, rope_scaling_factor=4.0,
                       mbs=1, loss_chunk_size=8192),
    # Falcon-7B: 32 layers, h=4544, 71 heads, MQA (1 kv head), parallel attn
    "falcon-7b": dict(num_layers=32, hidden_size=4544, ffn_hidden_size=18176,
                      num_attention_heads=71, num_attention_heads_kv=1,
                      vocab=65024, seq=2048, model_name="falcon"),
    # one TP4xPP2 rank's shard of Llama-2-70B: 40 layers (one PP stage),
    # heads/kv-heads/ffn divided by tp=4, hidden kept full (TP shards only
    # the head and ffn dimensions) — single-GPU memory/timing proxy for the
    # per-rank footprint of the 8-GPU 70B config (VERDICT #3 validation)
    "llama2-70b-shard8": dict(num_layers=40, hidden_size=8192,
                              ffn_hidden_size=7168, num_attention_heads=16,
                              num_attention_heads_kv=2, vocab=32000,
                              seq=4096, model_name="llama2",
                              kv_channels=128),
    "gpt-125m": dict(num_layers=12, hidden_size=768, ffn_hidden_size=3072,
                     num_attention_heads=12, num_attention_heads_kv=12,
                     vocab=50304, seq=1024, model_name="gpt"),
    "llama2-tiny": dict(num_layers=4, hidden_size=512, ffn_hidden_size=1376,
                        num_attention_heads=8, num_attention_heads_kv=8,
                        vocab=32000, seq=512, model_name="llama2"),
}


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", "0"))
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)

    have_gpu = torch.cuda.is_available()
    spec = dict(MODEL_SPECS[args.model])
    if not have_gpu and args.model in ("llama2-7b", "llama2-70b",
                                       "llama2-70b-shard8", "mistral-7b",
                                       "falcon-7b"):
        # CPU plumbing check shrinks the model but keeps the code path
        spec = dict(MODEL_SPECS["llama2-tiny"])
        spec["model_name"] = MODEL_SPECS[args.model]["model_name"]

    if args.layers:
        spec["num_layers"] = args.layers
    seq = args.seq_len or spec["seq"]
    tp = args.tp if args.tp is not None else 1
    pp = args.pp
    assert world_size % (tp * pp) == 0
    dp = world_size // (tp * pp)
    # mbs sweep on MI355X (profiles/r01): 15.5k @ mbs4, 16.7k @ mbs8 with
    # the wide-workgroup FA kernels
    mbs = args.micro_batch_size or spec.get("mbs") or (8 if have_gpu else 1)
    if args.dtype == "fp8" and args.micro_batch_size is None and have_gpu:
        # fp8's weight copies + quantized activations don't fit mbs8 x 2
        # accumulation microbatches in 288 GB
        mbs = 6
    # pp > 1 needs several in-flight microbatches to fill the 1F1B pipeline;
    # at pp == 1, 2 accumulation microbatches amortize the optimizer step
    # (measured +1.5% tokens/s at 7B mbs8, gpurun_out/r2_8.log)
    gbs = args.global_batch or (mbs * dp * (2 * pp if pp > 1 else 2))

    dtype_flags = {}
    if args.dtype == "bf16" and have_gpu:
        dtype_flags["bf16"] = True
    elif args.dtype == "fp16" and have_gpu:
        dtype_flags["fp16"] = True
    elif args.dtype == "fp8" and have_gpu:
        # fp8 fwd/dgrad GEMMs over bf16 params (megatron_amd/fp8.py) — a
        # separate recipe, never the headline bf16 number
        dtype_flags["bf16"] = True
        dtype_flags["fp8"] = True
        if args.fp8_wgrad:
            dtype_flags["fp8_wgrad"] = True

    from megatron_amd.config import TrainingConfig, set_config
    from megatron_amd.initialize import initialize_megatron

    cfg = TrainingConfig(
        num_layers=spec["num_layers"], hidden_size=spec["hidden_size"],
        ffn_hidden_size=spec["ffn_hidden_size"],
        num_attention_heads=spec["num_attention_heads"],
        num_attention_heads_kv=spec["num_attention_heads_kv"],
        kv_channels=spec.get("kv_channels"),
        max_position_embeddings=max(seq, 4096),
        seq_length=seq, micro_batch_size=mbs, global_batch_size=gbs,
        tensor_model_parallel_size=tp, pipeline_model_parallel_size=pp,
        train_iters=args.steps + args.warmup + 1,
        lr=1e-5, min_lr=1e-6, lr_decay_style="constant",
        lr_warmup_iters=0, clip_grad=1.0,
        hidden_dropout=0.0, attention_dropout=0.0,
        use_flash_attn=True,
        gradient_accumulation_fusion=not args.no_grad_accum_fusion,
        recompute_granularity=(
            "full" if (args.recompute or spec.get("recompute"))
            else "selective"
        ),
        recompute_method=(
            "uniform" if (args.recompute or spec.get("recompute")) else None
        ),
        sequence_parallel=(tp > 1),
        # bucketed async all-reduce hidden behind backward beats ZeRO-1's
        # exposed whole-buffer reduce-scatter + all-gather here: 288 GB HBM3E
        # leaves no memory pressure at 7B, so spend memory for overlap
        overlap_grad_reduce=(dp > 1),
        model_name=spec["model_name"],
        sliding_window_size=spec.get("sliding_window_size"),
        rope_scaling_factor=spec.get("rope_scaling_factor", 1.0),
        loss_chunk_size=(args.loss_chunk_size
                         if args.loss_chunk_size is not None
                         else spec.get("loss_chunk_size", 0)),
        rank=rank, world_size=world_size, local_rank=local_rank,
        **dtype_flags,
    )
    cfg.finalize()
    cfg.pad_vocab_size(spec["vocab"])
    set_config(cfg)

    initialize_megatron(cfg=cfg)

    import megatron_amd.parallel as mpu
    from megatron_amd.models import MODEL_CLASSES, ModelType
    from megatron_amd.training import get_model, train_step
    from megatron_amd.optim import (
        get_megatron_optimizer, get_optimizer_param_scheduler,
    )
    from megatron_amd.utils import get_ltor_masks_and_position_ids
    import functools

    def model_provider(pre_process=True, post_process=True):
        model_cls = MODEL_CLASSES[cfg.model_name]
        return model_cls(cfg, parallel_output=True, pre_process=pre_process,
                         post_process=post_process)

    model = get_model(model_provider, ModelType.encoder_or_decoder, cfg=cfg)
    optimizer = get_megatron_optimizer(model, cfg)
    opt_sched = get_optimizer_param_scheduler(optimizer, cfg)

    device = torch.device("cuda", local_rank) if have_gpu else torch.device("cpu")
    vocab = cfg.padded_vocab_size

    def synthetic_batch_iterator():
        g = torch.Generator(device="cpu").manual_seed(1234 + rank)
        while True:
            tokens = torch.randint(
                0, spec["vocab"], (mbs, seq + 1), generator=g,
                dtype=torch.int64,
            ).to(device)
            yield {"text": tokens}

    data_iter = synthetic_batch_iterator()

    def forward_step_func(data_iterator, model):
        data = next(data_iterator)
        tokens_ = data["text"]
        tokens = tokens_[:, :-1].contiguous()
        labels = tokens_[:, 1:].contiguous()
        attention_mask, loss_mask, position_ids = (
            get_ltor_masks_and_position_ids(tokens, 0, False, False, False)
        )
        output_tensor = model(tokens, position_ids, None, labels=labels)

        def loss_func(loss_mask, output_tensor):
            losses = output_tensor.float()
            loss_mask_ = loss_mask.view(-1).float()
            loss = torch.sum(losses.view(-1) * loss_mask_) / loss_mask_.sum()
            return loss, {"lm loss": loss.detach()}

        return output_tensor, functools.partial(loss_func, loss_mask)

    def barrier_sync():
        if torch.distributed.is_initialized() and world_size > 1:
            torch.distributed.barrier()
        if have_gpu:
            torch.cuda.synchronize()

    # warmup
    for _ in range(args.warmup):
        train_step(forward_step_func, data_iter, model, optimizer, opt_sched,
                   cfg)

    barrier_sync()
    t0 = time.time()
    if os.environ.get("MEGATRON_AMD_TORCH_PROF"):
        # diagnostic mode: attribute kernels to aten ops (not for timing)
        from torch.profiler import ProfilerActivity, profile

        with profile(activities=[ProfilerActivity.CUDA,
                                 ProfilerActivity.CPU]) as prof:
            for _ in range(args.steps):
                train_step(forward_step_func, data_iter, model, optimizer,
                           opt_sched, cfg)
            barrier_sync()
        print(prof.key_averages().table(
            sort_by="self_cuda_time_total", row_limit=30,
        ), file=sys.stderr)
    else:
        for i in range(args.steps):
            ld = train_step(forward_step_func, data_iter, model, optimizer,
                            opt_sched, cfg)
            if os.environ.get("MEGATRON_AMD_PRINT_LOSS") and rank == 0:
                # convergence evidence mode (adds a sync per step — not for
                # timing-quality numbers)
                try:
                    loss = ld[0]["lm loss"].item()
                    print(f"step {i}: lm loss {loss:.4f}", file=sys.stderr)
                except Exception:
                    pass
        barrier_sync()
    elapsed = time.time() - t0

    # MAX over ranks
    elapsed_t = torch.tensor(
        [elapsed], dtype=torch.float64,
        device=device if have_gpu else "cpu",
    )
    if torch.distributed.is_initialized() and world_size > 1:
        torch.distributed.all_reduce(elapsed_t,
                                     op=torch.distributed.ReduceOp.MAX)
    elapsed = elapsed_t.item()

    if have_gpu:
        print(f"peak allocated {torch.cuda.max_memory_allocated() / 2**30:.1f}"
              f" GiB / reserved "
              f"{torch.cuda.max_memory_reserved() / 2**30:.1f} GiB",
              file=sys.stderr)
    ms_per_step = elapsed / args.steps * 1000.0
    tokens_per_step = gbs * seq
    tokens_per_sec = tokens_per_step / (elapsed / args.steps)

    baseline = 7100.0  # tokens/s/node, 8xA100, Llama-2-7B seq1024

    if rank == 0:
        # metric names ONLY what this invocation measured (the round-1
        # verdict flagged the old string for naming the unmeasured 70B
        # TP4xPP2 half); `--model llama2-70b --tp 4 --pp 2` on an 8-GPU node
        # measures that config and names it accordingly
        pretty = {"llama2-7b": "Llama-2-7B", "llama2-70b": "Llama-2-70B",
                  "mistral-7b": "Mistral-7B", "falcon-7b": "Falcon-7B"}
        metric = (f"tokens/sec (node) {pretty.get(args.model, args.model)} "
                  f"{args.dtype} seq{seq}")
        if tp * pp > 1:
            metric += f" tp{tp}xpp{pp}"
        result = {
            "metric": metric,
            "value": round(tokens_per_sec, 1),
            "unit": "tokens/s",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": round(tokens_per_sec / baseline, 4)
            if have_gpu and args.model == "llama2-7b" else None,
            "dtype": (args.dtype if have_gpu else "fp32"),
            "data": "synthetic",
            "config": {
                "model": (args.model if have_gpu else f"{args.model}(cpu-tiny)")
                + (f"-L{args.layers}" if args.layers else ""),
                "global_batch": gbs,
                "seq_len": seq,
                "parallelism": f"tp{tp}_pp{pp}_dp{dp}",
            },
        }
    else:
        result = None

    # tear down BEFORE printing: RCCL writes a version banner to stdout at
    # communicator destruction, which must not land after the JSON line
    if torch.distributed.is_initialized():
        torch.distributed.destroy_process_group()
    if result is not None:
        sys.stdout.flush()
        print(json.dumps(result), flush=True)
    # exit without running C-level exit handlers: RCCL/HIP flush version
    # banners to stdout at library unload, which would land after the JSON
    # contract line. rocprofv3 needs its exit handlers to write the trace —
    # set MEGATRON_AMD_CLEAN_EXIT=0 when profiling.
    sys.stdout.flush()
    sys.stderr.flush()
    if os.environ.get("MEGATRON_AMD_CLEAN_EXIT", "1") != "0":
        os._exit(0)


if __name__ == "__main__":
    main()
