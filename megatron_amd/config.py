"""Explicit configuration object (replaces the reference's global argparse namespace).

The reference (megatron/arguments.py:15-1103, megatron/global_vars.py:14-21) funnels
all configuration through a process-global argparse.Namespace reached via get_args().
Here configuration is an explicit dataclass threaded through the runtime; a
module-level "current config" exists only for CLI-driven flows (finetune.py) and the
few leaf call sites where threading it would be pure noise. Flag names match the
reference's underscore style (--micro_batch_size etc.) so launch scripts carry over.
"""

from __future__ import annotations

import argparse
import dataclasses
from dataclasses import dataclass, field
from typing import List, Optional

import torch


@dataclass
class TrainingConfig:
    # -- model architecture (reference arguments.py "network size" group) --
    num_layers: int = 2
    hidden_size: int = 128
    ffn_hidden_size: Optional[int] = None
    num_attention_heads: int = 4
    num_attention_heads_kv: Optional[int] = None  # GQA/MQA: kv heads (None -> = heads)
    kv_channels: Optional[int] = None
    max_position_embeddings: int = 2048
    make_vocab_size_divisible_by: int = 128
    padded_vocab_size: Optional[int] = None
    layernorm_epsilon: float = 1e-5
    apply_residual_connection_post_layernorm: bool = False
    use_bias: bool = True               # bias on linear layers (Llama: False)
    use_rms_norm: bool = False          # RMSNorm instead of LayerNorm
    use_post_ln: bool = False           # post-LN (default pre-LN)
    glu_activation: Optional[str] = None  # None|'liglu'|'geglu'|'reglu'|'swiglu'
    position_embedding_type: str = "absolute"  # 'absolute' | 'rotary'
    rope_theta: float = 10000.0
    rope_scaling_factor: float = 1.0
    tie_embed_logits: bool = True       # tie lm head to word embeddings
    parallel_attn: bool = False         # Falcon parallel attention+MLP
    parallel_layernorm: bool = False    # Falcon-40B separate ln_mlp
    sliding_window_size: Optional[int] = None  # Mistral SWA
    onnx_safe: bool = False
    bert_binary_head: bool = True

    # -- init / numerics --
    init_method_std: float = 0.02
    init_method_xavier_uniform: bool = False
    apply_query_key_layer_scaling: bool = False
    attention_softmax_in_fp32: bool = False
    fp32_residual_connection: bool = False
    fp16: bool = False
    bf16: bool = False
    # fp8 (e4m3) forward/dgrad GEMMs via hipBLASLt _scaled_mm (megatron_amd/
    # fp8.py) — gfx950 fp8 MFMA is 2x bf16. wgrad stays bf16/fp32-accum;
    # params/optimizer/checkpoints unchanged. NOT used for headline bf16
    # numbers; enable with --fp8 for the fp8 recipe.
    fp8: bool = False
    loss_scale: Optional[float] = None
    initial_loss_scale: float = 2 ** 32
    min_loss_scale: float = 1.0
    loss_scale_window: float = 1000
    hysteresis: int = 2
    accumulate_allreduce_grads_in_fp32: bool = True
    fp32_allreduce: bool = False

    # -- dropout / regularization --
    attention_dropout: float = 0.1
    hidden_dropout: float = 0.1
    weight_decay: float = 0.01
    start_weight_decay: Optional[float] = None
    end_weight_decay: Optional[float] = None
    weight_decay_incr_style: str = "constant"
    clip_grad: float = 1.0
    adam_beta1: float = 0.9
    adam_beta2: float = 0.999
    adam_eps: float = 1e-8
    sgd_momentum: float = 0.9
    optimizer: str = "adam"

    # -- learning-rate schedule --
    lr: Optional[float] = None
    lr_decay_style: str = "linear"
    lr_decay_iters: Optional[int] = None
    lr_decay_samples: Optional[int] = None
    lr_warmup_fraction: Optional[float] = None
    lr_warmup_iters: int = 0
    lr_warmup_samples: int = 0
    min_lr: float = 0.0
    override_opt_param_scheduler: bool = False
    use_checkpoint_opt_param_scheduler: bool = False

    # -- batch / schedule --
    micro_batch_size: int = 1
    global_batch_size: Optional[int] = None
    rampup_batch_size: Optional[List[int]] = None
    train_iters: Optional[int] = None
    train_samples: Optional[int] = None
    eval_iters: int = 100
    eval_interval: int = 1000
    exit_interval: Optional[int] = None
    exit_duration_in_mins: Optional[int] = None
    exit_signal_handler: bool = False
    skip_iters: List[int] = field(default_factory=list)
    adlr_autoresume: bool = False
    adlr_autoresume_interval: int = 1000

    # -- downstream tasks (tasks/main.py; reference tasks/main.py:24-48) --
    task: Optional[str] = None
    epochs: int = 0
    pretrained_checkpoint: Optional[str] = None
    train_data: Optional[List[str]] = None
    valid_data: Optional[List[str]] = None
    overlapping_eval: int = 32
    keep_last: bool = False
    strict_lambada: bool = False
    # msdp (reference tasks/msdp/main.py:18-43)
    sample_input_file: Optional[str] = None
    sample_output_file: Optional[str] = None
    prompt_file: Optional[str] = None
    prompt_type: Optional[str] = None
    num_prompt_examples: int = 10
    guess_file: Optional[str] = None
    answer_file: Optional[str] = None
    out_seq_length: int = 100
    api_prompt: bool = False
    megatron_api_url: Optional[str] = None
    # orqa (reference tasks/orqa/, megatron/arguments.py biencoder group)
    qa_data_dev: Optional[str] = None
    qa_data_test: Optional[str] = None
    evidence_data_path: Optional[str] = None
    biencoder_shared_query_context_model: bool = False
    report_topk_accuracies: List[int] = field(default_factory=lambda: [1, 5, 20])
    match: str = "string"
    # hipGraph-captured single-token decode (inference/forward_step.py).
    # Opt-in: one replay replaces the per-token launch storm; enable for
    # serving (tools/run_text_generation_server.py turns it on)
    use_hip_graph_decode: bool = True   # hipGraph-replayed decode (TP1/PP1)
    fp8_wgrad: bool = False  # with fp8: also run the wgrad GEMM in e4m3
    #   (cast-transpose operands, fp32 GEMM output accumulated into
    #   main_grad) — experimental, off by default even under fp8

    # -- parallelism --
    tensor_model_parallel_size: int = 1
    pipeline_model_parallel_size: int = 1
    pipeline_model_parallel_split_rank: Optional[int] = None
    num_layers_per_virtual_pipeline_stage: Optional[int] = None
    virtual_pipeline_model_parallel_size: Optional[int] = None
    sequence_parallel: bool = False
    use_distributed_optimizer: bool = False
    DDP_impl: str = "local"
    use_contiguous_buffers_in_local_ddp: bool = True
    overlap_grad_reduce: bool = False
    # elements (fp32) per overlapped all-reduce bucket; sized for the xGMI
    # per-link bandwidth (160 MB buckets keep the ring busy without delaying
    # the first launch) — the N=2..8 scaling tuning knob
    overlap_bucket_numel: int = 40_000_000
    scatter_gather_tensors_in_pipeline: bool = True
    variable_seq_lengths: bool = False
    no_async_tensor_model_parallel_allreduce: bool = False
    gradient_accumulation_fusion: bool = True
    standalone_embedding_stage: bool = False

    # -- distributed runtime --
    distributed_backend: str = "nccl"
    distributed_timeout_minutes: int = 10
    local_rank: Optional[int] = None
    rank: int = 0
    world_size: int = 1
    data_parallel_size: int = 1
    use_cpu_initialization: bool = False
    perform_initialization: bool = True
    empty_unused_memory_level: int = 0

    # -- activation recompute --
    recompute_granularity: Optional[str] = None   # None|'full'|'selective'
    recompute_method: Optional[str] = None        # 'uniform'|'block'
    recompute_num_layers: int = 1
    distribute_saved_activations: bool = False

    # -- attention implementation --
    use_flash_attn: bool = True
    # chunk the LM-head GEMM + cross entropy over the sequence (0 = off):
    # bounds the fp32 logits transient at long sequence / large vocab
    loss_chunk_size: int = 0
    masked_softmax_fusion: bool = True
    bias_gelu_fusion: bool = True
    bias_dropout_fusion: bool = True

    # -- data --
    data_path: Optional[List[str]] = None
    split: str = "969, 30, 1"
    train_data_path: Optional[List[str]] = None
    valid_data_path: Optional[List[str]] = None
    test_data_path: Optional[List[str]] = None
    seq_length: Optional[int] = None
    encoder_seq_length: Optional[int] = None
    decoder_seq_length: Optional[int] = None
    sample_rate: float = 1.0
    mask_prob: float = 0.15
    short_seq_prob: float = 0.1
    mmap_warmup: bool = False
    num_workers: int = 2
    tokenizer_type: Optional[str] = None
    vocab_file: Optional[str] = None
    merge_file: Optional[str] = None
    vocab_extra_ids: int = 0
    vocab_extra_ids_list: Optional[str] = None
    new_tokens: bool = True
    data_impl: str = "infer"
    reset_position_ids: bool = False
    reset_attention_mask: bool = False
    eod_mask_loss: bool = False
    train_data_exact_num_epochs: Optional[int] = None
    data_sharding: bool = True
    dataloader_type: str = "single"
    scalar_loss_mask: float = 0.0

    # -- checkpointing --
    save: Optional[str] = None
    save_interval: Optional[int] = None
    no_save_optim: bool = False
    no_save_rng: bool = False
    load: Optional[str] = None
    no_load_optim: bool = False
    no_load_rng: bool = False
    finetune: bool = False
    use_checkpoint_args: bool = False
    exit_on_missing_checkpoint: bool = False
    load_iters: Optional[int] = None

    # -- logging / metrics --
    log_interval: int = 100
    log_params_norm: bool = False
    log_num_zeros_in_grad: bool = False
    log_timers_to_tensorboard: bool = False
    log_memory_to_tensorboard: bool = False
    log_validation_ppl_to_tensorboard: bool = False
    timing_log_level: int = 0
    timing_log_option: str = "minmax"
    barrier_with_L1_time: bool = True
    tensorboard_dir: Optional[str] = None
    tensorboard_log_interval: int = 1
    tensorboard_queue_size: int = 1000
    wandb_logger: bool = False
    wandb_project: Optional[str] = None
    wandb_entity: Optional[str] = None
    wandb_name: Optional[str] = None
    wandb_id: Optional[str] = None
    wandb_api_key: Optional[str] = None
    metrics: List[str] = field(default_factory=list)

    # -- inference --
    inference_batch_times_seqlen_threshold: int = 512
    max_tokens_to_oom: int = 12000

    # -- misc --
    seed: int = 1234
    data_parallel_random_init: bool = False
    init_method: Optional[str] = None
    model_name: Optional[str] = None   # 'llama'|'llama2'|'codellama'|'falcon'|'mistral'|'gpt'
    model_type: Optional[str] = None
    num_layers_per_stage: Optional[List[int]] = None

    # derived (filled by finalize()):
    params_dtype: torch.dtype = torch.float32
    consumed_train_samples: int = 0
    consumed_valid_samples: int = 0
    curriculum_learning: bool = False
    iteration: int = 0
    do_train: bool = True
    do_valid: bool = False
    do_test: bool = False

    # ------------------------------------------------------------------
    def finalize(self) -> "TrainingConfig":
        """Derive dependent fields and enforce invariants.

        Mirrors the semantics of the reference's validate_args
        (megatron/arguments.py:53-350) without the argparse coupling.
        """
        assert not (self.fp16 and self.bf16), "fp16 and bf16 are exclusive"
        self.params_dtype = (
            torch.half if self.fp16 else torch.bfloat16 if self.bf16 else torch.float32
        )

        # parallel sizes
        mp = self.tensor_model_parallel_size * self.pipeline_model_parallel_size
        assert self.world_size % mp == 0, (
            f"world size {self.world_size} not divisible by TPxPP {mp}"
        )
        self.data_parallel_size = self.world_size // mp

        # overlapped bucket all-reduce and the distributed optimizer's
        # whole-buffer reduce-scatter are alternative DP reduction paths
        assert not (self.overlap_grad_reduce and self.use_distributed_optimizer), (
            "overlap_grad_reduce requires the non-distributed optimizer"
        )

        if self.global_batch_size is None:
            self.global_batch_size = self.micro_batch_size * self.data_parallel_size
        assert (
            self.global_batch_size % (self.micro_batch_size * self.data_parallel_size)
            == 0
        ), "global batch must be a multiple of micro_batch * dp"

        # sample-based training: derive the iteration budget (reference
        # validate_args converts train_samples for the constant-batch case)
        if self.train_iters is None and self.train_samples is not None:
            assert self.rampup_batch_size is None, (
                "use --train_iters with batch-size rampup"
            )
            self.train_iters = self.train_samples // self.global_batch_size
            if self.lr_decay_samples is not None and self.lr_decay_iters is None:
                self.lr_decay_iters = (
                    self.lr_decay_samples // self.global_batch_size
                )
            if self.lr_warmup_samples and not self.lr_warmup_iters:
                self.lr_warmup_iters = (
                    self.lr_warmup_samples // self.global_batch_size
                )

        # virtual pipeline
        if self.num_layers_per_virtual_pipeline_stage is not None:
            assert self.pipeline_model_parallel_size > 2, (
                "interleaved schedule needs pp > 2"
            )
            layers_per_stage = self.num_layers // self.pipeline_model_parallel_size
            assert layers_per_stage % self.num_layers_per_virtual_pipeline_stage == 0
            self.virtual_pipeline_model_parallel_size = (
                layers_per_stage // self.num_layers_per_virtual_pipeline_stage
            )

        # architecture derivations (reference arguments.py:233-241)
        if self.ffn_hidden_size is None:
            self.ffn_hidden_size = 4 * self.hidden_size
        if self.kv_channels is None:
            assert self.hidden_size % self.num_attention_heads == 0
            self.kv_channels = self.hidden_size // self.num_attention_heads
        if self.num_attention_heads_kv is None:
            self.num_attention_heads_kv = self.num_attention_heads

        if self.seq_length is not None:
            assert self.max_position_embeddings >= self.seq_length or (
                self.position_embedding_type == "rotary"
            ), "seq_length exceeds max_position_embeddings"

        # sequence parallel needs TP > 1 to do anything
        if self.tensor_model_parallel_size == 1:
            self.sequence_parallel = False
        if self.sequence_parallel:
            # async dgrad all-reduce is replaced by RS/AG pairs under SP
            self.no_async_tensor_model_parallel_allreduce = True

        if self.use_distributed_optimizer:
            assert self.DDP_impl == "local"
            assert self.use_contiguous_buffers_in_local_ddp

        if self.recompute_granularity == "full":
            assert self.recompute_method in ("uniform", "block", None)

        if self.lr_warmup_fraction is not None:
            assert self.lr_warmup_iters == 0 and self.lr_warmup_samples == 0

        if self.weight_decay is not None:
            if self.start_weight_decay is None:
                self.start_weight_decay = self.weight_decay
            if self.end_weight_decay is None:
                self.end_weight_decay = self.weight_decay

        return self

    def pad_vocab_size(self, orig_vocab_size: int) -> int:
        """Pad vocab to a multiple of make_vocab_size_divisible_by * tp
        (reference megatron/tokenizer/tokenizer.py:49-62)."""
        mult = self.make_vocab_size_divisible_by * self.tensor_model_parallel_size
        after = orig_vocab_size
        while after % mult != 0:
            after += 1
        self.padded_vocab_size = after
        return after


# ---------------------------------------------------------------------------
# module-level current config (explicit replacement of get_args())
_CONFIG: Optional[TrainingConfig] = None


def set_config(cfg: TrainingConfig) -> None:
    global _CONFIG
    _CONFIG = cfg


def get_config() -> TrainingConfig:
    assert _CONFIG is not None, "config not set — call set_config() / initialize()"
    return _CONFIG


def config_is_set() -> bool:
    return _CONFIG is not None


# ---------------------------------------------------------------------------
# argparse front-end with the reference's underscore flag names


def _add_bool(parser, name, default, help=""):
    # reference uses --no_X style for disabling defaults-on flags
    if default:
        parser.add_argument(f"--no_{name}", dest=name, action="store_false", help=help)
    else:
        parser.add_argument(f"--{name}", action="store_true", help=help)


def build_parser(extra_args_provider=None) -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(description="megatron_amd", allow_abbrev=False)
    flds = {f.name: f for f in dataclasses.fields(TrainingConfig)}
    skip = {"params_dtype", "consumed_train_samples", "consumed_valid_samples",
            "iteration", "do_train", "do_valid", "do_test", "data_parallel_size",
            "rank", "world_size", "curriculum_learning",
            "virtual_pipeline_model_parallel_size", "padded_vocab_size"}
    for name, f in flds.items():
        if name in skip:
            continue
        if f.type in ("bool",) or isinstance(f.default, bool):
            _add_bool(p, name, f.default)
        elif f.type.startswith("List") or "List" in str(f.type):
            default = None if isinstance(f.default, dataclasses._MISSING_TYPE) else f.default
            p.add_argument(f"--{name}", nargs="*", default=default)
        else:
            typ = {"int": int, "float": float, "str": str,
                   "Optional[int]": int, "Optional[float]": float,
                   "Optional[str]": str}.get(str(f.type), str)
            default = None if isinstance(f.default, dataclasses._MISSING_TYPE) else f.default
            p.add_argument(f"--{name}", type=typ, default=default)
    if extra_args_provider is not None:
        p = extra_args_provider(p)
    return p


def parse_config(extra_args_provider=None, args_list=None, defaults=None) -> TrainingConfig:
    import os

    parser = build_parser(extra_args_provider)
    ns, _unknown = parser.parse_known_args(args_list)
    kwargs = {k: v for k, v in vars(ns).items()
              if k in {f.name for f in dataclasses.fields(TrainingConfig)}}
    if defaults:
        for k, v in defaults.items():
            if parser.get_default(k) == kwargs.get(k):  # not overridden on CLI
                kwargs[k] = v
    cfg = TrainingConfig(**kwargs)
    # extra_args_provider flags that are not dataclass fields (e.g. --port
    # of the generation server) still ride on the config object
    field_names = {f.name for f in dataclasses.fields(TrainingConfig)}
    for k, v in vars(ns).items():
        if k not in field_names:
            setattr(cfg, k, v)
    # environment (torchrun)
    cfg.rank = int(os.environ.get("RANK", "0"))
    cfg.world_size = int(os.environ.get("WORLD_SIZE", "1"))
    if cfg.local_rank is None:
        cfg.local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    cfg.finalize()
    return cfg
