"""End-to-end `pretrain()` driver on CPU/gloo (world 1): the full reference
entry path — finetune.py providers -> initialize -> _train loop with eval
interval -> end-of-training eval -> checkpoint save -> resume from it.
(reference finetune.py:242-270 + training.py:55-169)."""

import os
import sys

import pytest
import torch
import torch.multiprocessing as mp

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def _cfg(tmpdir, train_iters, load=None):
    from megatron_amd.config import TrainingConfig

    return TrainingConfig(
        model_name="llama2",
        num_layers=2, hidden_size=32, num_attention_heads=4,
        num_attention_heads_kv=2, seq_length=32, max_position_embeddings=64,
        micro_batch_size=2, global_batch_size=4,  # 2 microbatches
        train_iters=train_iters, lr=1e-3, min_lr=1e-4,
        lr_decay_style="cosine", lr_warmup_iters=1,
        eval_interval=2, eval_iters=1, log_interval=1,
        save=tmpdir, load=load, save_interval=100,
        hidden_dropout=0.0, attention_dropout=0.0,
        use_cpu_initialization=True, clip_grad=1.0,
        world_size=1, rank=0, make_vocab_size_divisible_by=16,
    ).finalize()


def _worker(rank, port, tmpdir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = "0"
    os.environ["WORLD_SIZE"] = "1"
    import torch.distributed as dist

    import finetune
    from megatron_amd import parallel as mpu
    from megatron_amd.models import ModelType
    from megatron_amd.training import pretrain

    cfg = _cfg(tmpdir, train_iters=4)
    cfg.pad_vocab_size(128)

    pretrain(
        finetune.train_valid_test_datasets_provider,
        finetune.model_provider,
        ModelType.encoder_or_decoder,
        finetune.forward_step,
        cfg=cfg,
    )

    # the driver saved the final checkpoint
    assert os.path.isfile(os.path.join(tmpdir, "latest_checkpointed_iteration.txt"))
    with open(os.path.join(tmpdir, "latest_checkpointed_iteration.txt")) as f:
        assert f.read().strip() == "4"
    assert os.path.isdir(os.path.join(tmpdir, "iter_0000004"))

    mpu.destroy_model_parallel()
    dist.destroy_process_group()


def _worker_resume(rank, port, tmpdir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = "0"
    os.environ["WORLD_SIZE"] = "1"
    import torch.distributed as dist

    import finetune
    from megatron_amd import parallel as mpu
    from megatron_amd.models import ModelType
    from megatron_amd.training import pretrain

    cfg = _cfg(tmpdir, train_iters=6, load=tmpdir)
    # extending train_iters past the saved schedule requires the override
    # flag, like the reference's --override-opt_param-scheduler
    cfg.override_opt_param_scheduler = True
    cfg.pad_vocab_size(128)

    pretrain(
        finetune.train_valid_test_datasets_provider,
        finetune.model_provider,
        ModelType.encoder_or_decoder,
        finetune.forward_step,
        cfg=cfg,
    )
    # resumed at 4, trained to 6
    assert cfg.iteration == 6
    with open(os.path.join(tmpdir, "latest_checkpointed_iteration.txt")) as f:
        assert f.read().strip() == "6"

    mpu.destroy_model_parallel()
    dist.destroy_process_group()


def test_pretrain_end_to_end_and_resume(tmp_path):
    d = str(tmp_path)
    mp.spawn(_worker, args=(29641, d), nprocs=1, join=True)
    mp.spawn(_worker_resume, args=(29642, d), nprocs=1, join=True)


def test_bench_torchrun_two_ranks(tmp_path):
    """The driver's multi-GPU invocation shape — torch.distributed.run with
    one rank per 'GPU' — must work end to end (cpu/gloo here): JSON contract
    line parses, n_gpus == world size, dp2 geometry chosen."""
    import json
    import subprocess
    import sys

    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29917", os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--steps", "1", "--warmup", "0"],
        capture_output=True, text=True, timeout=600, cwd=REPO,
    )
    assert r.returncode == 0, r.stderr[-1500:]
    line = [ln for ln in r.stdout.splitlines() if ln.startswith("{")][-1]
    res = json.loads(line)
    assert res["n_gpus"] == 2
    assert res["config"]["parallelism"] == "tp1_pp1_dp2"
