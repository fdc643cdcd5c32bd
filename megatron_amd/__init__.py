"""megatron_amd — MI355X-native 3D-parallel LLM training framework.

A from-scratch AMD CDNA4 (gfx950) framework with the capabilities of
epfLLM/Megatron-LLM: tensor/pipeline/data parallelism + Megatron-style
sequence parallelism, Llama / Llama-2 / Code-Llama / Falcon / Mistral model
families, GQA/MQA, RoPE (+scaling), RMSNorm, SwiGLU, sliding-window
attention, BF16/FP16 mixed precision, ZeRO-1-style distributed optimizer,
HF<->Megatron weight conversion, and a text-generation server.

Compute path: PyTorch-ROCm (hipBLASLt GEMMs) + hand-written HIP/CDNA4
kernels (megatron_amd/ops/csrc) + RCCL over xGMI through torch.distributed.
"""

__version__ = "0.1.0"

from .config import TrainingConfig, get_config, set_config  # noqa: F401
