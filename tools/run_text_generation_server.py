"""Start the REST text-generation server (reference
tools/run_text_generation_server.py).

  torchrun --nproc_per_node 2 tools/run_text_generation_server.py \
      --model_name llama2 --load /ckpt --tokenizer_type SentencePieceTokenizer \
      --vocab_file tok.model --tensor_model_parallel_size 2 ...
"""

import os
import sys

sys.path.append(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from megatron_amd import parallel as mpu  # noqa: E402
from megatron_amd.checkpointing import load_checkpoint  # noqa: E402
from megatron_amd.config import get_config  # noqa: E402
from megatron_amd.initialize import initialize_megatron  # noqa: E402
from megatron_amd.inference.server import MegatronServer, run_worker_loop  # noqa: E402
from megatron_amd.models import MODEL_CLASSES, ModelType  # noqa: E402
from megatron_amd.training import get_model  # noqa: E402


def model_provider(pre_process=True, post_process=True):
    cfg = get_config()
    model_cls = MODEL_CLASSES[cfg.model_name or "gpt"]
    return model_cls(cfg, parallel_output=False, pre_process=pre_process,
                     post_process=post_process)


def add_text_generate_args(parser):
    group = parser.add_argument_group(title="text generation")
    group.add_argument("--port", type=int, default=5000)
    group.add_argument("--temperature", type=float, default=1.0)
    group.add_argument("--top_p", type=float, default=0.0)
    group.add_argument("--top_k", type=int, default=0)
    return parser


if __name__ == "__main__":
    initialize_megatron(
        extra_args_provider=add_text_generate_args,
        args_defaults={"no_load_rng": True, "no_load_optim": True,
                       "use_hip_graph_decode": True},
    )
    cfg = get_config()
    model = get_model(model_provider, ModelType.encoder_or_decoder,
                      wrap_with_ddp=False)
    if cfg.load is not None:
        load_checkpoint(model, None, None, cfg)
    assert len(model) == 1
    model = model[0]
    model.eval()

    if (
        mpu.is_pipeline_first_stage()
        and mpu.get_tensor_model_parallel_rank() == 0
    ):
        server = MegatronServer(model)
        server.run(port=int(os.environ.get("PORT", getattr(cfg, "port",
                                                           5000))))
    else:
        run_worker_loop(model)
