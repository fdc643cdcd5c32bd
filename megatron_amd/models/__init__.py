from .enums import AttnMaskType, AttnType, LayerType, ModelType  # noqa: F401
from .gpt_model import (  # noqa: F401
    CodeLlamaModel,
    FalconModel,
    GPTModel,
    LlamaModel,
    MistralModel,
)
from .language_model import (  # noqa: F401
    init_method_normal,
    parallel_lm_logits,
    scaled_init_method_normal,
)
from .bert_model import BertModel  # noqa: F401
from .module import Float16Module, MegatronModule  # noqa: F401
from .t5_model import T5Model  # noqa: F401
from .norms import LayerNorm, RMSNorm  # noqa: F401
from .transformer import (  # noqa: F401
    CoreAttention,
    ParallelAttention,
    ParallelMLP,
    ParallelTransformer,
    ParallelTransformerLayer,
)

MODEL_CLASSES = {
    "gpt": GPTModel,
    "llama": LlamaModel,
    "llama2": LlamaModel,
    "codellama": CodeLlamaModel,
    "falcon": FalconModel,
    "mistral": MistralModel,
}
