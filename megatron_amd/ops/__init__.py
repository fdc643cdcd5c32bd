"""HIP/CDNA4 ops for MI355X (gfx950).

functional: autograd ops (rmsnorm, layernorm, softmax, glu, rope,
bias_dropout_add, flash attention) dispatching to the in-tree _C.so.
ext: extension loader. build: ahead-of-time hipcc build driver.
"""

from . import ext  # noqa: F401
from .functional import (  # noqa: F401
    apply_rope,
    bias_dropout_add,
    flash_attention,
    glu_activation,
    layernorm,
    rmsnorm,
    scaled_masked_softmax,
    swiglu,
)
