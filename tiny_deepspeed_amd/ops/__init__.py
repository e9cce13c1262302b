"""Op layer: device-dispatched op functions (HIP kernels on gfx950, torch on CPU).

Public surface mirrors the reference's ops re-exports
(``/root/reference/tiny_deepspeed/core/module/ops/__init__.py:4-18``) plus
the additional fused hot ops the MI355X build hand-writes (gelu, attention,
cross-entropy, fused optimizer updates).
"""

from .linear import (
    linear_forward,
    linear_input_grad,
    linear_weight_grad,
    linear_bias_grad,
)
from .layernorm import (layernorm_fwd, layernorm_fwd_res, layernorm_dx,
                        layernorm_dwdb)
from .embedding import embedding_forward, embedding_weight_grad
from .gelu import gelu, gelu_fwd, gelu_bwd
from .attention import causal_attention, fused_causal_attention
from .cross_entropy import cross_entropy, cross_entropy_fwd, cross_entropy_bwd
from .optim_ops import adamw_step, sgd_step
from .autotuner import RuntimeAutoTuner, default_tuner
from ._ext import ext_available, get_ext
from .utils import acc_dtype

__all__ = [
    "linear_forward", "linear_input_grad", "linear_weight_grad", "linear_bias_grad",
    "layernorm_fwd", "layernorm_fwd_res", "layernorm_dx", "layernorm_dwdb",
    "embedding_forward", "embedding_weight_grad",
    "gelu", "gelu_fwd", "gelu_bwd",
    "causal_attention", "fused_causal_attention",
    "cross_entropy", "cross_entropy_fwd", "cross_entropy_bwd",
    "adamw_step", "sgd_step",
    "RuntimeAutoTuner", "default_tuner",
    "ext_available", "get_ext",
    "acc_dtype",
]
