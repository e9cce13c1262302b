"""tiny_deepspeed_amd — an MI355X-native minimal ZeRO training engine.

A from-scratch PyTorch-ROCm framework with the capabilities of
liangyuwang/Tiny-DeepSpeed: DDP + ZeRO-1/2/3 data-parallel
training on a single 8xMI355X node, with hand-written CDNA4 HIP kernels for
the hot ops (LayerNorm, GELU, embedding, cross-entropy, fused AdamW/SGD,
fused causal attention) and RCCL collectives over xGMI on a dedicated HIP
stream, overlapped with backward.

Public API surface mirrors the reference's
(``/root/reference/tiny_deepspeed/core/__init__.py:5-23``):
SGD, AdamW, DDP, DDPSGD, DDPAdamW, Zero1*, Zero2*, Zero3*, partition_tensors.
"""

from .optim import SGD, AdamW
from .parallel import (
    DDP, DDPSGD, DDPAdamW,
    Zero1, Zero1SGD, Zero1AdamW,
    Zero2, Zero2SGD, Zero2AdamW,
    Zero3, Zero3SGD, Zero3AdamW,
    Zero2Flat, Zero2FlatSGD, Zero2FlatAdamW,
    Single,
    partition_tensors,
)

__version__ = "0.1.0"

__all__ = [
    "SGD", "AdamW",
    "DDP", "DDPSGD", "DDPAdamW",
    "Zero1", "Zero1SGD", "Zero1AdamW",
    "Zero2", "Zero2SGD", "Zero2AdamW",
    "Zero3", "Zero3SGD", "Zero3AdamW",
    "Zero2Flat", "Zero2FlatSGD", "Zero2FlatAdamW",
    "Single",
    "partition_tensors",
]
