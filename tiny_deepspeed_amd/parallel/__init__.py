"""Parallelism layer: DP (DDP) and ZeRO-1/2/3 strategies over RCCL/xGMI.

Parity with ``/root/reference/tiny_deepspeed/core/zero/`` (SURVEY.md
components #14-19): model wrappers, strategy modules, distributed
optimizers, layer-swap machinery and the partitioner.
"""

from .comm import CommContext, default_comm
from .partition import partition_tensors, MI355X_HBM_BYTES
from .wrapper import ModelWrapper, Single
from .ddp import DDP, DDPSGD, DDPAdamW
from .zero1 import Zero1, Zero1SGD, Zero1AdamW
from .zero2 import Zero2, Zero2SGD, Zero2AdamW
from .zero3 import Zero3, Zero3SGD, Zero3AdamW
from .flat import Zero2Flat, Zero2FlatSGD, Zero2FlatAdamW

__all__ = [
    "CommContext", "default_comm",
    "partition_tensors", "MI355X_HBM_BYTES",
    "ModelWrapper", "Single",
    "DDP", "DDPSGD", "DDPAdamW",
    "Zero1", "Zero1SGD", "Zero1AdamW",
    "Zero2", "Zero2SGD", "Zero2AdamW",
    "Zero3", "Zero3SGD", "Zero3AdamW",
    "Zero2Flat", "Zero2FlatSGD", "Zero2FlatAdamW",
]
