"""Optimizers: from-scratch SGD and AdamW over named parameters, with fused
CDNA4 update kernels on GPU (parity: /root/reference/tiny_deepspeed/core/optim/)."""

from .base import Optimizer
from .sgd import SGD
from .adamw import AdamW

__all__ = ["Optimizer", "SGD", "AdamW"]
