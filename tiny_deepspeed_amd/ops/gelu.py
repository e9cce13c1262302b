"""GELU (tanh approximation) forward/backward, autograd-wrapped.

The reference leaves GELU to torch autograd inside the MLP
(``/root/reference/example/model.py:94``); here it is a hand-written
elementwise CDNA4 kernel (bf16x8 vectorized, HBM-bound) with fwd+bwd,
dispatched against the torch composite through the runtime autotuner
(reference candidate-list architecture, ``ops/linear.py:9-17``).
"""

import math

import torch

from . import _ext
from .autotuner import default_tuner

_C0 = math.sqrt(2.0 / math.pi)
_C1 = 0.044715


def gelu_fwd_hip(x):
    return _ext.get_ext().gelu_fwd(x.contiguous())


def gelu_fwd_torch(x):
    xf = x.float()
    y = 0.5 * xf * (1.0 + torch.tanh(_C0 * (xf + _C1 * xf * xf * xf)))
    return y.to(x.dtype)


def gelu_fwd(x, tuner=None):
    if not _ext.use_native(x):
        return gelu_fwd_torch(x)
    tuner = tuner if tuner is not None else default_tuner()
    if tuner is not None:
        return tuner.choose("gelu_fwd", [gelu_fwd_hip, gelu_fwd_torch], x)
    return gelu_fwd_hip(x)


def gelu_bwd_hip(dy, x):
    return _ext.get_ext().gelu_bwd(dy.contiguous(), x.contiguous())


def gelu_bwd_torch(dy, x):
    xf = x.float()
    dyf = dy.float()
    t = torch.tanh(_C0 * (xf + _C1 * xf * xf * xf))
    dt = (1.0 - t * t) * _C0 * (1.0 + 3.0 * _C1 * xf * xf)
    dx = dyf * (0.5 * (1.0 + t) + 0.5 * xf * dt)
    return dx.to(x.dtype)


def gelu_bwd(dy, x, tuner=None):
    if not _ext.use_native(dy):
        return gelu_bwd_torch(dy, x)
    tuner = tuner if tuner is not None else default_tuner()
    if tuner is not None:
        return tuner.choose("gelu_bwd", [gelu_bwd_hip, gelu_bwd_torch], dy, x)
    return gelu_bwd_hip(dy, x)


class _GeluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.save_for_backward(x)
        return gelu_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        return gelu_bwd(dy, x)


def gelu(x):
    return _GeluFn.apply(x)
