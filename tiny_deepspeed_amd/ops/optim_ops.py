"""Fused optimizer-update ops (AdamW, SGD w/ momentum).

The reference's AdamW/SGD update math is a chain of 7+ torch kernel launches
per parameter (``/root/reference/tiny_deepspeed/core/optim/adamw.py:32-59``,
``sgd.py:28-46``). Here each update is ONE CDNA4 kernel (csrc/optim.hip):
m, v, bias-correction and the parameter write happen in a single HBM pass.
Optimizer state (m, v, momentum) is fp32; when the parameter is bf16 a
separate fp32 master copy is updated and the bf16 param is written from it.

CPU fallback implements identical math in torch (used as the numerics
oracle by tests).
"""

import math

import torch

from . import _ext


def adamw_step(param, grad, exp_avg, exp_avg_sq, master, step, lr, beta1, beta2,
               eps, weight_decay, max_exp_avg_sq=None):
    """In-place AdamW update. `master` is the fp32 copy when param is not fp32
    (None for fp32 params). `step` is the 1-based global step count.
    `max_exp_avg_sq` enables amsgrad."""
    if _ext.use_native(param):
        _ext.get_ext().adamw_step(
            param, grad, exp_avg, exp_avg_sq,
            master if master is not None else param,
            max_exp_avg_sq if max_exp_avg_sq is not None else exp_avg,
            master is not None, max_exp_avg_sq is not None,
            float(lr), float(beta1), float(beta2), float(eps),
            float(weight_decay), int(step),
        )
        return
    p32 = master if master is not None else param
    g32 = grad.float()
    p32.mul_(1.0 - lr * weight_decay)
    exp_avg.mul_(beta1).add_(g32, alpha=1.0 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(g32, g32, value=1.0 - beta2)
    bc1 = 1.0 - beta1 ** step
    bc2 = 1.0 - beta2 ** step
    if max_exp_avg_sq is not None:
        torch.maximum(max_exp_avg_sq, exp_avg_sq, out=max_exp_avg_sq)
        denom = (max_exp_avg_sq / bc2).sqrt().add_(eps)
    else:
        denom = (exp_avg_sq / bc2).sqrt().add_(eps)
    p32.addcdiv_(exp_avg, denom, value=-lr / bc1)
    if master is not None:
        param.copy_(p32.to(param.dtype))


def sgd_step(param, grad, momentum_buf, master, lr, momentum, dampening,
             weight_decay, nesterov, maximize, first_step):
    """In-place SGD update matching torch.optim.SGD semantics."""
    if _ext.use_native(param):
        _ext.get_ext().sgd_step(
            param, grad,
            momentum_buf if momentum_buf is not None else param,
            master if master is not None else param,
            momentum_buf is not None, master is not None,
            float(lr), float(momentum), float(dampening),
            float(weight_decay), bool(nesterov), bool(maximize),
            bool(first_step),
        )
        return
    p32 = master if master is not None else param
    g32 = grad.float()
    if maximize:
        g32 = -g32
    if weight_decay != 0.0:
        g32 = g32.add(p32, alpha=weight_decay)
    if momentum_buf is not None:
        if first_step:
            momentum_buf.copy_(g32)
        else:
            momentum_buf.mul_(momentum).add_(g32, alpha=1.0 - dampening)
        if nesterov:
            g32 = g32.add(momentum_buf, alpha=momentum)
        else:
            g32 = momentum_buf
    p32.add_(g32, alpha=-lr)
    if master is not None:
        param.copy_(p32.to(param.dtype))
