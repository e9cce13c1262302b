"""Linear module with overridable forward/backward callbacks.

Same load-bearing design as the reference's callback indirection
(``/root/reference/tiny_deepspeed/core/module/linear.py:16-92``): forward
and backward route through a torch.autograd.Function whose bodies are
overridable instance methods, which is what lets the parallel strategies
(parallel/ddp.py etc.) inject RCCL collectives between the dW computation
and the dX computation for compute/comm overlap.

Backward contract: `backward_callback(dy, x, weight)` returns
(dx, dweight_or_None, dbias_or_None). The base class computes local grads
with no communication.
"""

import torch
import torch.nn as nn

from .. import ops


class _LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, module):
        ctx.module = module
        y = module.forward_callback(x, weight, bias)
        ctx.save_for_backward(x)
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        dx, dw, db = ctx.module.backward_callback(dy, x)
        return dx, dw, db, None


class Linear(nn.Linear):
    """Drop-in nn.Linear whose fwd/bwd go through the op layer."""

    def __init__(self, in_features, out_features, bias=True, device=None,
                 dtype=None, auto_tune=False):
        super().__init__(in_features, out_features, bias=bias, device=device, dtype=dtype)
        self.tuner = ops.RuntimeAutoTuner() if auto_tune else None

    # --- overridable callbacks -------------------------------------------
    def forward_callback(self, x, weight, bias):
        return ops.linear_forward(x, weight, bias, tuner=self.tuner)

    def backward_callback(self, dy, x):
        dw = ops.linear_weight_grad(dy, x, tuner=self.tuner) if self.weight.requires_grad else None
        db = (
            ops.linear_bias_grad(dy, tuner=self.tuner)
            if (self.bias is not None and self.bias.requires_grad)
            else None
        )
        dx = ops.linear_input_grad(dy, self.weight, tuner=self.tuner)
        self._assert_grad_shapes(dw, db)
        return dx, dw, db

    def _assert_grad_shapes(self, dw, db):
        if dw is not None:
            assert dw.shape == self.weight.shape, (
                f"dW shape {tuple(dw.shape)} != weight {tuple(self.weight.shape)}"
            )
        if db is not None and self.bias is not None:
            assert db.shape == self.bias.shape

    def forward(self, x):
        return _LinearFn.apply(x, self.weight, self.bias, self)
