"""LayerNorm module with overridable callbacks.

Parity with ``/root/reference/tiny_deepspeed/core/module/normalization.py:19-109``
including its restrictions: elementwise affine with bias required, and only
last-dimension normalization is supported. Forward saves (x, mean, rstd);
backward computes dx (+ dw/db stripe partials) via the CDNA4 kernel pair.
"""

import torch
import torch.nn as nn

from .. import ops


class _LayerNormResFn(torch.autograd.Function):
    """Fused residual-add + LayerNorm: returns (h, y) with h = x + res.
    Backward folds the residual-stream gradient dh into dx (one kernel)."""

    @staticmethod
    def forward(ctx, x, res, weight, bias, module):
        ctx.module = module
        # don't materialize an all-zeros dh when h is unused (ln_f discards
        # it) — backward treats dh=None as the plain no-fold path
        ctx.set_materialize_grads(False)
        h, y, mean, rstd = module.forward_res_callback(x, res, weight, bias)
        ctx.save_for_backward(h, mean, rstd)
        return h, y

    @staticmethod
    def backward(ctx, dh, dy):
        h, mean, rstd = ctx.saved_tensors
        if dy is None:  # y unused: gradient flows through h alone
            return dh, dh, None, None, None
        dx, dw, db = ctx.module.backward_callback(dy, h, mean, rstd, dh=dh)
        return dx, dx, dw, db, None


class _LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, module):
        ctx.module = module
        y, mean, rstd = module.forward_callback(x, weight, bias)
        ctx.save_for_backward(x, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, mean, rstd = ctx.saved_tensors
        dx, dw, db = ctx.module.backward_callback(dy, x, mean, rstd)
        return dx, dw, db, None


class LayerNorm(nn.LayerNorm):
    def __init__(self, normalized_shape, eps=1e-5, elementwise_affine=True,
                 bias=True, device=None, dtype=None, auto_tune=False):
        if not elementwise_affine or not bias:
            raise NotImplementedError(
                "tiny_deepspeed_amd.LayerNorm requires elementwise_affine=True "
                "and bias=True (reference parity: normalization.py:34-38)"
            )
        super().__init__(normalized_shape, eps=eps,
                         elementwise_affine=elementwise_affine, bias=bias,
                         device=device, dtype=dtype)
        if len(self.normalized_shape) != 1:
            raise NotImplementedError(
                "only last-dim LayerNorm is supported (reference parity: "
                "normalization.py:62-63)"
            )
        self.tuner = ops.RuntimeAutoTuner() if auto_tune else None

    # --- overridable callbacks -------------------------------------------
    def forward_callback(self, x, weight, bias):
        return ops.layernorm_fwd(x, weight, bias, eps=self.eps, tuner=self.tuner)

    def forward_res_callback(self, x, res, weight, bias):
        return ops.layernorm_fwd_res(x, res, weight, bias, eps=self.eps,
                                     tuner=self.tuner)

    def backward_callback(self, dy, x, mean, rstd, dh=None):
        dx, ws = ops.layernorm_dx(dy, x, self.weight, mean, rstd, dh=dh,
                                  tuner=self.tuner)
        if self.weight.requires_grad:
            dw, db = ops.layernorm_dwdb(ws, dtype=self.weight.dtype, tuner=self.tuner)
            self._assert_grad_shapes(dw, db)
        else:
            dw = db = None
        return dx, dw, db

    def _assert_grad_shapes(self, dw, db):
        if dw is not None:
            assert dw.shape == self.weight.shape
        if db is not None:
            assert db.shape == self.bias.shape

    def forward(self, x):
        return _LayerNormFn.apply(x, self.weight, self.bias, self)

    def forward_fused(self, x, res):
        """(h, y) = (x + res, LN(x + res)) with the add fused in-kernel."""
        return _LayerNormResFn.apply(x, res, self.weight, self.bias, self)
