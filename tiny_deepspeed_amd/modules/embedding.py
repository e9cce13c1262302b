"""Embedding module with overridable callbacks.

Parity with ``/root/reference/tiny_deepspeed/core/module/embedding.py:15-98``
(padding_idx plumbing; backward returns (None, grad_weight)). max_norm is
not supported (it mutates the weight in forward — the reference plumbs but
never exercises it; we raise instead of silently ignoring).
"""

import torch
import torch.nn as nn

from .. import ops


class _EmbeddingFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, idx, weight, module):
        ctx.module = module
        ctx.save_for_backward(idx)
        return module.forward_callback(idx, weight)

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        dw = ctx.module.backward_callback(dy, idx)
        return None, dw, None


class Embedding(nn.Embedding):
    def __init__(self, num_embeddings, embedding_dim, padding_idx=None,
                 max_norm=None, norm_type=2.0, scale_grad_by_freq=False,
                 sparse=False, _weight=None, device=None, dtype=None,
                 auto_tune=False):
        if max_norm is not None or scale_grad_by_freq or sparse:
            raise NotImplementedError(
                "max_norm/scale_grad_by_freq/sparse are not supported"
            )
        super().__init__(num_embeddings, embedding_dim, padding_idx=padding_idx,
                         _weight=_weight, device=device, dtype=dtype)
        self.tuner = ops.RuntimeAutoTuner() if auto_tune else None

    # --- overridable callbacks -------------------------------------------
    def forward_callback(self, idx, weight):
        return ops.embedding_forward(weight, idx, padding_idx=self.padding_idx,
                                     tuner=self.tuner)

    def backward_callback(self, dy, idx):
        if not self.weight.requires_grad:
            return None
        dw = ops.embedding_weight_grad(idx, dy, self.num_embeddings,
                                       padding_idx=self.padding_idx,
                                       tuner=self.tuner)
        assert dw.shape == self.weight.shape
        return dw

    def forward(self, idx):
        return _EmbeddingFn.apply(idx, self.weight, self)
