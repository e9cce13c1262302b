"""Embedding op functions: gather forward, scatter-add weight grad.

Capability parity with
``/root/reference/tiny_deepspeed/core/module/ops/embedding.py:11-68``.

MI355X design: forward is a vectorized row-gather HIP kernel; the weight
grad is an atomic scatter-add into an fp32 accumulation buffer (bf16 grads
would lose counts and bf16 atomics are slow), converted to the param dtype
at the end. padding_idx rows are skipped in the grad like torch.
"""

import torch

from . import _ext


def embedding_forward(weight, idx, padding_idx=None, tuner=None):
    if _ext.use_native(weight):
        return _ext.get_ext().embedding_fwd(weight, idx.contiguous().view(-1)).view(
            *idx.shape, weight.shape[1]
        )
    return torch.nn.functional.embedding(idx, weight, padding_idx=padding_idx)


def embedding_weight_grad(idx, dy, num_embeddings, padding_idx=None, tuner=None):
    flat_idx = idx.reshape(-1)
    dy2 = dy.reshape(-1, dy.shape[-1])
    if _ext.use_native(dy2):
        dw32 = _ext.get_ext().embedding_bwd(
            dy2.contiguous(), flat_idx.contiguous(), num_embeddings,
            -1 if padding_idx is None else int(padding_idx),
        )
        return dw32.to(dy.dtype)
    dw = torch.zeros(num_embeddings, dy.shape[-1], dtype=torch.float32, device=dy.device)
    if padding_idx is not None:
        keep = flat_idx != padding_idx
        flat_idx = flat_idx[keep]
        dy2 = dy2[keep]
    dw.index_add_(0, flat_idx, dy2.float())
    return dw.to(dy.dtype)
