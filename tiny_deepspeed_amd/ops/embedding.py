"""Embedding op functions: gather forward, scatter-add weight grad.

Capability parity with
``/root/reference/tiny_deepspeed/core/module/ops/embedding.py:11-68``,
candidate-list dispatch like the reference (``ops/linear.py:9-17``).

MI355X design: forward is a vectorized row-gather HIP kernel; the weight
grad is an atomic scatter-add into an fp32 accumulation buffer (bf16 grads
would lose counts and bf16 atomics are slow), converted to the param dtype
at the end. padding_idx rows are skipped in the grad like torch.
"""

import torch

from . import _ext
from .autotuner import default_tuner


def emb_fwd_hip(weight, idx, padding_idx=None):
    return _ext.get_ext().embedding_fwd(
        weight, idx.contiguous().view(-1)
    ).view(*idx.shape, weight.shape[1])


def emb_fwd_torch(weight, idx, padding_idx=None):
    return torch.nn.functional.embedding(idx, weight, padding_idx=padding_idx)


def embedding_forward(weight, idx, padding_idx=None, tuner=None):
    if not _ext.use_native(weight):
        return emb_fwd_torch(weight, idx, padding_idx)
    tuner = tuner if tuner is not None else default_tuner()
    if tuner is not None:
        return tuner.choose("emb_fwd", [emb_fwd_hip, emb_fwd_torch],
                            weight, idx, padding_idx)
    return emb_fwd_hip(weight, idx, padding_idx)


def emb_bwd_hip(flat_idx, dy2, num_embeddings, padding_idx):
    dw32 = _ext.get_ext().embedding_bwd(
        dy2.contiguous(), flat_idx.contiguous(), num_embeddings,
        -1 if padding_idx is None else int(padding_idx),
    )
    return dw32.to(dy2.dtype)


def emb_bwd_torch(flat_idx, dy2, num_embeddings, padding_idx):
    dw = torch.zeros(num_embeddings, dy2.shape[-1], dtype=torch.float32,
                     device=dy2.device)
    if padding_idx is not None:
        keep = flat_idx != padding_idx
        flat_idx = flat_idx[keep]
        dy2 = dy2[keep]
    dw.index_add_(0, flat_idx, dy2.float())
    return dw.to(dy2.dtype)


def embedding_weight_grad(idx, dy, num_embeddings, padding_idx=None, tuner=None):
    flat_idx = idx.reshape(-1)
    dy2 = dy.reshape(-1, dy.shape[-1])
    if not _ext.use_native(dy2):
        return emb_bwd_torch(flat_idx, dy2, num_embeddings, padding_idx)
    tuner = tuner if tuner is not None else default_tuner()
    if tuner is not None:
        return tuner.choose("emb_bwd", [emb_bwd_hip, emb_bwd_torch],
                            flat_idx, dy2, num_embeddings, padding_idx)
    return emb_bwd_hip(flat_idx, dy2, num_embeddings, padding_idx)
