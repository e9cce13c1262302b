"""Fused softmax cross-entropy over the vocabulary, autograd-wrapped.

The reference uses F.cross_entropy (``/root/reference/example/model.py:156``).
Here: a fused CDNA4 kernel pair — forward computes per-row max/logsumexp and
the NLL in one HBM pass over the (B*T, 50304) logits; backward writes
dlogits = (softmax - onehot) * scale in one pass. fp32 accumulation,
logits dtype in/out. Mean reduction over rows (ignore_index supported).
Dispatch is {CDNA4 kernel, torch composite} through the runtime autotuner
(reference candidate-list architecture, ``ops/linear.py:9-17``).
"""

import torch

from . import _ext
from .autotuner import default_tuner


def _kernel_supported(logits2d):
    """The CDNA4 kernels vector-load rows: V must keep row bases 16B-aligned
    (V % 8 bf16 / V % 4 fp32 — guarded in tdsa_ce_fwd/bwd). Off-width vocabs
    (e.g. unpadded 50257) use the composite path."""
    w = 8 if logits2d.dtype == torch.bfloat16 else 4
    return logits2d.shape[-1] % w == 0


def ce_fwd_hip(logits2d, targets, ignore_index):
    return _ext.get_ext().cross_entropy_fwd(
        logits2d.contiguous(), targets.contiguous(), ignore_index
    )


def ce_fwd_torch(logits2d, targets, ignore_index):
    lf = logits2d.float()
    m = lf.max(dim=-1).values
    lse = m + (lf - m.unsqueeze(-1)).exp().sum(dim=-1).log()
    valid = targets != ignore_index
    tgt = targets.clamp_min(0)
    picked = lf.gather(1, tgt.unsqueeze(1)).squeeze(1)
    losses = torch.where(valid, lse - picked, torch.zeros_like(lse))
    return losses.sum(), lse, valid.sum()


def cross_entropy_fwd(logits2d, targets, ignore_index=-100, tuner=None):
    """Returns (loss_sum[fp32 scalar], lse[rows fp32], n_valid[int64 scalar])."""
    if not (_ext.use_native(logits2d) and _kernel_supported(logits2d)):
        return ce_fwd_torch(logits2d, targets, ignore_index)
    tuner = tuner if tuner is not None else default_tuner()
    if tuner is not None:
        return tuner.choose("ce_fwd", [ce_fwd_hip, ce_fwd_torch],
                            logits2d, targets, ignore_index)
    return ce_fwd_hip(logits2d, targets, ignore_index)


def ce_bwd_hip(dloss, logits2d, targets, lse, n_valid, ignore_index, out=None):
    return _ext.get_ext().cross_entropy_bwd(
        logits2d.contiguous(), targets.contiguous(), lse,
        float(dloss), int(n_valid), ignore_index, out,
    )


def ce_bwd_torch(dloss, logits2d, targets, lse, n_valid, ignore_index, out=None):
    lf = logits2d.float()
    soft = (lf - lse.unsqueeze(-1)).exp()
    valid = (targets != ignore_index).unsqueeze(1)
    tgt = targets.clamp_min(0)
    soft.scatter_add_(
        1, tgt.unsqueeze(1), -torch.ones_like(tgt, dtype=torch.float32).unsqueeze(1)
    )
    scale = float(dloss) / max(int(n_valid), 1)
    dlogits = torch.where(valid, soft * scale, torch.zeros_like(soft))
    if out is not None:
        out.copy_(dlogits.to(logits2d.dtype))
        return out
    return dlogits.to(logits2d.dtype)


def cross_entropy_bwd(dloss, logits2d, targets, lse, n_valid,
                      ignore_index=-100, tuner=None, out=None):
    """out: optional pre-allocated dlogits destination (e.g. a row-slice of
    the fused lm_head+CE full-dlogits buffer) — skips an extra copy."""
    if not (_ext.use_native(logits2d) and _kernel_supported(logits2d)):
        return ce_bwd_torch(dloss, logits2d, targets, lse, n_valid,
                            ignore_index, out)
    tuner = tuner if tuner is not None else default_tuner()
    if tuner is not None:
        return tuner.choose("ce_bwd", [ce_bwd_hip, ce_bwd_torch],
                            dloss, logits2d, targets, lse, n_valid,
                            ignore_index, out)
    return ce_bwd_hip(dloss, logits2d, targets, lse, n_valid, ignore_index,
                      out)


class _CrossEntropyFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits2d, targets, ignore_index):
        loss_sum, lse, n_valid = cross_entropy_fwd(logits2d, targets, ignore_index)
        ctx.save_for_backward(logits2d, targets, lse, n_valid)
        ctx.ignore_index = ignore_index
        n = max(int(n_valid), 1)
        return loss_sum / n

    @staticmethod
    def backward(ctx, dloss):
        logits2d, targets, lse, n_valid = ctx.saved_tensors
        dlogits = cross_entropy_bwd(
            dloss, logits2d, targets, lse, n_valid, ctx.ignore_index
        )
        return dlogits, None, None


def cross_entropy(logits, targets, ignore_index=-100):
    """Mean cross-entropy; logits (..., V), targets (...) int64."""
    logits2d = logits.reshape(-1, logits.shape[-1])
    return _CrossEntropyFn.apply(logits2d, targets.reshape(-1), ignore_index)
