"""Linear (affine) op functions: forward, input-grad, weight-grad, bias-grad.

Capability parity with the reference ops
(``/root/reference/tiny_deepspeed/core/module/ops/linear.py:9-75``).

MI355X design note: plain GEMMs belong to the library path — torch.matmul on
ROCm dispatches to hipBLASLt/rocBLAS which drive the MFMA matrix cores
directly, so these four functions are thin shape adapters around matmul with
the right transpose forms:

  fwd : out[M,N] = x[M,K] @ W[N,K]^T (+ b)      (NT GEMM)
  dX  : dx[M,K]  = dy[M,N] @ W[N,K]             (NN GEMM)
  dW  : dw[N,K]  = dy[M,N]^T @ x[M,K]           (TN GEMM)
  db  : db[N]    = sum_M dy[M,N]                (column reduce, HIP kernel)

The *fused* hot ops (layernorm, gelu, attention, cross-entropy, optimizers,
embedding) are hand-written CDNA4 kernels — see the sibling modules.
"""

import torch

from . import _ext


def _flatten_batch(t):
    # (B, T, E) -> (B*T, E); 2-D tensors pass through.
    return t.reshape(-1, t.shape[-1])


def linear_forward(x, weight, bias=None, tuner=None):
    out = torch.matmul(x, weight.t())
    if bias is not None:
        out = out + bias
    return out


def linear_input_grad(dy, weight, tuner=None):
    return torch.matmul(dy, weight)


def linear_weight_grad(dy, x, tuner=None):
    dy2 = _flatten_batch(dy)
    x2 = _flatten_batch(x)
    return torch.matmul(dy2.t(), x2)


def linear_bias_grad(dy, tuner=None):
    dy2 = _flatten_batch(dy)
    if _ext.use_native(dy2):
        return _ext.get_ext().column_sum(dy2.contiguous())
    return dy2.sum(dim=0)
