"""Dtype policy helpers.

Parity with the reference's acc-dtype table
(``/root/reference/tiny_deepspeed/core/module/ops/utils.py:10-16``): all
reduced-precision compute accumulates in fp32 (int8 in int32). The HIP
kernels hard-code the same policy; this table is the Python-side statement
of it (used by tests and by ops that pick accumulation buffers).
"""

import torch

ACC_DTYPE = {
    torch.float16: torch.float32,
    torch.bfloat16: torch.float32,
    torch.float32: torch.float32,
    torch.int8: torch.int32,
}


def acc_dtype(dtype: torch.dtype) -> torch.dtype:
    return ACC_DTYPE.get(dtype, torch.float32)
