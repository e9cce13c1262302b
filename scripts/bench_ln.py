"""Isolated LayerNorm fwd/bwd timing at GPT-2 medium shapes (within-box
tuning of the launcher knobs TDSA_LN_GRID / TDSA_LN_STRIPES)."""

import os
import sys
import time

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

import torch

from tiny_deepspeed_amd import ops


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    M, N = 32768, 1024
    x = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
    res = torch.randn_like(x)
    w = torch.randn(N, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(N, device="cuda", dtype=torch.bfloat16)
    y, mean, rstd = ops.layernorm_fwd(x, w, b)
    dy = torch.randn_like(x)
    dh = torch.randn_like(x)
    t1 = timeit(lambda: ops.layernorm_fwd(x, w, b))
    t2 = timeit(lambda: ops.layernorm_fwd_res(x, res, w, b))
    t3 = timeit(lambda: ops.layernorm_dx(dy, x, w, mean, rstd, dh=dh))
    def full_bwd():
        dx, ws = ops.layernorm_dx(dy, x, w, mean, rstd, dh=dh)
        ops.layernorm_dwdb(ws, dtype=torch.bfloat16)
    t4 = timeit(full_bwd)
    gb = M * N * 2 / 2**30
    print(f"fwd       {t1*1e6:7.1f} us  ({3*gb/t1:6.0f} GB/s streams)")
    print(f"fwd+res   {t2*1e6:7.1f} us  ({4*gb/t2:6.0f} GB/s)")
    print(f"bwd dx+dh {t3*1e6:7.1f} us  ({4*gb/t3:6.0f} GB/s)")
    print(f"bwd full  {t4*1e6:7.1f} us")


if __name__ == "__main__":
    main()
