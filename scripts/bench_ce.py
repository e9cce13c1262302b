"""Isolated cross-entropy fwd/bwd timing at GPT-2 medium scale."""

import os
import sys
import time

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

import torch

import importlib
ce = importlib.import_module("tiny_deepspeed_amd.ops.cross_entropy")


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    R, V = 32768, 50304
    logits = torch.randn(R, V, device="cuda", dtype=torch.bfloat16)
    tgt = torch.randint(0, V, (R,), device="cuda")
    loss_sum, lse, n_valid = ce.cross_entropy_fwd(logits, tgt)
    gb = R * V * 2 / 2**30
    t1 = timeit(lambda: ce.cross_entropy_fwd(logits, tgt))
    t2 = timeit(lambda: ce.cross_entropy_bwd(1.0, logits, tgt, lse, R))
    print(f"ce fwd {t1*1e6:8.1f} us  ({gb/t1:6.0f} GB/s)")
    print(f"ce bwd {t2*1e6:8.1f} us  ({2*gb/t2:6.0f} GB/s)")


if __name__ == "__main__":
    main()
