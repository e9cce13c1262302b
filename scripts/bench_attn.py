"""Isolated attention kernel benchmark (fwd / bwd), with TF/s accounting.

Usage (GPU box): python scripts/bench_attn.py [--T 1024] [--B 8] [--H 16]
Causal flops: ~0.5 * 2 gemms * 2*B*H*T^2*D each for fwd; bwd ~2.5x fwd.
"""

import argparse
import math
import os
import sys
import time

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

import torch

from tiny_deepspeed_amd import _C


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
    p = argparse.ArgumentParser()
    p.add_argument("--B", type=int, default=8)
    p.add_argument("--H", type=int, default=16)
    p.add_argument("--T", type=int, default=1024)
    p.add_argument("--iters", type=int, default=20)
    args = p.parse_args()
    B, H, T, D = args.B, args.H, args.T, 64
    scale = 1.0 / math.sqrt(D)
    torch.manual_seed(0)
    q = torch.randn(B, H, T, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    o, lse = _C.attention_fwd(q, k, v, scale)
    do = torch.randn_like(o)

    t_fwd = timeit(lambda: _C.attention_fwd(q, k, v, scale), args.iters)
    t_bwd = timeit(lambda: _C.attention_bwd(q, k, v, o, lse, do, scale),
                   args.iters)
    # causal effective flops
    f_fwd = 0.5 * 2 * (2 * B * H * T * T * D)
    f_bwd = 0.5 * 5 * (2 * B * H * T * T * D)  # dkv recompute S,dP + dV,dK; dq S,dP,dQ
    print(f"fwd: {t_fwd*1e6:8.1f} us  {f_fwd/t_fwd/1e12:7.1f} TF/s (causal-effective)")
    print(f"bwd: {t_bwd*1e6:8.1f} us  {f_bwd/t_bwd/1e12:7.1f} TF/s (causal-effective)")

    # sdpa comparison (rocm flash attention via torch, if available)
    qf = q.clone(); kf = k.clone(); vf = v.clone()
    def sdpa():
        return torch.nn.functional.scaled_dot_product_attention(
            qf, kf, vf, is_causal=True, scale=scale)
    try:
        t_sdpa = timeit(sdpa, args.iters)
        print(f"torch sdpa fwd: {t_sdpa*1e6:8.1f} us  {f_fwd/t_sdpa/1e12:7.1f} TF/s")
    except Exception as e:
        print("sdpa failed:", e)


if __name__ == "__main__":
    main()
