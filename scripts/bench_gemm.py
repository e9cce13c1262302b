"""Per-shape GEMM throughput for the GPT-2 model's hot shapes, as driven by
torch.matmul (hipBLASLt). Baseline for deciding whether a hand-written MFMA
GEMM can beat the library path on these shapes.

Usage on GPU box:  python scripts/bench_gemm.py [--model gpt2-medium] [--tokens 32768]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

import torch

PRESETS = {
    "gpt2-small": (768, 50304),
    "gpt2-medium": (1024, 50304),
    "gpt2-large": (1280, 50304),
    "gpt2-xl": (1600, 50304),
}


def timeit(fn, iters=30, warmup=5):
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
    p.add_argument("--model", default="gpt2-medium")
    p.add_argument("--tokens", type=int, default=32768)
    args = p.parse_args()
    E, V = PRESETS[args.model]
    M = args.tokens
    # (name, N, K) for out[M,N] = x[M,K] @ W[N,K]^T and its grads
    shapes = [
        ("qkv", 3 * E, E),
        ("attn_proj", E, E),
        ("mlp_fc", 4 * E, E),
        ("mlp_proj", E, 4 * E),
        ("lm_head", V, E),
    ]
    total_ms = 0.0
    total_tf = 0.0
    for name, N, K in shapes:
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
        dy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
        flops = 2.0 * M * N * K
        t_fwd = timeit(lambda: torch.matmul(x, w.t()))        # NT
        t_dx = timeit(lambda: torch.matmul(dy, w))            # NN
        t_dw = timeit(lambda: torch.matmul(dy.t(), x))        # TN
        for tag, t in (("fwd NT", t_fwd), ("dX  NN", t_dx), ("dW  TN", t_dw)):
            print(f"{name:10s} {tag}  M={M} N={N} K={K}: "
                  f"{t*1e6:9.1f} us  {flops/t/1e12:7.0f} TF/s")
            total_ms += t * 1e3
            total_tf += flops / 1e12
        del x, w, dy
    print(f"\nTOTAL (one fwd+dX+dW pass over the shape set): {total_ms:.2f} ms, "
          f"aggregate {total_tf/(total_ms/1e3):.0f} TF/s")


if __name__ == "__main__":
    main()
