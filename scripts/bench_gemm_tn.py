"""TN dW GEMM: hand-written CDNA4 kernel vs hipBLASLt on the model's
weight-grad shape families (dw[N,K] = dy[M,N]^T @ x[M,K], M = B*T).

The measurement that settles SURVEY 2.10B / VERDICT r1 item 3: whichever
implementation wins per shape is what the autotuner dispatches; this script
records the verdict for BASELINE.md. Run on an MI355X:
    python scripts/bench_gemm_tn.py [--iters 50]
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from tiny_deepspeed_amd.ops import get_ext

# gpt2-medium b32 families + lm_head (fused-CE dW) + gpt2-small/large dims
SHAPES = [
    (32768, 3072, 1024),   # c_attn dW
    (32768, 1024, 1024),   # attn proj dW
    (32768, 4096, 1024),   # c_fc dW
    (32768, 1024, 4096),   # mlp proj dW
    (32768, 50304, 1024),  # lm_head dW
    (32768, 2304, 768),    # gpt2-small c_attn
    (32768, 1280, 5120),   # gpt2-large mlp proj
]


def time_fn(fn, iters, warmup=10):
    for _ in range(warmup):
        fn()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    e.synchronize()
    return s.elapsed_time(e) / iters  # ms


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=50)
    args = ap.parse_args()
    ext = get_ext()
    torch.manual_seed(0)
    print(f"{'M':>6} {'N':>6} {'K':>6} {'hip_ms':>8} {'hip_TF':>7} "
          f"{'lib_ms':>8} {'lib_TF':>7} {'winner':>7} {'maxerr':>9}")
    for M, N, K in SHAPES:
        dy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16) * 0.05
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        flop = 2.0 * M * N * K
        hip_ms = time_fn(lambda: ext.gemm_tn(dy, x), args.iters)
        lib_ms = time_fn(lambda: torch.matmul(dy.t(), x), args.iters)
        out = ext.gemm_tn(dy, x).float()
        ref = torch.matmul(dy.t(), x).float()
        err = (out - ref).abs().max().item() / max(ref.abs().max().item(), 1e-6)
        win = "hip" if hip_ms < lib_ms else "lib"
        print(f"{M:>6} {N:>6} {K:>6} {hip_ms:8.3f} {flop/hip_ms/1e9:7.0f} "
              f"{lib_ms:8.3f} {flop/lib_ms/1e9:7.0f} {win:>7} {err:9.2e}")


if __name__ == "__main__":
    main()
