"""On-GPU probe: determine the v_mfma_f32_16x16x32_bf16 fragment layout and
validate the attention LDS staging path. Run on an MI355X box; prints a
verdict per hypothesis and dumps raw D on mismatch."""

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

import torch

from tiny_deepspeed_amd import _C


def main():
    torch.manual_seed(0)
    dev = "cuda"
    # integer-valued bf16-exact data, asymmetric (transpose-detecting)
    A = (torch.randint(-8, 8, (16, 32), device=dev).to(torch.bfloat16))
    B = (torch.randint(-8, 8, (32, 16), device=dev).to(torch.bfloat16))
    ref = A.float() @ B.float()
    for variant in (0, 1):
        D = _C.dbg_mfma(A, B, variant)
        checks = {
            "D==A@B": (D - ref).abs().max().item(),
            "D==(A@B).T": (D - ref.t()).abs().max().item(),
        }
        print(f"variant {variant}: " + "  ".join(
            f"{k}: {v:.4f}" for k, v in checks.items()))
    # if nothing matched, dump for offline permutation analysis
    D0 = _C.dbg_mfma(A, B, 0)
    torch.save({"A": A.cpu(), "B": B.cpu(), "D0": D0.cpu(),
                "D1": _C.dbg_mfma(A, B, 1).cpu()},
               "gpurun_out/mfma_probe.pt")

    # staging probe: transposed image through scalar swizzled writes
    M = torch.randn(64, 64, device=dev, dtype=torch.bfloat16)
    out_t = _C.dbg_stage(M, 1)
    err_t = (out_t.float() - M.float().t()).abs().max().item()
    out_r = _C.dbg_stage(M, 0)
    err_r = (out_r.float() - M.float()).abs().max().item()
    print(f"stage rowmajor roundtrip err: {err_r:.4f} (expected 0)")
    print(f"stage transposed err vs M^T: {err_t:.4f} (expected 0)")


def probe_tr16():
    import torch
    from tiny_deepspeed_amd import _C
    dev = "cuda"
    Mr = torch.arange(64, device=dev).view(64, 1).expand(64, 64).contiguous().to(torch.bfloat16)
    Mc = torch.arange(64, device=dev).view(1, 64).expand(64, 64).contiguous().to(torch.bfloat16)
    Rr = _C.dbg_tr16(Mr)  # [dt][s][half][j][lane] -> row index read
    Rc = _C.dbg_tr16(Mc)  # -> col index read
    # hypothesis: lane l gets row 16s+8*(g4>>1)+4*half+j, col dt*32+16*(g4&1)+(l&15)
    import itertools
    ok = True
    for dt, s, half, j in itertools.product(range(2), range(4), range(2), range(4)):
        for l in range(64):
            g4 = l >> 4
            er = 16*s + 8*(g4 >> 1) + 4*half + j
            ec = dt*32 + 16*(g4 & 1) + (l & 15)
            ar = int(Rr[dt, s, half, j, l].item())
            ac = int(Rc[dt, s, half, j, l].item())
            if (ar, ac) != (er, ec):
                if ok:
                    print(f"tr16 MISMATCH first at dt{dt} s{s} h{half} j{j} l{l}: got (r{ar},c{ac}) want (r{er},c{ec})")
                ok = False
    print("tr16 hypothesis:", "OK" if ok else "FAILED")
    if not ok:
        # dump mapping for (dt,s,half)=(0,0,0)
        for j in range(4):
            print("j", j, [(int(Rr[0,0,0,j,l]), int(Rc[0,0,0,j,l])) for l in range(0,64,4)])


if __name__ == "__main__":
    main()
    probe_tr16()
