"""LayerNorm op functions (last-dim normalization, affine w/ bias).

Capability parity with the reference's Triton kernels
(``/root/reference/tiny_deepspeed/core/module/ops/layernorm.py:46-298``),
re-designed as CDNA4 HIP kernels (csrc/layernorm.hip):

  layernorm_fwd   : per-row mean/rstd in fp32, y = (x-mu)*rstd*w + b
  layernorm_dx    : dx = (w*dy - (xhat*c1 + c2)) * rstd, plus per-workgroup
                    partial dw/db accumulated into fp32 stripe buffers
                    (replacing the reference's spin-lock atomic_cas scheme,
                    SURVEY.md 2.10A, with a conflict-free two-pass reduce)
  layernorm_dwdb  : column-reduce of the stripe buffers -> dw[N], db[N]

Dispatch follows the reference's candidate-list architecture
(``ops/layernorm.py:46-74`` routes through RuntimeAutoTuner): on GPU the
{CDNA4 kernel, torch composite} pair is timed per shape by the tuner and the
winner cached. CPU uses the torch fp32 reference (also the test oracle).
"""

import torch

from . import _ext
from .autotuner import default_tuner

# Number of fp32 partial-stripe rows for the dw/db reduction. Each dx
# workgroup accumulates its rows into stripe (block_id % N_STRIPES); the
# dwdb kernel column-reduces the stripes. 1024 blocks cover 256 CUs well.
N_STRIPES = 256


# --- forward candidates ----------------------------------------------------
def ln_fwd_hip(x, weight, bias, eps):
    return _ext.get_ext().layernorm_fwd(x.contiguous(), weight, bias, eps)


def ln_fwd_torch(x, weight, bias, eps):
    xf = x.float()
    mean = xf.mean(dim=-1)
    var = xf.var(dim=-1, unbiased=False)
    rstd = (var + eps).rsqrt()
    y = (xf - mean.unsqueeze(-1)) * rstd.unsqueeze(-1)
    y = y * weight.float() + bias.float()
    return y.to(x.dtype), mean, rstd


def layernorm_fwd(x, weight, bias, eps=1e-5, tuner=None):
    """Returns (y, mean, rstd); mean/rstd are fp32 per-row tensors."""
    if not _ext.use_native(x):
        return ln_fwd_torch(x, weight, bias, eps)
    tuner = tuner if tuner is not None else default_tuner()
    if tuner is not None:
        return tuner.choose("ln_fwd", [ln_fwd_hip, ln_fwd_torch],
                            x, weight, bias, eps)
    return ln_fwd_hip(x, weight, bias, eps)


def ln_fwd_res_hip(x, res, weight, bias, eps):
    y, mean, rstd, h = _ext.get_ext().layernorm_fwd(
        x.contiguous(), weight, bias, eps, res.contiguous()
    )
    return h, y, mean, rstd


def ln_fwd_res_torch(x, res, weight, bias, eps):
    h = x + res
    y, mean, rstd = ln_fwd_torch(h, weight, bias, eps)
    return h, y, mean, rstd


def layernorm_fwd_res(x, res, weight, bias, eps=1e-5, tuner=None):
    """Fused residual + LayerNorm: h = x + res; y = LN(h).
    Returns (h, y, mean, rstd) — h is the residual stream, written once by
    the kernel instead of a separate elementwise add pass."""
    if not _ext.use_native(x):
        return ln_fwd_res_torch(x, res, weight, bias, eps)
    tuner = tuner if tuner is not None else default_tuner()
    if tuner is not None:
        return tuner.choose("ln_fwd_res", [ln_fwd_res_hip, ln_fwd_res_torch],
                            x, res, weight, bias, eps)
    return ln_fwd_res_hip(x, res, weight, bias, eps)


# --- backward-dx candidates ------------------------------------------------
def ln_dx_hip(dy, x, weight, mean, rstd, dh):
    dx, pdw, pdb = _ext.get_ext().layernorm_bwd_dx(
        dy.contiguous(), x.contiguous(), weight, mean, rstd, N_STRIPES, dh
    )
    return dx, (pdw, pdb)


def ln_dx_torch(dy, x, weight, mean, rstd, dh):
    xf = x.float()
    dyf = dy.float()
    wf = weight.float()
    xhat = (xf - mean.unsqueeze(-1)) * rstd.unsqueeze(-1)
    wdy = wf * dyf
    c1 = (xhat * wdy).mean(dim=-1, keepdim=True)
    c2 = wdy.mean(dim=-1, keepdim=True)
    dx = (wdy - (xhat * c1 + c2)) * rstd.unsqueeze(-1)
    if dh is not None:
        dx = dx + dh.float()
    return dx.to(x.dtype), (dy, x, mean, rstd)


def layernorm_dx(dy, x, weight, mean, rstd, dh=None, tuner=None):
    """Input gradient (+ optional fused add of the residual-stream grad dh).
    On the HIP path this also produces stripe partials for dw/db (returned
    as an opaque workspace consumed by layernorm_dwdb; the torch candidate
    returns the saved tensors for recompute instead — both accepted)."""
    if not _ext.use_native(dy):
        return ln_dx_torch(dy, x, weight, mean, rstd, dh)
    tuner = tuner if tuner is not None else default_tuner()
    if tuner is not None:
        return tuner.choose("ln_dx", [ln_dx_hip, ln_dx_torch],
                            dy, x, weight, mean, rstd, dh)
    return ln_dx_hip(dy, x, weight, mean, rstd, dh)


def layernorm_dwdb(workspace, dtype=None, tuner=None):
    """Reduce partials to (dw, db)."""
    if len(workspace) == 2:  # HIP dx path: fp32 stripe buffers
        pdw, pdb = workspace
        if _ext.use_native(pdw):
            ext = _ext.get_ext()
            dw, db = ext.layernorm_bwd_dwdb(pdw, pdb)
            if dtype is not None:
                dw, db = dw.to(dtype), db.to(dtype)
            return dw, db
        dw = pdw.sum(dim=0)
        db = pdb.sum(dim=0)
    else:  # torch dx path: recompute from saved tensors
        dy, x, mean, rstd = workspace
        xf = x.float().reshape(-1, x.shape[-1])
        dyf = dy.float().reshape(-1, dy.shape[-1])
        xhat = (xf - mean.reshape(-1, 1)) * rstd.reshape(-1, 1)
        dw = (dyf * xhat).sum(dim=0)
        db = dyf.sum(dim=0)
    if dtype is not None:
        dw, db = dw.to(dtype), db.to(dtype)
    return dw, db
