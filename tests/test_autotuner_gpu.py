"""Tuned dispatch on hardware: the default tuner must be live, time real
{CDNA4 kernel, torch} candidate pairs with hipEvents, and the hand-written
kernels must win their home shapes (the north star requires the HIP path to
be the one that actually runs)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

from tiny_deepspeed_amd import ops
from tiny_deepspeed_amd.ops import _ext
from tiny_deepspeed_amd.ops.autotuner import RuntimeAutoTuner


@pytest.fixture(scope="module", autouse=True)
def _require_ext():
    assert torch.cuda.is_available()
    assert _ext.ext_available()


def test_default_tuner_live_on_gpu(monkeypatch):
    monkeypatch.delenv("TDSA_AUTOTUNE", raising=False)
    assert ops.default_tuner() is not None


def test_layernorm_tuned_dispatch_picks_hip():
    tuner = RuntimeAutoTuner(warmup=3, iters=10)
    x = torch.randn(4096, 1024, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(1024, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(1024, device="cuda", dtype=torch.bfloat16)
    y, mean, rstd = ops.layernorm_fwd(x, w, b, tuner=tuner)
    ref = torch.nn.functional.layer_norm(x.float(), (1024,), w.float(), b.float())
    assert (y.float() - ref).abs().max().item() < 2e-2 * ref.abs().max().item()
    choices = tuner.choices()
    assert any(k[0] == "ln_fwd" for k in choices)
    won = [v for k, v in choices.items() if k[0] == "ln_fwd"][0]
    assert won == "ln_fwd_hip", f"library beat the CDNA4 LN kernel: {choices}"


def test_attention_tuned_dispatch_picks_hip():
    tuner = RuntimeAutoTuner(warmup=2, iters=5)
    torch.manual_seed(0)
    q, k, v = (torch.randn(2, 4, 256, 64, device="cuda", dtype=torch.bfloat16)
               for _ in range(3))
    from tiny_deepspeed_amd.ops import attention as attn
    o, lse = tuner.choose("attn_fwd", [attn.attn_fwd_hip,
                                       attn.attn_fwd_composite],
                          q, k, v, 0.125)
    ref, _ = attn._composite_fwd(q, k, v, 0.125)
    assert (o.float() - ref.float()).abs().max().item() < 2e-2
    won = [v_ for k_, v_ in tuner.choices().items() if k_[0] == "attn_fwd"][0]
    assert won == "attn_fwd_hip", f"composite beat the fused kernel: {tuner.choices()}"


def test_linear_dw_tuned_dispatch_runs():
    # dW candidate list: {MFMA TN kernel (if built), hipBLASLt}; whichever
    # wins, the result must match the library and a choice must be recorded
    # when there are >1 candidates.
    tuner = RuntimeAutoTuner(warmup=2, iters=5)
    dy = torch.randn(64, 2048, 1024, device="cuda", dtype=torch.bfloat16) * 0.02
    x = torch.randn(64, 2048, 512, device="cuda", dtype=torch.bfloat16)
    dw = ops.linear_weight_grad(dy, x, tuner=tuner)
    ref = torch.matmul(dy.reshape(-1, 1024).t().float(), x.reshape(-1, 512).float())
    err = (dw.float() - ref).abs().max().item()
    assert err < 0.5, f"dW err {err}"  # bf16 accum tolerance at M=128k
    if hasattr(_ext.get_ext(), "gemm_tn"):
        assert any(k[0] == "linear_dw" for k in tuner.choices())


def test_model_step_populates_default_tuner(monkeypatch):
    monkeypatch.delenv("TDSA_AUTOTUNE", raising=False)
    import tiny_deepspeed_amd.ops.autotuner as at
    monkeypatch.setattr(at, "_DEFAULT_TUNER", None)
    from tiny_deepspeed_amd.models import GPTConfig, GPT2Model
    from tiny_deepspeed_amd import Single, AdamW
    cfg = GPTConfig(n_layer=2, n_head=4, n_embd=256, block_size=256,
                    vocab_size=1024)
    model = GPT2Model(cfg).to(device="cuda", dtype=torch.bfloat16)
    wrapped = Single(model)
    opt = AdamW(wrapped.named_parameters(), lr=1e-4)
    x = torch.randint(0, 1024, (2, 256), device="cuda")
    _, loss = wrapped(x, x)
    loss.backward()
    opt.step()
    tuner = at._DEFAULT_TUNER
    assert tuner is not None
    ops_tuned = {k[0] for k in tuner.choices()}
    # the hot ops all dispatched through measured choice (the model has
    # bias=False linears so no linear_db; blocks use the fused-res LN)
    for name in ("gelu_fwd", "gelu_bwd", "ce_fwd", "ce_bwd", "ln_dx",
                 "attn_fwd_packed", "attn_bwd_packed", "emb_fwd", "emb_bwd"):
        assert name in ops_tuned, f"{name} not tuned: {sorted(ops_tuned)}"
    assert ops_tuned & {"ln_fwd", "ln_fwd_res"}, sorted(ops_tuned)
    torch.cuda.synchronize()
