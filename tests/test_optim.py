"""Optimizer math vs torch.optim references (fp32), bf16 master-weight path,
state_dict roundtrip, and the fixed global step counter."""

import pytest
import torch

from tiny_deepspeed_amd import AdamW, SGD


def _params(seed=0, dtype=torch.float32):
    torch.manual_seed(seed)
    lin = torch.nn.Linear(8, 8).to(dtype)
    return lin


def _run(opt_factory, torch_factory, iters=5):
    ours_mod = _params(0)
    ref_mod = _params(0)
    ours = opt_factory(ours_mod)
    ref = torch_factory(ref_mod)
    torch.manual_seed(42)
    for _ in range(iters):
        x = torch.randn(4, 8)
        (ours_mod(x).square().mean()).backward()
        (ref_mod(x).square().mean()).backward()
        ours.step()
        ref.step()
        ref.zero_grad()
    for p1, p2 in zip(ours_mod.parameters(), ref_mod.parameters()):
        assert torch.allclose(p1, p2, atol=1e-6), (p1 - p2).abs().max()


def test_adamw_matches_torch():
    _run(
        lambda m: AdamW(m.named_parameters(), lr=1e-2, betas=(0.9, 0.999),
                        eps=1e-8, weight_decay=0.1),
        lambda m: torch.optim.AdamW(m.parameters(), lr=1e-2, betas=(0.9, 0.999),
                                    eps=1e-8, weight_decay=0.1),
    )


def test_adamw_amsgrad_matches_torch():
    _run(
        lambda m: AdamW(m.named_parameters(), lr=1e-2, weight_decay=0.1,
                        amsgrad=True),
        lambda m: torch.optim.AdamW(m.parameters(), lr=1e-2, weight_decay=0.1,
                                    amsgrad=True),
    )


def test_sgd_momentum_matches_torch():
    _run(
        lambda m: SGD(m.named_parameters(), lr=1e-2, momentum=0.9,
                      weight_decay=0.01),
        lambda m: torch.optim.SGD(m.parameters(), lr=1e-2, momentum=0.9,
                                  weight_decay=0.01),
    )


def test_sgd_nesterov_matches_torch():
    _run(
        lambda m: SGD(m.named_parameters(), lr=1e-2, momentum=0.9, nesterov=True),
        lambda m: torch.optim.SGD(m.parameters(), lr=1e-2, momentum=0.9,
                                  nesterov=True),
    )


def test_global_step_counter_not_per_param():
    # the reference advanced t per parameter; ours is global
    m = _params(0)
    opt = AdamW(m.named_parameters(), lr=1e-2)
    x = torch.randn(2, 8)
    m(x).sum().backward()
    opt.step()
    assert opt.t == 1
    m(x).sum().backward()
    opt.step()
    assert opt.t == 2


def test_bf16_master_weights():
    torch.manual_seed(0)
    mod = torch.nn.Linear(16, 16).to(torch.bfloat16)
    opt = AdamW(mod.named_parameters(), lr=1e-3)
    assert all(v.dtype == torch.float32 for v in opt.master.values())
    x = torch.randn(4, 16, dtype=torch.bfloat16)
    mod(x).float().square().mean().backward()
    before = {n: v.clone() for n, v in opt.master.items()}
    opt.step()
    for n, v in opt.master.items():
        assert not torch.equal(v, before[n])
        # bf16 param tracks the master copy
        p = dict(mod.named_parameters())[n]
        assert torch.allclose(p.float(), v, atol=1e-2)


def test_state_dict_roundtrip():
    m1 = _params(0)
    opt1 = AdamW(m1.named_parameters(), lr=1e-2)
    x = torch.randn(4, 8)
    m1(x).sum().backward()
    opt1.step()
    sd = opt1.state_dict()
    m2 = _params(1)
    opt2 = AdamW(m2.named_parameters(), lr=1e-2)
    opt2.load_state_dict(sd)
    assert opt2.t == opt1.t
    for n in opt1.exp_avg:
        assert torch.allclose(opt1.exp_avg[n], opt2.exp_avg[n])
        assert torch.allclose(opt1.exp_avg_sq[n], opt2.exp_avg_sq[n])


@pytest.mark.parametrize("nesterov", [False, True])
def test_sgd_matches_torch_optim(nesterov):
    torch.manual_seed(0)
    p1 = torch.nn.Parameter(torch.randn(64, 32))
    p2 = torch.nn.Parameter(p1.detach().clone())
    ours = SGD([("p", p1)], lr=0.05, momentum=0.9, weight_decay=0.01,
               nesterov=nesterov)
    ref = torch.optim.SGD([p2], lr=0.05, momentum=0.9, weight_decay=0.01,
                          nesterov=nesterov)
    for _ in range(4):
        g = torch.randn_like(p1)
        p1.grad = g.clone()
        p2.grad = g.clone()
        ours.step()
        ref.step()
    assert torch.allclose(p1, p2, atol=1e-6), (p1 - p2).abs().max()
