"""GPT-2 model: shapes, causality, wrapper swap integrity."""

import torch
import torch.nn as nn

import tiny_deepspeed_amd as tdsa
from tiny_deepspeed_amd import modules as tmods
from tiny_deepspeed_amd.models import GPTConfig, GPT2Model


def _small_cfg(**kw):
    return GPTConfig(block_size=32, vocab_size=64, n_layer=2, n_head=2,
                     n_embd=16, **kw)


def test_forward_shapes_and_loss():
    torch.manual_seed(0)
    model = GPT2Model(_small_cfg())
    x = torch.randint(0, 64, (3, 32))
    y = torch.randint(0, 64, (3, 32))
    logits, loss = model(x, y)
    assert logits.shape == (3, 32, 64)
    assert loss.dim() == 0 and torch.isfinite(loss)
    logits2, loss2 = model(x)
    assert loss2 is None


def test_causality():
    torch.manual_seed(0)
    model = GPT2Model(_small_cfg()).eval()
    x = torch.randint(0, 64, (1, 32))
    with torch.no_grad():
        base, _ = model(x)
        x2 = x.clone()
        x2[0, 20:] = (x2[0, 20:] + 1) % 64  # perturb the future
        pert, _ = model(x2)
    assert torch.allclose(base[0, :19], pert[0, :19], atol=1e-5)
    assert not torch.allclose(base[0, 20:], pert[0, 20:], atol=1e-5)


def test_attention_backends_agree():
    torch.manual_seed(0)
    m1 = GPT2Model(_small_cfg(attention="fused"))
    m2 = GPT2Model(_small_cfg(attention="math"))
    m2.load_state_dict(m1.state_dict())
    x = torch.randint(0, 64, (2, 32))
    l1, _ = m1(x)
    l2, _ = m2(x)
    assert torch.allclose(l1, l2, atol=1e-4)


def test_block_size_assert():
    model = GPT2Model(_small_cfg())
    x = torch.randint(0, 64, (1, 33))
    try:
        model(x)
        assert False, "expected assertion"
    except AssertionError:
        pass


def test_presets():
    small = GPTConfig.named("gpt2-small")
    xl = GPTConfig.named("gpt2-xl")
    assert small.n_layer == 12 and small.n_embd == 768
    assert xl.n_layer == 48 and xl.n_embd == 1600


def test_single_wrapper_swaps_and_preserves_params():
    torch.manual_seed(0)
    raw = GPT2Model(_small_cfg())
    orig = {n: p for n, p in raw.named_parameters()}
    model = tdsa.Single(raw)
    # same Parameter objects, swapped module classes
    for n, p in model.named_parameters():
        assert p is orig[n]
    kinds = {type(m) for m in model.module.modules()}
    assert tmods.Linear in kinds
    assert tmods.LayerNorm in kinds
    assert tmods.Embedding in kinds
    assert not any(type(m) is nn.Linear for m in model.module.modules())


def test_wrapper_matches_raw_model_loss():
    torch.manual_seed(0)
    cfg = _small_cfg()
    raw = GPT2Model(cfg)
    import copy

    wrapped = tdsa.Single(copy.deepcopy(raw))
    x = torch.randint(0, 64, (2, 32))
    y = torch.randint(0, 64, (2, 32))
    l1, loss1 = raw(x, y)
    l2, loss2 = wrapped(x, y)
    assert torch.allclose(loss1, loss2, atol=1e-5)
    loss1.backward()
    loss2.backward()
    for (n, p1), (_, p2) in zip(raw.named_parameters(),
                                wrapped.named_parameters()):
        assert torch.allclose(p1.grad, p2.grad, atol=1e-4), n


def test_unsupported_module_errors():
    class Weird(nn.Module):
        def __init__(self):
            super().__init__()
            self.w = nn.Parameter(torch.randn(4))
            self.lin = nn.Linear(4, 4)

        def forward(self, x):
            return self.lin(x) + self.w

    try:
        tdsa.Single(Weird())
        assert False, "expected RuntimeError"
    except RuntimeError as e:
        assert "unsupported" in str(e)


def test_model_with_dropout_trains_cpu():
    """config.dropout > 0 end to end on the composite CPU path (the GPU
    path fuses dropout in-kernel; both share the model-level plumbing)."""
    from tiny_deepspeed_amd import Single, AdamW

    torch.manual_seed(0)
    cfg = GPTConfig(n_layer=2, n_head=2, n_embd=32, block_size=32,
                    vocab_size=64, dropout=0.1)
    model = Single(GPT2Model(cfg))
    opt = AdamW(model.named_parameters(), lr=1e-3)
    g = torch.Generator().manual_seed(1)
    x = torch.randint(0, 64, (2, 32), generator=g)
    y = torch.randint(0, 64, (2, 32), generator=g)
    losses = []
    for _ in range(5):
        _, loss = model(x, y)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0]
    # eval() disables dropout: two eval forwards are identical
    model.module.eval()
    l1 = model(x, y)[1].item()
    l2 = model(x, y)[1].item()
    assert l1 == l2
    model.module.train()
