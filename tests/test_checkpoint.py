"""Checkpoint save/load round trip (capability the reference lacks,
SURVEY.md §5.4): model weights, optimizer state (incl. fp32 moments and the
global step counter) restore exactly and training resumes identically."""

import os

import torch

from tiny_deepspeed_amd import Single, AdamW
from tiny_deepspeed_amd.models import GPTConfig, GPT2Model
from tiny_deepspeed_amd.utils.checkpoint import save_checkpoint, load_checkpoint


def _tiny():
    cfg = GPTConfig(n_layer=2, n_head=2, n_embd=32, block_size=32,
                    vocab_size=64)
    torch.manual_seed(0)
    model = Single(GPT2Model(cfg))
    opt = AdamW(model.named_parameters(), lr=1e-3)
    return cfg, model, opt


def _step(model, opt, seed):
    g = torch.Generator().manual_seed(seed)
    x = torch.randint(0, 64, (2, 16), generator=g)
    y = torch.randint(0, 64, (2, 16), generator=g)
    _, loss = model(x, y)
    loss.backward()
    opt.step()
    return loss.item()


def test_checkpoint_roundtrip(tmp_path):
    cfg, model, opt = _tiny()
    for i in range(3):
        _step(model, opt, i)
    path = os.path.join(tmp_path, "ckpt.pt")
    save_checkpoint(path, model, opt, step=3)

    # continue training from the live state
    ref_losses = [_step(model, opt, 10 + i) for i in range(2)]

    # fresh model + optimizer, restore, replay the same steps
    cfg2, model2, opt2 = _tiny()
    step = load_checkpoint(path, model2, opt2)
    assert step == 3
    assert opt2.t == opt.t - 2  # t advanced by the 2 extra steps above
    new_losses = [_step(model2, opt2, 10 + i) for i in range(2)]
    assert new_losses == ref_losses


def test_checkpoint_model_only(tmp_path):
    cfg, model, opt = _tiny()
    _step(model, opt, 0)
    path = os.path.join(tmp_path, "m.pt")
    save_checkpoint(path, model, step=1)
    _, model2, _ = _tiny()
    load_checkpoint(path, model2)
    for (n1, p1), (n2, p2) in zip(model.named_parameters(),
                                  model2.named_parameters()):
        assert n1 == n2
        assert torch.equal(p1, p2), n1
