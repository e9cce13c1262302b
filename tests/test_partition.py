"""Partitioner: contiguity, evenness, meta planning, malloc, capacity check."""

from collections import OrderedDict

import pytest
import torch

from tiny_deepspeed_amd import partition_tensors


def _named(sizes):
    return OrderedDict(
        (f"p{i}", torch.empty(s, device="meta")) for i, s in enumerate(sizes)
    )


def test_contiguous_assignment():
    parts, _ = partition_tensors(_named([10, 10, 10, 10]), ["cpu"] * 2)
    vals = list(parts.values())
    assert vals == sorted(vals)  # contiguous, monotone
    assert set(vals) == {0, 1}


def test_roughly_even_split():
    parts, _ = partition_tensors(_named([100] * 8), ["cpu"] * 4)
    from collections import Counter

    counts = Counter(parts.values())
    assert all(counts[r] == 2 for r in range(4))


def test_works_on_meta_and_malloc():
    named = _named([4, 4])
    parts, tensors = partition_tensors(named, ["cpu", "cpu"], malloc=True)
    assert tensors is not None
    for n, t in tensors.items():
        assert not t.is_meta
        assert t.device.type == "cpu"


def test_empty_part_warns():
    with pytest.warns(UserWarning):
        partition_tensors(_named([100]), ["cpu"] * 4)


def test_capacity_check_raises():
    named = OrderedDict(a=torch.empty(1000, device="meta", dtype=torch.float32))
    with pytest.raises(RuntimeError):
        partition_tensors(named, ["cpu"], capacity_bytes=100)


def test_evenness_priority_bounds():
    with pytest.raises(ValueError):
        partition_tensors(_named([4]), ["cpu"], evenness_priority=2.0)


def test_model_partition_covers_all_params():
    from tiny_deepspeed_amd.models import GPTConfig, GPT2Model

    with torch.device("meta"):
        m = GPT2Model(GPTConfig(n_layer=2, n_head=2, n_embd=32, vocab_size=128,
                                block_size=64))
    named = OrderedDict(m.named_parameters())
    parts, _ = partition_tensors(named, ["cpu"] * 3)
    assert set(parts.keys()) == set(named.keys())
    loads = [0, 0, 0]
    for n, r in parts.items():
        loads[r] += named[n].numel()
    total = sum(loads)
    assert max(loads) < 0.75 * total  # no rank hoards everything
