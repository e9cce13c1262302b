"""Property-based invariants for the greedy contiguous partitioner
(hypothesis): every tensor assigned, ranks contiguous & monotone, loads
bounded, meta-planning never allocates."""

from collections import OrderedDict

import torch
from hypothesis import given, settings, strategies as st

from tiny_deepspeed_amd import partition_tensors


@st.composite
def tensor_sets(draw):
    n = draw(st.integers(min_value=1, max_value=40))
    sizes = [draw(st.integers(min_value=1, max_value=5000)) for _ in range(n)]
    ranks = draw(st.integers(min_value=1, max_value=8))
    evenness = draw(st.sampled_from([0.0, 0.5, 1.0]))
    return sizes, ranks, evenness


@given(tensor_sets())
@settings(max_examples=60, deadline=None)
def test_partition_invariants(case):
    sizes, n_ranks, evenness = case
    with torch.device("meta"):
        named = OrderedDict(
            (f"t{i}", torch.empty(s)) for i, s in enumerate(sizes)
        )
    parts, out = partition_tensors(
        named, ranks_map=["cpu"] * n_ranks, evenness_priority=evenness,
    )
    assert out is None
    # every tensor assigned to a valid rank
    assert set(parts.keys()) == set(named.keys())
    ranks = list(parts.values())
    assert all(0 <= r < n_ranks for r in ranks)
    # contiguity: rank sequence is non-decreasing (greedy walk)
    assert ranks == sorted(ranks)
    # no rank's load exceeds total (sanity) and the max part is bounded by
    # ideal + the largest tensor (greedy guarantee)
    total = sum(sizes)
    ideal = total / n_ranks
    loads = [0] * n_ranks
    for (name, r), s in zip(parts.items(), sizes):
        loads[r] += s
    assert max(loads) <= ideal + max(sizes) + 1
