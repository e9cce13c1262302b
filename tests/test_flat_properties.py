"""Property-based invariants for the flat-bucket ZeRO-2 engine
(hypothesis): arbitrary parameter shapes and bucket sizes must produce
buckets whose views tile the parameters exactly (no overlap, no gap,
values preserved), padded to the world size, and the publish protocol
must deliver exactly the concatenated grads into the shards."""

import numpy as np
import torch
import torch.nn as nn
from hypothesis import given, settings, strategies as st

from tiny_deepspeed_amd.parallel.comm import CommContext
from tiny_deepspeed_amd.parallel.flat import FlatShardEngine


class _FakeComm(CommContext):
    """World-`w` arithmetic without a process group: reduce_scatter takes
    this rank's slice (world-1 semantics generalized for slot math)."""

    def __init__(self, rank, world):
        super().__init__()
        self.rank = rank
        self.world_size = world
        self.fired = []

    def reduce_scatter_avg(self, out_shard, in_flat):
        n = out_shard.numel()
        out_shard.copy_(in_flat[self.rank * n:(self.rank + 1) * n])
        self.fired.append(n)
        return out_shard

    def all_gather_flat(self, out_flat, in_shard):
        return out_flat


@st.composite
def param_sets(draw):
    n = draw(st.integers(min_value=1, max_value=12))
    shapes = []
    for _ in range(n):
        dims = draw(st.integers(min_value=1, max_value=2))
        shapes.append(tuple(draw(st.integers(min_value=1, max_value=64))
                            for _ in range(dims)))
    bucket_bytes = draw(st.sampled_from([64, 1024, 1 << 20]))
    world = draw(st.integers(min_value=1, max_value=8))
    rank = draw(st.integers(min_value=0, max_value=world - 1))
    return shapes, bucket_bytes, world, rank


@settings(max_examples=60, deadline=None)
@given(param_sets())
def test_flat_engine_tiling_and_publish(spec):
    shapes, bucket_bytes, world, rank = spec
    torch.manual_seed(0)
    params = []
    for i, shp in enumerate(shapes):
        p = nn.Parameter(torch.randn(*shp))
        p._tdsa_name = f"p{i}"
        params.append((f"p{i}", p))
    originals = {n: p.detach().clone() for n, p in params}

    comm = _FakeComm(rank, world)
    eng = FlatShardEngine(list(params), comm, bucket_bytes=bucket_bytes)

    # 1. every param is a view of exactly one bucket; values preserved
    for n, p in params:
        torch.testing.assert_close(p.data, originals[n])
        base = p.data.untyped_storage().data_ptr()
        owners = [b for b in eng.buckets
                  if b.flat.untyped_storage().data_ptr() == base]
        assert len(owners) == 1
    # 2. bucket layout: offsets tile [0, total) without overlap; padding
    #    to a multiple of world, zero-filled
    for b in eng.buckets:
        spans = sorted(b.offs.values())
        pos = 0
        for off, k in spans:
            assert off == pos
            pos += k
        assert b.flat.numel() % world == 0
        assert pos <= b.flat.numel() < pos + world  # minimal padding
        if b.flat.numel() > pos:
            assert b.flat[pos:].abs().sum().item() == 0
        assert b.grad_shard.numel() * world == b.grad.numel()
    # 3. publish all grads armed -> every bucket fires once, and the shard
    #    holds this rank's slice of the concatenated grads
    grads = {n: torch.randn_like(p) for n, p in params}
    for n, p in params:
        eng.publish(p, grads[n], armed=True)
    for b in eng.buckets:
        assert b.fired
        flat_ref = torch.zeros_like(b.grad)
        for n, (off, k) in b.offs.items():
            flat_ref[off:off + k] = grads[n].reshape(-1)
        ns = b.grad_shard.numel()
        torch.testing.assert_close(
            b.grad_shard, flat_ref[rank * ns:(rank + 1) * ns])
    # 4. a second armed round re-fires (latch reset)
    for n, p in params:
        eng.publish(p, grads[n], armed=True)
    assert len(comm.fired) == 2 * len(eng.buckets)
