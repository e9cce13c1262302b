"""CommContext unit behavior without an initialized process group (world=1):
every collective is a no-op returning its input; sync()/wait_gather() are
safe with nothing outstanding; bucketed broadcast is a no-op."""

import torch

from tiny_deepspeed_amd.parallel.comm import CommContext


def test_world1_noop_collectives():
    comm = CommContext()
    assert comm.world_size == 1 and comm.rank == 0
    t = torch.randn(8)
    ref = t.clone()
    assert comm.all_reduce_avg(t) is t
    assert torch.equal(t, ref)
    assert comm.reduce_avg_to(t, 0) is t
    assert comm.broadcast(t, 0) is t
    assert comm.gather_broadcast(t, 0) is t
    comm.broadcast_bucketed([(t, 0)])
    assert torch.equal(t, ref)
    comm.sync()
    comm.wait_gather()
    s = comm.all_reduce_scalar_avg(torch.tensor(3.0))
    assert s.item() == 3.0
