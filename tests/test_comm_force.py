"""TDSA_COMM_FORCE=1: collectives must EXECUTE at world size 1 instead of
early-returning — this is what lets 1-GPU leases exercise the real
RCCL/comm-stream path. Here the same flag is proven on gloo world-1:
collectives run (values preserved — averaging over world 1 is identity) and
a forced DDP/ZeRO training run matches the unforced one exactly."""

import os

import pytest
import torch
import torch.distributed as dist

from tests.dist_utils import free_port
from tiny_deepspeed_amd.parallel.comm import CommContext


@pytest.fixture
def world1_pg(monkeypatch):
    monkeypatch.setenv("MASTER_ADDR", "127.0.0.1")
    monkeypatch.setenv("MASTER_PORT", str(free_port()))
    monkeypatch.setenv("TDSA_COMM_FORCE", "1")
    dist.init_process_group("gloo", rank=0, world_size=1)
    yield
    dist.destroy_process_group()


def test_forced_collectives_execute_world1(world1_pg):
    comm = CommContext()
    assert comm.force and not comm._inactive()
    t = torch.randn(64)
    ref = t.clone()
    comm.all_reduce_avg(t)
    comm.sync()
    torch.testing.assert_close(t, ref)  # avg over world 1 == identity
    comm.reduce_avg_to(t, 0)
    comm.broadcast(t, 0)
    comm.gather_broadcast(t, 0)
    comm.sync()
    torch.testing.assert_close(t, ref)
    s = comm.all_reduce_scalar_avg(torch.tensor(5.0))
    assert s.item() == 5.0


def test_forced_bucketed_broadcast_world1(world1_pg):
    comm = CommContext()
    small = [torch.randn(100) for _ in range(6)]
    big = torch.randn(1 << 21)
    refs = [t.clone() for t in small] + [big.clone()]
    comm.broadcast_bucketed([(t, 0) for t in small] + [(big, 0)],
                            bucket_bytes=1 << 12)
    comm.sync()
    for t, r in zip(small + [big], refs):
        torch.testing.assert_close(t, r)


def _train_losses(force, steps=3):
    os.environ["TDSA_COMM_FORCE"] = "1" if force else "0"
    try:
        from tiny_deepspeed_amd.models import GPTConfig, GPT2Model
        from tiny_deepspeed_amd import DDP, DDPAdamW

        torch.manual_seed(7)
        cfg = GPTConfig(n_layer=2, n_head=2, n_embd=64, block_size=64,
                        vocab_size=128)
        model = GPT2Model(cfg)
        comm = CommContext()
        assert comm.force == force
        wrapped = DDP(model, comm=comm)
        opt = DDPAdamW(wrapped.named_parameters(), lr=1e-3, comm=comm)
        g = torch.Generator().manual_seed(3)
        x = torch.randint(0, 128, (2, 64), generator=g)
        y = torch.randint(0, 128, (2, 64), generator=g)
        losses = []
        for _ in range(steps):
            wrapped.require_backward_grad_sync = True
            _, loss = wrapped(x, y)
            loss.backward()
            opt.step()
            losses.append(loss.item())
        return losses
    finally:
        os.environ.pop("TDSA_COMM_FORCE", None)


def test_forced_ddp_training_matches_unforced(world1_pg):
    forced = _train_losses(True)
    plain = _train_losses(False)
    assert forced == pytest.approx(plain, rel=0, abs=0), (forced, plain)
    assert forced[-1] < forced[0]


def test_single_communicator_mode(world1_pg, monkeypatch):
    """TDSA_COMM_SINGLE=1: both channels share one communicator and one
    stream — the safety fallback for concurrent-communicator hazards."""
    monkeypatch.setenv("TDSA_COMM_SINGLE", "1")
    comm = CommContext()
    assert comm.single
    assert comm.pg["gather"] is comm.pg["reduce"]
    t = torch.randn(32)
    ref = t.clone()
    comm.gather_broadcast(t, 0)
    comm.wait_gather()
    comm.all_reduce_avg(t)
    comm.sync()
    torch.testing.assert_close(t, ref)
