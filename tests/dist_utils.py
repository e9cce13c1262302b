"""Helpers for multi-process gloo tests (world_size 2 on CPU).

Uses the *spawn* start method: forking a pytest process that has already
run torch work (OMP / autograd engine threads) deadlocks in the child.
Worker functions must live in an importable module (tests/_dist_workers.py)
— PYTHONPATH is propagated so spawned children can unpickle them.
"""

import os
import socket

import torch.distributed as dist
import torch.multiprocessing as mp

_REPO_ROOT = os.path.abspath(os.path.join(os.path.dirname(__file__), ".."))


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _entry(rank, world, port, fn, args, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from tiny_deepspeed_amd.parallel.comm import default_comm

    default_comm(refresh=True)
    try:
        result = fn(rank, world, *args)
        q.put((rank, "ok", result))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, "err", f"{e}\n{traceback.format_exc()}"))
        raise
    finally:
        dist.destroy_process_group()


def run_distributed(fn, world=2, args=(), timeout=180):
    """Run fn(rank, world, *args) in `world` spawned processes over gloo.
    `fn` must be defined in an importable module. Returns {rank: result}."""
    prev_pp = os.environ.get("PYTHONPATH")
    os.environ["PYTHONPATH"] = (
        _REPO_ROOT if not prev_pp else f"{_REPO_ROOT}{os.pathsep}{prev_pp}"
    )
    try:
        ctx = mp.get_context("spawn")
        q = ctx.SimpleQueue()
        port = free_port()
        procs = [
            ctx.Process(target=_entry, args=(r, world, port, fn, args, q))
            for r in range(world)
        ]
        for p in procs:
            p.start()
        results = {}
        for _ in range(world):
            rank, status, payload = q.get()
            if status == "err":
                for p in procs:
                    p.terminate()
                raise RuntimeError(f"rank {rank} failed:\n{payload}")
            results[rank] = payload
        for p in procs:
            p.join(timeout=timeout)
    finally:
        if prev_pp is None:
            os.environ.pop("PYTHONPATH", None)
        else:
            os.environ["PYTHONPATH"] = prev_pp
    return results
