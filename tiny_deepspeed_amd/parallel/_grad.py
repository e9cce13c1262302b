"""Shared gradient-publication logic for the parallel strategies.

Strategy modules manage ``param.grad`` directly (their autograd Functions
return None for parameter gradients) so that the async collective's
in-place result is guaranteed to be the tensor the optimizer reads —
no dependence on autograd's AccumulateGrad steal-vs-clone behavior.

Gradient accumulation (require_backward_grad_sync=False iterations) is
handled correctly: the collective covers the ACCUMULATED gradient, not
just the last microbatch's contribution (the reference reduces only the
current dW — SURVEY.md 2.9 note).
"""

from .wrapper import take_sync

# publication modes
ALLREDUCE = "allreduce"        # DDP: average everywhere, grad replicated
REDUCE_KEEP = "reduce_keep"    # ZeRO-1: average onto owner, local grad kept
REDUCE_SHARD = "reduce_shard"  # ZeRO-2/3: average onto owner, dropped elsewhere
FLAT = "flat"                  # flat-bucket ZeRO-2: slot into the engine's
#                                bucket; one reduce_scatter per full bucket


def publish_grad(comm, param, dw, mode):
    """Accumulate dw into param.grad, launching the strategy's collective
    (async, comm stream) when the once-per-iteration latch is armed."""
    if dw is None:
        return
    if mode == FLAT:
        # the engine owns accumulation, bucket completion and the collective
        param._tdsa_engine.publish(param, dw, take_sync(param))
        return
    # ZeRO-3 non-owners have 0-numel param storage: full-shape gradients
    # accumulate in a side slot instead of .grad (torch enforces shape).
    sharded_param = param.numel() == 0
    prev = getattr(param, "_tdsa_accum", None) if sharded_param else param.grad
    if prev is not None:
        dw = prev + dw
    if not take_sync(param):
        if sharded_param:
            param._tdsa_accum = dw
        else:
            param.grad = dw
        return
    if mode == ALLREDUCE:
        param.grad = comm.all_reduce_avg(dw)
    elif mode == REDUCE_KEEP:
        comm.reduce_avg_to(dw, param._tdsa_owner)
        param.grad = dw
    elif mode == REDUCE_SHARD:
        comm.reduce_avg_to(dw, param._tdsa_owner)
        if comm.rank == param._tdsa_owner:
            param.grad = dw
        else:
            # REAL gradient release: the last host reference dies here; the
            # stream-ordered allocator reclaims the block once RCCL is done
            # (record_stream was called by the comm context). On gloo the
            # host reference is pinned until sync() instead.
            comm.keep_until_sync(dw)
            if sharded_param:
                param._tdsa_accum = None
            else:
                param.grad = None
    else:
        raise ValueError(mode)
