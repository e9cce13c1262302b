"""Communication context: RCCL collectives on dedicated HIP streams.

The MI355X replacement for the reference's inline torch.distributed calls
(call-site inventory: SURVEY.md 2.9). Deliberate design changes:

1. The reference issues async collectives then immediately calls
   ``torch.cuda.synchronize()`` (``/root/reference/tiny_deepspeed/core/zero/
   ddp/module.py:17-24``) — a host sync that defeats the overlap it
   advertises. Here every collective is enqueued on a
   dedicated comm stream ordered against compute with hipEvents
   (stream waits), so backward's dX GEMMs run while grads fly over xGMI.
   The only rendezvous is ``sync()`` — a device-side stream wait.

2. Two channels, two communicators: gradient reduces ride the "reduce"
   channel; ZeRO-3's just-in-time parameter gathers ride a separate
   "gather" channel with its own stream AND its own process group —
   RCCL requires identical collective order per communicator, and with
   one communicator a layer's gather would have to drain every pending
   grad reduce first, serializing backward.

3. Gradients are AVERAGED (pre-scaled by 1/world then SUM-reduced, which
   works on both RCCL and gloo), fixing the reference's summed-grads /
   effective-LR-scales-with-world quirk.

Memory lifetime: tensors touched on a comm stream are marked with
``record_stream`` so the stream-ordered caching allocator defers reuse
until the collective has passed — this is what makes ZeRO-2's non-owner
gradient release REAL (the reference faked it and asked for a C++ plugin,
``zero2/module.py:31``).

On CPU (gloo backend, used by the no-GPU tests) there are no streams;
collectives run async_op=True and sync()/wait_gather() wait the handles.
"""

import os

import torch
import torch.distributed as dist

REDUCE = "reduce"
GATHER = "gather"


class CommContext:
    def __init__(self, process_group=None, use_comm_stream=True):
        self.initialized = dist.is_available() and dist.is_initialized()
        # TDSA_COMM_FORCE=1: run every collective even at world size 1.
        # Lets 1-GPU leases exercise the real RCCL enqueue / comm-stream
        # ordering / record_stream lifetime path that the world-1
        # early-returns below would otherwise skip (hardware evidence for
        # the multi-GPU machinery before an 8-GPU node exists).
        self.force = os.environ.get("TDSA_COMM_FORCE", "0") == "1"
        # TDSA_COMM_SINGLE=1: one communicator + one comm stream for BOTH
        # channels. Safety fallback for ZeRO-3 at scale: concurrent
        # communicators can deadlock if their kernels interleave differently
        # across ranks; with a single stream the enqueue order is the
        # (identical) program order on every rank. Costs gather/reduce
        # serialization — use only if the dual-communicator mode misbehaves.
        self.single = os.environ.get("TDSA_COMM_SINGLE", "0") == "1"
        self.pg = {REDUCE: process_group, GATHER: process_group}
        if self.initialized:
            self.rank = dist.get_rank(process_group)
            self.world_size = dist.get_world_size(process_group)
            if ((self.world_size > 1 or self.force) and process_group is None
                    and not self.single):
                # second communicator for the gather channel (must be
                # constructed collectively, identical on all ranks)
                self.pg[GATHER] = dist.new_group(backend=dist.get_backend())
        else:
            self.rank = 0
            self.world_size = 1
            self.force = False
        self.is_cuda = torch.cuda.is_available()
        # streams are keyed on the BACKEND, not on GPU visibility: a gloo
        # process group on a GPU box moves CPU tensors, and record_stream
        # on those raises (found running the combined CPU+GPU suite on an
        # MI355X box). Uninitialized contexts keep streams harmless — every
        # collective early-returns at _inactive().
        backend = (str(dist.get_backend()) if self.initialized else None)
        use_streams = (self.is_cuda and use_comm_stream
                       and (backend is None or "nccl" in backend))
        reduce_stream = torch.cuda.Stream() if use_streams else None
        self.streams = {
            REDUCE: reduce_stream,
            GATHER: (reduce_stream if self.single else torch.cuda.Stream())
                    if use_streams else None,
        }
        self._works = {REDUCE: [], GATHER: []}
        self._keepalive = []
        # gloo flat broadcasts whose scatter-back is deferred to sync():
        # (work, flat, bucket, numels) per channel
        self._pending_flat = {REDUCE: [], GATHER: []}
        # observability: per-collective launch counts and payload bytes
        # (read by bench.py's comm-stats line; negligible overhead)
        self.stats = {}

    def _count(self, kind, t):
        c = self.stats.setdefault(kind, [0, 0])
        c[0] += 1
        c[1] += t.numel() * t.element_size()

    # ------------------------------------------------------------------ #
    def _inactive(self):
        """True when collectives should no-op (world 1, unless forced)."""
        if not self.initialized:
            return True
        return self.world_size == 1 and not self.force

    def _launch(self, tensors, collective, channel):
        stream = self.streams[channel]
        if stream is not None:
            stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(stream):
                collective()
                for t in tensors:
                    t.record_stream(stream)
        else:
            work = collective(async_op=True)
            if work is not None:
                self._works[channel].append(work)

    def sync(self):
        """Order subsequent compute after ALL outstanding communication.
        Device-side wait on GPU (no host sync); work.wait() on gloo."""
        for stream in self.streams.values():
            if stream is not None:
                torch.cuda.current_stream().wait_stream(stream)
        for works in self._works.values():
            for w in works:
                w.wait()
            works.clear()
        for channel in (REDUCE, GATHER):
            self._drain_flat(channel)
        self._keepalive.clear()

    def wait_gather(self):
        """Order subsequent compute after outstanding gathers only."""
        stream = self.streams[GATHER]
        if stream is not None:
            torch.cuda.current_stream().wait_stream(stream)
        for w in self._works[GATHER]:
            w.wait()
        self._works[GATHER].clear()
        self._drain_flat(GATHER)

    def _drain_flat(self, channel):
        """Complete deferred gloo flat broadcasts: wait + scatter-back."""
        for work, flat, bucket, numels, src in self._pending_flat[channel]:
            work.wait()
            if self.rank != src:
                off = 0
                for t, n in zip(bucket, numels):
                    t.view(-1).copy_(flat[off:off + n])
                    off += n
        self._pending_flat[channel].clear()

    def keep_until_sync(self, t):
        """Pin a tensor's host reference until the next sync() (gloo path;
        on GPU record_stream already guarantees device lifetime)."""
        if self.streams[REDUCE] is None:
            self._keepalive.append(t)

    # --- collectives ---------------------------------------------------- #
    def all_reduce_avg(self, t):
        """Async in-place average-all-reduce; returns t."""
        if self._inactive():
            return t
        self._count("all_reduce", t)

        def run(async_op=False):
            t.div_(self.world_size)
            return dist.all_reduce(t, op=dist.ReduceOp.SUM,
                                   group=self.pg[REDUCE], async_op=async_op)

        self._launch([t], run, REDUCE)
        return t

    def reduce_avg_to(self, t, owner):
        """Async in-place average-reduce to `owner`; valid only there."""
        if self._inactive():
            return t
        self._count("reduce", t)

        def run(async_op=False):
            t.div_(self.world_size)
            return dist.reduce(t, dst=owner, op=dist.ReduceOp.SUM,
                               group=self.pg[REDUCE], async_op=async_op)

        self._launch([t], run, REDUCE)
        return t

    def broadcast(self, t, src, channel=REDUCE):
        """Async in-place broadcast from `src`; returns t."""
        if self._inactive():
            return t
        self._count("broadcast", t)

        def run(async_op=False):
            return dist.broadcast(t, src=src, group=self.pg[channel],
                                  async_op=async_op)

        self._launch([t], run, channel)
        return t

    def gather_broadcast(self, t, src):
        return self.broadcast(t, src, channel=GATHER)

    def broadcast_bucketed(self, tensors_with_src, bucket_bytes=1 << 22,
                           channel=REDUCE):
        """Broadcast many tensors, coalescing consecutive SMALL same-owner
        tensors into flat buckets (one collective each). Addresses the
        reference's 'communication bucketing' TODO (README.md:71): the GPT-2
        census has ~100 sub-1MB payloads whose per-collective launch latency
        dominates their transfer time. Large tensors broadcast directly
        (a flatten round trip would double their local traffic)."""
        if self._inactive():
            return
        groups = []  # (src, [tensors]) of consecutive small same-src tensors
        for t, src in tensors_with_src:
            nbytes = t.numel() * t.element_size()
            if nbytes >= bucket_bytes:
                groups.append((src, None, t))
                continue
            if (groups and groups[-1][1] is not None
                    and groups[-1][0] == src
                    and groups[-1][2] + nbytes <= bucket_bytes
                    and groups[-1][1][0].dtype == t.dtype):
                groups[-1][1].append(t)
                groups[-1] = (src, groups[-1][1], groups[-1][2] + nbytes)
            else:
                groups.append((src, [t], nbytes))
        for src, bucket, t_or_bytes in groups:
            if bucket is None:
                self.broadcast(t_or_bytes, src, channel=channel)
            elif len(bucket) == 1:
                self.broadcast(bucket[0], src, channel=channel)
            else:
                self._broadcast_flat(bucket, src, channel)

    def _broadcast_flat(self, bucket, src, channel):
        stream = self.streams[channel]
        numels = [t.numel() for t in bucket]
        if self.rank == src:
            flat = torch.cat([t.reshape(-1) for t in bucket])
        else:
            flat = torch.empty(sum(numels), dtype=bucket[0].dtype,
                               device=bucket[0].device)
        if stream is not None:
            stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(stream):
                dist.broadcast(flat, src=src, group=self.pg[channel])
                if self.rank != src:
                    off = 0
                    for t, n in zip(bucket, numels):
                        t.view(-1).copy_(flat[off:off + n])
                        off += n
                flat.record_stream(stream)
                for t in bucket:
                    t.record_stream(stream)
        else:
            # gloo: stay async until sync()/wait_gather() — the scatter-back
            # to non-src ranks is deferred with the completion handle
            work = dist.broadcast(flat, src=src, group=self.pg[channel],
                                  async_op=True)
            self._pending_flat[channel].append((work, flat, bucket, numels, src))

    def reduce_scatter_avg(self, out_shard, in_flat):
        """Average-reduce `in_flat` (numel == world * shard) across ranks and
        scatter equal shards: this rank's shard lands in `out_shard`.
        Async on the REDUCE stream. The flat-bucket ZeRO-2 grad collective —
        one call replaces a bucketful of per-tensor reduces."""
        if self._inactive():
            out_shard.copy_(in_flat[: out_shard.numel()])
            return out_shard
        self._count("reduce_scatter", in_flat)

        def run(async_op=False):
            in_flat.div_(self.world_size)
            if dist.get_backend(self.pg[REDUCE]) == "gloo":
                # gloo lacks reduce_scatter_tensor: all-reduce then slice
                # (CPU test path only; synchronous is fine there)
                work = dist.all_reduce(in_flat, op=dist.ReduceOp.SUM,
                                       group=self.pg[REDUCE],
                                       async_op=async_op)
                if work is not None:
                    work.wait()
                n = out_shard.numel()
                out_shard.copy_(in_flat[self.rank * n:(self.rank + 1) * n])
                return None
            return dist.reduce_scatter_tensor(
                out_shard, in_flat, op=dist.ReduceOp.SUM,
                group=self.pg[REDUCE], async_op=async_op)

        self._launch([out_shard, in_flat], run, REDUCE)
        return out_shard

    def all_gather_flat(self, out_flat, in_shard):
        """All-gather equal shards into `out_flat` (in-place friendly:
        in_shard may alias out_flat's own-rank slice). Async, REDUCE stream."""
        if self._inactive():
            n = in_shard.numel()
            if out_flat[:n].data_ptr() != in_shard.data_ptr():
                out_flat[:n].copy_(in_shard)
            return out_flat
        self._count("all_gather", out_flat)

        def run(async_op=False):
            if dist.get_backend(self.pg[REDUCE]) == "gloo":
                chunks = list(out_flat.chunk(self.world_size))
                work = dist.all_gather(chunks, in_shard.clone(),
                                       group=self.pg[REDUCE],
                                       async_op=async_op)
                if work is not None:
                    work.wait()
                return None
            return dist.all_gather_into_tensor(out_flat, in_shard,
                                               group=self.pg[REDUCE],
                                               async_op=async_op)

        self._launch([out_flat, in_shard], run, REDUCE)
        return out_flat

    def all_reduce_scalar_avg(self, t):
        """Synchronous scalar average (loss logging)."""
        if self._inactive():
            return t
        dist.all_reduce(t, op=dist.ReduceOp.SUM, group=self.pg[REDUCE])
        t.div_(self.world_size)
        return t


_DEFAULT = None


def default_comm(refresh=False):
    """Process-wide CommContext (create after dist.init_process_group)."""
    global _DEFAULT
    if _DEFAULT is None or refresh:
        _DEFAULT = CommContext()
    return _DEFAULT
