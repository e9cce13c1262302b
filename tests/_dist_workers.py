"""Worker functions executed inside spawned gloo ranks (see dist_utils)."""

from collections import OrderedDict

import torch

CFG = dict(block_size=32, vocab_size=96, n_layer=2, n_head=2, n_embd=32)
ITERS = 4


def make_cfg():
    from tiny_deepspeed_amd.models import GPTConfig

    return GPTConfig(**CFG)


def batch(seed=0):
    g = torch.Generator().manual_seed(seed)
    x = torch.randint(0, CFG["vocab_size"], (2, CFG["block_size"]), generator=g)
    y = torch.randint(0, CFG["vocab_size"], (2, CFG["block_size"]), generator=g)
    return x, y


def single_device_losses():
    import tiny_deepspeed_amd as tdsa
    from tiny_deepspeed_amd.models import GPT2Model

    torch.manual_seed(0)
    model = tdsa.Single(GPT2Model(make_cfg()))
    opt = tdsa.AdamW(model.named_parameters(), lr=1e-3, weight_decay=0.01)
    x, y = batch()
    losses = []
    for _ in range(ITERS):
        _, loss = model(x, y)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    return losses


def train_strategy(rank, world, strategy):
    import tiny_deepspeed_amd as tdsa
    from tiny_deepspeed_amd.models import GPT2Model

    torch.manual_seed(0)
    model = GPT2Model(make_cfg())
    ranks_map = ["cpu"] * world
    x, y = batch()  # same data on all ranks -> parity with single device
    if strategy == "ddp":
        model = tdsa.DDP(model)
        opt = tdsa.DDPAdamW(model.named_parameters(), lr=1e-3, weight_decay=0.01)
    else:
        with torch.device("meta"):
            meta = GPT2Model(make_cfg())
        parts, _ = tdsa.partition_tensors(
            OrderedDict(meta.named_parameters()), ranks_map
        )
        wrap = {"zero1": tdsa.Zero1, "zero2": tdsa.Zero2, "zero3": tdsa.Zero3}[strategy]
        optc = {"zero1": tdsa.Zero1AdamW, "zero2": tdsa.Zero2AdamW,
                "zero3": tdsa.Zero3AdamW}[strategy]
        model = wrap(model, parts)
        opt = optc(model.named_parameters(), lr=1e-3, weight_decay=0.01,
                   param_part_table=parts, ranks_map=ranks_map)
    losses = []
    for _ in range(ITERS):
        model.require_backward_grad_sync = True
        _, loss = model(x, y)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    return losses


def ddp_grads_averaged(rank, world):
    import tiny_deepspeed_amd as tdsa

    torch.manual_seed(0)
    lin = torch.nn.Linear(8, 8, bias=False)
    model = tdsa.DDP(torch.nn.Sequential(lin))
    torch.manual_seed(100 + rank)
    x = torch.randn(4, 8)
    model.require_backward_grad_sync = True
    model(x).square().mean().backward()
    model.comm.sync()
    return lin.weight.grad.clone()


def zero2_shard_check(rank, world):
    import tiny_deepspeed_amd as tdsa
    from tiny_deepspeed_amd.models import GPT2Model

    torch.manual_seed(0)
    model = GPT2Model(make_cfg())
    with torch.device("meta"):
        meta = GPT2Model(make_cfg())
    parts, _ = tdsa.partition_tensors(
        OrderedDict(meta.named_parameters()), ["cpu"] * world
    )
    model = tdsa.Zero2(model, parts)
    x, y = batch()
    model.require_backward_grad_sync = True
    _, loss = model(x, y)
    loss.backward()
    model.comm.sync()
    owned_with_grad, unowned_with_grad = 0, 0
    for n, p in model.named_parameters():
        if parts[n] == rank:
            owned_with_grad += int(p.grad is not None)
        else:
            unowned_with_grad += int(p.grad is not None)
    return owned_with_grad, unowned_with_grad


def zero3_shard_check(rank, world):
    import tiny_deepspeed_amd as tdsa
    from tiny_deepspeed_amd.models import GPT2Model

    torch.manual_seed(0)
    model = GPT2Model(make_cfg())
    with torch.device("meta"):
        meta = GPT2Model(make_cfg())
    parts, _ = tdsa.partition_tensors(
        OrderedDict(meta.named_parameters()), ["cpu"] * world
    )
    model = tdsa.Zero3(model, parts)
    owned_elems = sum(
        p.numel() for n, p in model.named_parameters() if parts[n] == rank
    )
    unowned_elems = sum(
        p.numel() for n, p in model.named_parameters() if parts[n] != rank
    )
    assert unowned_elems == 0  # params actually sharded
    x, y = batch()
    model.require_backward_grad_sync = True
    _, loss = model(x, y)
    loss.backward()
    model.comm.sync()
    return owned_elems, float(loss.item())


def zero3_meta_init(rank, world):
    import tiny_deepspeed_amd as tdsa
    from tiny_deepspeed_amd.models import GPT2Model

    with torch.device("meta"):
        model = GPT2Model(make_cfg())
        parts, _ = tdsa.partition_tensors(
            OrderedDict(model.named_parameters()), ["cpu"] * world
        )
    model = tdsa.Zero3(model, parts, device="cpu")
    opt = tdsa.Zero3AdamW(model.named_parameters(), lr=1e-3,
                          param_part_table=parts, ranks_map=["cpu"] * world)
    for n, p in model.named_parameters():
        assert not p.is_meta
        if parts[n] == rank:
            assert p.numel() > 0
        else:
            assert p.numel() == 0
    x, y = batch()
    losses = []
    for _ in range(3):
        model.require_backward_grad_sync = True
        _, loss = model(x, y)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    # meta materialization must honor the model's init scheme (std 0.02):
    # the nn-default inits started near loss ~10 instead of ~ln(vocab)
    assert losses[0] < 6.0, losses
    return losses


def grad_accumulation(rank, world):
    import tiny_deepspeed_amd as tdsa

    torch.manual_seed(0)
    lin = torch.nn.Linear(8, 8, bias=False)
    model = tdsa.DDP(torch.nn.Sequential(lin))
    torch.manual_seed(200 + rank)
    xs = [torch.randn(4, 8) for _ in range(3)]
    for i, x in enumerate(xs):
        model.require_backward_grad_sync = i == len(xs) - 1
        model(x).square().mean().backward()
    model.comm.sync()
    return lin.weight.grad.clone()


def fused_lmhead_strategy_losses(rank, world, strategy):
    """Strategy training with fused_lm_head=True: dW publication must go
    through publish_weight_grad into the strategy's collective."""
    import tiny_deepspeed_amd as tdsa
    from tiny_deepspeed_amd.models import GPTConfig, GPT2Model

    cfg = GPTConfig(fused_lm_head=True, **CFG)
    torch.manual_seed(0)
    model = GPT2Model(cfg)
    x, y = batch()
    if strategy == "ddp":
        model = tdsa.DDP(model)
        opt = tdsa.DDPAdamW(model.named_parameters(), lr=1e-3,
                            weight_decay=0.01)
    elif strategy == "zero2flat":
        model = tdsa.Zero2Flat(model)
        opt = tdsa.Zero2FlatAdamW(model, lr=1e-3, weight_decay=0.01)
    else:
        with torch.device("meta"):
            meta = GPT2Model(cfg)
        parts, _ = tdsa.partition_tensors(
            OrderedDict(meta.named_parameters()), ["cpu"] * world)
        wrap = {"zero2": tdsa.Zero2, "zero3": tdsa.Zero3}[strategy]
        optc = {"zero2": tdsa.Zero2AdamW, "zero3": tdsa.Zero3AdamW}[strategy]
        model = wrap(model, parts)
        opt = optc(model.named_parameters(), lr=1e-3, weight_decay=0.01,
                   param_part_table=parts, ranks_map=["cpu"] * world)
    losses = []
    for _ in range(ITERS):
        model.require_backward_grad_sync = True
        logits, loss = model(x, y)
        assert logits is None  # fused path ran
        loss.backward()
        opt.step()
        losses.append(loss.item())
    return losses


def fused_lmhead_single_losses():
    import tiny_deepspeed_amd as tdsa
    from tiny_deepspeed_amd.models import GPTConfig, GPT2Model

    cfg = GPTConfig(fused_lm_head=True, **CFG)
    torch.manual_seed(0)
    model = tdsa.Single(GPT2Model(cfg))
    opt = tdsa.AdamW(model.named_parameters(), lr=1e-3, weight_decay=0.01)
    x, y = batch()
    losses = []
    for _ in range(ITERS):
        _, loss = model(x, y)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    return losses


def zero2flat_losses(rank, world, bucket_bytes=32 << 20):
    """Flat-bucket ZeRO-2: returns (losses, collective counts, n_buckets,
    n_params) so the test can assert loss parity AND the O(#params) ->
    O(#buckets) collective reduction."""
    import tiny_deepspeed_amd as tdsa
    from tiny_deepspeed_amd.models import GPT2Model

    torch.manual_seed(0)
    model = tdsa.Zero2Flat(GPT2Model(make_cfg()), bucket_bytes=bucket_bytes)
    opt = tdsa.Zero2FlatAdamW(model, lr=1e-3, weight_decay=0.01)
    comm = model.comm
    counts = {"rs": 0, "ag": 0}
    orig_rs = comm.reduce_scatter_avg
    orig_ag = comm.all_gather_flat

    def rs(out, inp):
        counts["rs"] += 1
        return orig_rs(out, inp)

    def ag(out, inp):
        counts["ag"] += 1
        return orig_ag(out, inp)

    comm.reduce_scatter_avg = rs
    comm.all_gather_flat = ag
    x, y = batch()
    losses = []
    for _ in range(ITERS):
        model.require_backward_grad_sync = True
        _, loss = model(x, y)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    n_params = len(list(model.named_parameters()))
    return losses, counts, len(model.engine.buckets), n_params


def zero2flat_grad_accum(rank, world):
    """Two no-sync microbatches then an armed one: the flat slots must
    accumulate; compare against a single armed pass over summed grads via
    the resulting loss trajectory (parity with per-tensor zero2)."""
    import tiny_deepspeed_amd as tdsa
    from tiny_deepspeed_amd.models import GPT2Model

    def run(flat):
        torch.manual_seed(0)
        m = GPT2Model(make_cfg())
        if flat:
            model = tdsa.Zero2Flat(m)
            opt = tdsa.Zero2FlatAdamW(model, lr=1e-3, weight_decay=0.01)
        else:
            parts, _ = tdsa.partition_tensors(
                OrderedDict((n, p) for n, p in m.named_parameters()),
                ["cpu"] * world)
            model = tdsa.Zero2(m, parts)
            opt = tdsa.Zero2AdamW(model.named_parameters(), lr=1e-3,
                                  weight_decay=0.01,
                                  param_part_table=parts,
                                  ranks_map=["cpu"] * world)
        losses = []
        for it in range(2):
            for micro in range(3):
                g = torch.Generator().manual_seed(10 * it + micro)
                x = torch.randint(0, CFG["vocab_size"],
                                  (2, CFG["block_size"]), generator=g)
                y = torch.randint(0, CFG["vocab_size"],
                                  (2, CFG["block_size"]), generator=g)
                model.require_backward_grad_sync = micro == 2
                _, loss = model(x, y)
                loss.backward()
                losses.append(loss.item())
            opt.step()
        return losses

    return run(True), run(False)


def _build_zero(rank, world, strategy):
    import tiny_deepspeed_amd as tdsa
    from tiny_deepspeed_amd.models import GPT2Model

    torch.manual_seed(0)
    model = GPT2Model(make_cfg())
    with torch.device("meta"):
        meta = GPT2Model(make_cfg())
    parts, _ = tdsa.partition_tensors(
        OrderedDict(meta.named_parameters()), ["cpu"] * world
    )
    wrap = {"zero1": tdsa.Zero1, "zero2": tdsa.Zero2}[strategy]
    optc = {"zero1": tdsa.Zero1AdamW, "zero2": tdsa.Zero2AdamW}[strategy]
    model = wrap(model, parts)
    opt = optc(model.named_parameters(), lr=1e-3, weight_decay=0.01,
               param_part_table=parts, ranks_map=["cpu"] * world)
    return model, opt


def _run_steps(model, opt, seeds):
    losses = []
    for s in seeds:
        g = torch.Generator().manual_seed(s)
        x = torch.randint(0, CFG["vocab_size"], (2, CFG["block_size"]),
                          generator=g)
        y = torch.randint(0, CFG["vocab_size"], (2, CFG["block_size"]),
                          generator=g)
        model.require_backward_grad_sync = True
        _, loss = model(x, y)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    return losses


def zero_ckpt_train_save(rank, world, strategy, path):
    """Train 2 steps, save the per-rank shard, continue 2 more steps;
    return the continuation losses (the resharded-resume target)."""
    from tiny_deepspeed_amd.utils.checkpoint import save_checkpoint

    model, opt = _build_zero(rank, world, strategy)
    _run_steps(model, opt, [0, 1])
    save_checkpoint(path, model, opt, step=2, rank=rank, world_size=world)
    import torch.distributed as dist
    dist.barrier()
    return _run_steps(model, opt, [10, 11])


def zero_ckpt_resume(rank, world, strategy, path):
    """Fresh build under a DIFFERENT world size, merge-load every shard,
    run the same continuation steps."""
    from tiny_deepspeed_amd.utils.checkpoint import load_checkpoint

    model, opt = _build_zero(rank, world, strategy)
    step, report = load_checkpoint(path, model, opt, rank=rank,
                                   return_report=True)
    assert step == 2
    assert not report.mismatched and not report.unexpected, repr(report)
    assert opt.t == 2
    return _run_steps(model, opt, [10, 11])


def broadcast_bucketed_roundtrip(rank, world):
    """Mixed owners/sizes through comm.broadcast_bucketed: every rank ends
    with the owner's values (small tensors ride flat buckets)."""
    from tiny_deepspeed_amd.parallel.comm import CommContext

    comm = CommContext()
    torch.manual_seed(100 + rank)  # ranks start DIFFERENT
    sizes = [(3,), (100,), (5, 5), (1 << 20,), (7,), (2, 2)]
    owners = [0, 1, 1, 0, 1, 0]
    tensors = [torch.randn(s) for s in sizes]
    expected = []
    for t, owner in zip(tensors, owners):
        g = torch.Generator().manual_seed(100 + owner)
        # regenerate what the owner drew for THIS tensor: replay its stream
        expected.append(None)  # filled below
    # replay each owner's full stream to know its values
    for owner in (0, 1):
        g = torch.random.manual_seed(100 + owner)
        vals = [torch.randn(s) for s in sizes]
        for i, o in enumerate(owners):
            if o == owner:
                expected[i] = vals[i]
    comm.broadcast_bucketed([(t, o) for t, o in zip(tensors, owners)],
                            bucket_bytes=1024)
    comm.sync()
    return all(torch.equal(t, e) for t, e in zip(tensors, expected))
