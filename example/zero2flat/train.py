"""Flat-bucket ZeRO-2 GPT-2 training — the reduce_scatter/all_gather flat
layout, alternative to example/zero2's per-tensor owner
mode. No partition table: parameters are numel-sharded inside ~32 MB flat
buckets and every rank updates its shard of every bucket.

Run: torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node N example/zero2flat/train.py
"""

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..", "..")))

import torch
import torch.distributed as dist

from example.common import init_distributed, synthetic_batch
from tiny_deepspeed_amd.models import GPTConfig, GPT2Model
from tiny_deepspeed_amd import Zero2Flat, Zero2FlatAdamW

rank, world_size, device = init_distributed()
torch.manual_seed(0)
dtype = torch.bfloat16 if device.type == "cuda" else torch.float32

config = GPTConfig()
model = GPT2Model(config).to(device=device, dtype=dtype)
model = Zero2Flat(model)
optimizer = Zero2FlatAdamW(model, lr=1e-5, weight_decay=1e-1)

x, y = synthetic_batch(config.vocab_size, 1, config.block_size, device, seed=rank)

for i in range(100):
    model.require_backward_grad_sync = True
    _, loss = model(x, y)
    loss.backward()
    optimizer.step()
    loss = model.comm.all_reduce_scalar_avg(loss.detach())
    if rank == 0:
        print(f"iter {i} loss: {loss.item():.4f}")

dist.destroy_process_group()
