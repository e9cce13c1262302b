"""ZeRO-2 GPT-2 training (parity: /root/reference/example/zero2/train.py).

Partition is planned on the meta device (no allocation), then the model is
materialized and wrapped.

Run: torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node N example/zero2/train.py
"""

import os
import sys
from collections import OrderedDict

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..", "..")))

import torch
import torch.distributed as dist

from example.common import init_distributed, synthetic_batch
from tiny_deepspeed_amd.models import GPTConfig, GPT2Model
from tiny_deepspeed_amd import Zero2, Zero2AdamW, partition_tensors

rank, world_size, device = init_distributed()
torch.manual_seed(0)
dtype = torch.bfloat16 if device.type == "cuda" else torch.float32

config = GPTConfig()
ranks_map = [f"{device.type}:{i}" if device.type == "cuda" else "cpu"
             for i in range(world_size)]
with torch.device("meta"):
    parts, _ = partition_tensors(
        OrderedDict(GPT2Model(config).named_parameters()),
        ranks_map=ranks_map, evenness_priority=0, verbose=(rank == 0),
    )

model = GPT2Model(config).to(device=device, dtype=dtype)
model = Zero2(model, parts)
optimizer = Zero2AdamW(model.named_parameters(), lr=1e-5, weight_decay=1e-1,
                       param_part_table=parts, ranks_map=ranks_map)

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
