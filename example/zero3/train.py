"""ZeRO-3 GPT-2 training with true meta-init (parity in intent:
/root/reference/example/zero3/train.py; the reference materializes the full
model on every rank — here each rank materializes ONLY its owned shard).

Run: torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node N example/zero3/train.py
"""

import os
import sys
from collections import OrderedDict

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..", "..")))

import torch
import torch.distributed as dist

from example.common import init_distributed, synthetic_batch
from tiny_deepspeed_amd.models import GPTConfig, GPT2Model
from tiny_deepspeed_amd import Zero3, Zero3AdamW, partition_tensors

rank, world_size, device = init_distributed()
torch.manual_seed(0)
dtype = torch.bfloat16 if device.type == "cuda" else torch.float32

config = GPTConfig()
ranks_map = [f"{device.type}:{i}" if device.type == "cuda" else "cpu"
             for i in range(world_size)]

# meta-construct: no allocation anywhere; each rank materializes only the
# parameters it owns when Zero3 wraps the meta model.
with torch.device("meta"):
    model = GPT2Model(config)
    parts, _ = partition_tensors(
        OrderedDict(model.named_parameters()),
        ranks_map=ranks_map, evenness_priority=0, verbose=(rank == 0),
        state_bytes_per_param=12,  # fp32 m+v+master for bf16 params
    )

model = model.to(dtype=dtype)
model = Zero3(model, parts, device=device)
optimizer = Zero3AdamW(model.named_parameters(), lr=1e-5, weight_decay=1e-1,
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
