"""Single-device GPT-2 training (parity: /root/reference/example/single_device/train.py).

Unlike the reference (which runs the raw torch model), the Single wrapper
routes Linear/LayerNorm/Embedding through the CDNA4 op stack so even the
no-distribution baseline exercises the HIP kernels on GPU.
"""

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..", "..")))

import torch

from example.common import synthetic_batch
from tiny_deepspeed_amd.models import GPTConfig, GPT2Model
from tiny_deepspeed_amd import Single, AdamW

torch.manual_seed(0)
device = torch.device("cuda", 0) if torch.cuda.is_available() else torch.device("cpu")
dtype = torch.bfloat16 if device.type == "cuda" else torch.float32

config = GPTConfig()
model = GPT2Model(config).to(device=device, dtype=dtype)
model = Single(model)
optimizer = AdamW(model.named_parameters(), lr=1e-5, weight_decay=1e-1)

x, y = synthetic_batch(config.vocab_size, 1, config.block_size, device, seed=0)

for i in range(100):
    _, loss = model(x, y)
    loss.backward()
    optimizer.step()
    print(f"iter {i} loss: {loss.item():.4f}")
