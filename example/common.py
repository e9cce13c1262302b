"""Shared plumbing for the example train scripts."""

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

import torch
import torch.distributed as dist


def init_distributed():
    """torchrun-style env init; RCCL on GPU, gloo on CPU-only hosts."""
    rank = int(os.getenv("LOCAL_RANK", os.getenv("RANK", "0")))
    world_size = int(os.getenv("WORLD_SIZE", "1"))
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    backend = "nccl" if torch.cuda.is_available() else "gloo"
    dist.init_process_group(backend=backend, init_method="env://",
                            world_size=world_size, rank=rank)
    if torch.cuda.is_available():
        torch.cuda.set_device(rank)
        device = torch.device("cuda", rank)
    else:
        device = torch.device("cpu")
    return rank, world_size, device


def synthetic_batch(vocab_size, batch, block, device, seed):
    g = torch.Generator().manual_seed(seed)
    x = torch.randint(0, vocab_size, (batch, block), generator=g).to(device)
    y = torch.randint(0, vocab_size, (batch, block), generator=g).to(device)
    return x, y
