"""Checkpoint / resume (absent in the reference — SURVEY.md 5.4).

Saves model + optimizer + step counter. For ZeRO strategies each rank saves
its own shard file (rank-suffixed) — optimizer state exists only on owners
and ZeRO-3 params only on owners — and load restores the same layout.
"""

import os

import torch


def save_checkpoint(path, model, optimizer=None, step=0, rank=0):
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    payload = {
        "step": step,
        "model": {k: v for k, v in model.state_dict().items()},
    }
    if optimizer is not None:
        payload["optimizer"] = optimizer.state_dict()
    torch.save(payload, _rank_path(path, rank))


def load_checkpoint(path, model, optimizer=None, rank=0, map_location="cpu"):
    payload = torch.load(_rank_path(path, rank), map_location=map_location,
                         weights_only=False)
    sd = payload["model"]
    own = dict(model.state_dict())
    # tolerate ZeRO-3 sharding: only copy tensors whose shapes match
    filtered = {k: v for k, v in sd.items()
                if k in own and own[k].shape == v.shape}
    model.load_state_dict(filtered, strict=False)
    if optimizer is not None and "optimizer" in payload:
        optimizer.load_state_dict(payload["optimizer"])
    return payload.get("step", 0)


def _rank_path(path, rank):
    if rank == 0:
        return path
    base, ext = os.path.splitext(path)
    return f"{base}.rank{rank}{ext}"
