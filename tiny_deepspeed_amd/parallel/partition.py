"""Greedy contiguous parameter partitioner ("cache rank map").

Capability parity with the reference's partition_tensors
(``/root/reference/tiny_deepspeed/core/zero/utils/partition.py:7-102``):
partitions an OrderedDict(name -> tensor) into len(ranks_map) contiguous
parts by numel, works on meta tensors (planning without allocation),
optional malloc materializes each tensor on its owner's device.

MI355X addition (SURVEY.md component #15): a per-rank capacity check sized
for 288 GB HBM3E per GPU — partitions whose largest part cannot fit (with
optimizer state overhead) raise early instead of OOMing mid-training.

evenness_priority in [0, 1] biases the rank boundary: 0 assigns each tensor
to the rank its *start* offset falls in (pure greedy fill), 1 assigns by
its *midpoint* (more even parts when tensors are large).
"""

import warnings
from collections import OrderedDict

import torch

MI355X_HBM_BYTES = 288 * (1 << 30)


def partition_tensors(named_tensors, ranks_map, evenness_priority=0.0,
                      malloc=False, verbose=False, capacity_bytes=None,
                      state_bytes_per_param=0):
    """Returns (part_assignment: OrderedDict name->rank_idx, tensors|None).

    ranks_map: list of device strings (e.g. ["cuda:0", ..., "cuda:7"]).
    state_bytes_per_param: extra bytes/element the owner will allocate
    (e.g. 12 for fp32 AdamW m+v+master of a bf16 param) — used only by the
    capacity check.
    """
    if not 0.0 <= evenness_priority <= 1.0:
        raise ValueError("evenness_priority must be in [0, 1]")
    n_ranks = len(ranks_map)
    names = list(named_tensors.keys())
    numels = [named_tensors[n].numel() for n in names]
    total = sum(numels)
    ideal = max(total / max(n_ranks, 1), 1)

    assignment = OrderedDict()
    loads = [0] * n_ranks
    cum = 0
    for name, numel in zip(names, numels):
        pos = cum + evenness_priority * (numel / 2.0)
        rank = min(n_ranks - 1, int(pos // ideal))
        assignment[name] = rank
        loads[rank] += numel
        cum += numel

    for r in range(n_ranks):
        if loads[r] == 0:
            warnings.warn(
                f"partition_tensors: rank {r} ({ranks_map[r]}) received no "
                f"tensors ({n_ranks} ranks for {len(names)} tensors)"
            )

    if capacity_bytes is None:
        capacity_bytes = MI355X_HBM_BYTES
    for r in range(n_ranks):
        elem = max(
            (named_tensors[n].element_size() for n in names if assignment[n] == r),
            default=0,
        )
        need = loads[r] * (elem + state_bytes_per_param)
        if need > capacity_bytes:
            raise RuntimeError(
                f"partition part {r} needs {need / (1 << 30):.1f} GiB "
                f"(> {capacity_bytes / (1 << 30):.1f} GiB capacity)"
            )

    if verbose:
        for r in range(n_ranks):
            print(f"[partition] rank {r} ({ranks_map[r]}): {loads[r]} params "
                  f"({100.0 * loads[r] / max(total, 1):.1f}%)")

    out_tensors = None
    if malloc:
        out_tensors = OrderedDict()
        for name, t in named_tensors.items():
            dev = torch.device(ranks_map[assignment[name]])
            if t.is_meta:
                out_tensors[name] = torch.empty(t.shape, dtype=t.dtype, device=dev)
            else:
                out_tensors[name] = t.to(dev)
    return assignment, out_tensors
