"""ZeRO-1 strategy: optimizer-state sharding.

Capability parity with ``/root/reference/tiny_deepspeed/core/zero/zero1/``:
gradients are average-REDUCED to the owning rank during backward (half the
traffic of all-reduce), optimizer state (m/v/momentum/master) exists only
on the owner, the owner updates its partition and broadcasts refreshed
parameters to all ranks at step end.
"""

import torch.nn as nn

from .. import optim as base_optim
from ._grad import REDUCE_KEEP
from ._zero_optim import _ZeroOptimMixin
from .ddp import Linear as _DDPLinear
from .ddp import LayerNorm as _DDPLayerNorm
from .ddp import Embedding as _DDPEmbedding
from .wrapper import ModelWrapper


class Linear(_DDPLinear):
    _mode = REDUCE_KEEP


class LayerNorm(_DDPLayerNorm):
    _mode = REDUCE_KEEP


class Embedding(_DDPEmbedding):
    _mode = REDUCE_KEEP


class Zero1(ModelWrapper):
    swap_map = {
        nn.Linear: Linear,
        nn.LayerNorm: LayerNorm,
        nn.Embedding: Embedding,
    }

    def __init__(self, module, parts, comm=None):
        super().__init__(module, parts=parts, comm=comm)


class Zero1SGD(_ZeroOptimMixin, base_optim.SGD):
    def __init__(self, parameters, param_part_table=None, ranks_map=None,
                 comm=None, **kw):
        self._setup_zero(param_part_table, ranks_map, comm)
        super().__init__(parameters, **kw)


class Zero1AdamW(_ZeroOptimMixin, base_optim.AdamW):
    def __init__(self, parameters, param_part_table=None, ranks_map=None,
                 comm=None, **kw):
        self._setup_zero(param_part_table, ranks_map, comm)
        super().__init__(parameters, **kw)
