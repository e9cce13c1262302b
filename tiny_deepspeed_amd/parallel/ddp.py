"""DDP strategy: replicated parameters, per-parameter async grad all-reduce
overlapped with backward's dX compute.

Capability parity with ``/root/reference/tiny_deepspeed/core/zero/ddp/``
(wrapper.py:15-40, module.py:27-147, optim.py:18-33). The collective is
enqueued on the comm stream right after dW is produced and the dX GEMM runs
concurrently — true overlap, without the reference's
torch.cuda.synchronize() flaw.
"""

import torch.nn as nn

from .. import modules as base
from .. import ops
from .. import optim as base_optim
from ._grad import publish_grad, ALLREDUCE
from .wrapper import ModelWrapper


class Linear(base.Linear):
    _mode = ALLREDUCE

    def backward_callback(self, dy, x):
        if self.weight.requires_grad:
            dw = ops.linear_weight_grad(dy, x, tuner=self.tuner)
            assert dw.shape == self.weight.shape
            publish_grad(self._comm, self.weight, dw, self._mode)
        if self.bias is not None and self.bias.requires_grad:
            db = ops.linear_bias_grad(dy, tuner=self.tuner)
            publish_grad(self._comm, self.bias, db, self._mode)
        # dX overlaps with the in-flight collectives above
        dx = ops.linear_input_grad(dy, self.weight, tuner=self.tuner)
        return dx, None, None

    def publish_weight_grad(self, dw):
        """Fused lm_head+CE path: route dW into this strategy's collective
        (all-reduce / reduce-to-owner per _mode) instead of autograd."""
        publish_grad(self._comm, self.weight, dw, self._mode)
        return None


class LayerNorm(base.LayerNorm):
    _mode = ALLREDUCE

    def backward_callback(self, dy, x, mean, rstd, dh=None):
        dx, ws = ops.layernorm_dx(dy, x, self.weight, mean, rstd, dh=dh,
                                  tuner=self.tuner)
        if self.weight.requires_grad:
            dw, db = ops.layernorm_dwdb(ws, dtype=self.weight.dtype, tuner=self.tuner)
            publish_grad(self._comm, self.weight, dw, self._mode)
            publish_grad(self._comm, self.bias, db, self._mode)
        return dx, None, None


class Embedding(base.Embedding):
    _mode = ALLREDUCE

    def backward_callback(self, dy, idx):
        if self.weight.requires_grad:
            dw = ops.embedding_weight_grad(idx, dy, self.num_embeddings,
                                           padding_idx=self.padding_idx,
                                           tuner=self.tuner)
            publish_grad(self._comm, self.weight, dw, self._mode)
        return None


class DDP(ModelWrapper):
    swap_map = {
        nn.Linear: Linear,
        nn.LayerNorm: LayerNorm,
        nn.Embedding: Embedding,
    }


class _DDPOptimMixin:
    """Local step; the only distributed action is waiting for in-flight
    grad all-reduces (device-side) before the first update."""

    def _setup_comm(self, comm):
        from .comm import default_comm

        self.comm = comm if comm is not None else default_comm()

    def pre_step(self):
        self.comm.sync()


class DDPSGD(_DDPOptimMixin, base_optim.SGD):
    def __init__(self, parameters, comm=None, **kw):
        super().__init__(parameters, **kw)
        self._setup_comm(comm)


class DDPAdamW(_DDPOptimMixin, base_optim.AdamW):
    def __init__(self, parameters, comm=None, **kw):
        super().__init__(parameters, **kw)
        self._setup_comm(comm)
