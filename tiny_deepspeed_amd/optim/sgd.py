"""SGD (momentum/dampening/nesterov/maximize/weight-decay) with fused updates.

Math parity with ``/root/reference/tiny_deepspeed/core/optim/sgd.py:10-46``
(torch.optim.SGD semantics). fp32 momentum buffers; bf16 params get an fp32
master copy.
"""

import torch

from .base import Optimizer
from .. import ops


class SGD(Optimizer):
    def __init__(self, parameters, lr=1e-3, momentum=0.0, dampening=0.0,
                 weight_decay=0.0, nesterov=False, maximize=False):
        super().__init__(parameters, lr)
        if momentum < 0.0:
            raise ValueError(f"Invalid momentum: {momentum}")
        if nesterov and (momentum <= 0.0 or dampening != 0.0):
            raise ValueError("Nesterov momentum requires momentum > 0 and zero dampening")
        self.momentum = momentum
        self.dampening = dampening
        self.weight_decay = weight_decay
        self.nesterov = nesterov
        self.maximize = maximize
        self.velocities = {}
        self.master = {}
        self._stepped = set()
        for name, p in self.params.items():
            if self._owns_state(name, p):
                self._init_state(name, p)

    def _owns_state(self, name, param):
        return param.numel() > 0

    def _init_state(self, name, p):
        if self.momentum != 0.0:
            self.velocities[name] = torch.zeros(p.shape, dtype=torch.float32,
                                                device=p.device)
        if p.dtype != torch.float32:
            self.master[name] = p.detach().float().clone()

    @torch.no_grad()
    def _apply_updates(self, items):
        if not items or not items[0][1].is_cuda or not ops.ext_available():
            return super()._apply_updates(items)
        params, grads, bufs, masters = [], [], [], []
        first = all(n not in self._stepped for n, _ in items)
        mixed = any((n in self._stepped) != (not first) for n, _ in items)
        if mixed:  # per-tensor first-step flags differ: fall back
            return super()._apply_updates(items)
        for name, p in items:
            params.append(p.data)
            grads.append(p.grad)
            bufs.append(self.velocities.get(name))
            masters.append(self.master.get(name))
            self._stepped.add(name)
        ops.get_ext().sgd_step_multi(
            params, grads, bufs, masters, self.lr, self.momentum,
            self.dampening, self.weight_decay, self.nesterov, self.maximize,
            first,
        )

    @torch.no_grad()
    def one_step(self, name, param):
        first = name not in self._stepped
        self._stepped.add(name)
        ops.sgd_step(
            param.data, param.grad, self.velocities.get(name),
            self.master.get(name), self.lr, self.momentum, self.dampening,
            self.weight_decay, self.nesterov, self.maximize, first,
        )

    def _state_tensors(self):
        return {"velocities": self.velocities, "master": self.master}
