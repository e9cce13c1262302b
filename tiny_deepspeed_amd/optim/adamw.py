"""AdamW with fused single-kernel updates on GPU.

Math parity with ``/root/reference/tiny_deepspeed/core/optim/adamw.py:10-59``
(decoupled weight decay, amsgrad option) with the per-parameter step-count
bug fixed (global t). fp32 moments; non-fp32 params get an
fp32 master copy so bf16 training doesn't lose update precision.
"""

import torch

from .base import Optimizer
from .. import ops


class AdamW(Optimizer):
    def __init__(self, parameters, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=1e-2, amsgrad=False):
        super().__init__(parameters, lr)
        if not 0.0 <= betas[0] < 1.0 or not 0.0 <= betas[1] < 1.0:
            raise ValueError(f"Invalid betas: {betas}")
        if eps <= 0.0:
            raise ValueError(f"Invalid eps: {eps}")
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.amsgrad = amsgrad
        self.exp_avg = {}
        self.exp_avg_sq = {}
        self.max_exp_avg_sq = {}
        self.master = {}
        for name, p in self.params.items():
            if self._owns_state(name, p):
                self._init_state(name, p)

    def _owns_state(self, name, param):
        """Which params this rank keeps optimizer state for (ZeRO-1/2/3
        subclasses restrict this to owned partitions)."""
        return param.numel() > 0

    def _init_state(self, name, p):
        dev = p.device
        self.exp_avg[name] = torch.zeros(p.shape, dtype=torch.float32, device=dev)
        self.exp_avg_sq[name] = torch.zeros(p.shape, dtype=torch.float32, device=dev)
        if self.amsgrad:
            self.max_exp_avg_sq[name] = torch.zeros(p.shape, dtype=torch.float32, device=dev)
        if p.dtype != torch.float32:
            self.master[name] = p.detach().float().clone()

    @torch.no_grad()
    def _apply_updates(self, items):
        if (not items or self.amsgrad or not items[0][1].is_cuda
                or not ops.ext_available()):
            return super()._apply_updates(items)
        params, grads, ms, vs, masters = [], [], [], [], []
        for name, p in items:
            params.append(p.data)
            grads.append(p.grad)
            ms.append(self.exp_avg[name])
            vs.append(self.exp_avg_sq[name])
            masters.append(self.master.get(name))
        ops.get_ext().adamw_step_multi(
            params, grads, ms, vs, masters, self.lr, self.beta1, self.beta2,
            self.eps, self.weight_decay, self.t,
        )

    @torch.no_grad()
    def one_step(self, name, param):
        ops.adamw_step(
            param.data, param.grad, self.exp_avg[name], self.exp_avg_sq[name],
            self.master.get(name), self.t, self.lr, self.beta1, self.beta2,
            self.eps, self.weight_decay,
            max_exp_avg_sq=self.max_exp_avg_sq.get(name),
        )

    def _state_tensors(self):
        return {
            "exp_avg": self.exp_avg,
            "exp_avg_sq": self.exp_avg_sq,
            "max_exp_avg_sq": self.max_exp_avg_sq,
            "master": self.master,
        }
