"""From-scratch optimizer base over an OrderedDict of named parameters.

Parity with ``/root/reference/tiny_deepspeed/core/optim/base.py:7-26``
(step() loops one_step(name, param) then clears grads), with two additions
the reference lacks: state_dict/load_state_dict for checkpointing
 and a pre_step/post_step hook pair the distributed
optimizers use to order communication around the update.
"""

from collections import OrderedDict

import torch

from ..utils.profiling import trace_range


class Optimizer:
    def __init__(self, parameters, lr):
        if lr < 0.0:
            raise ValueError(f"Invalid learning rate: {lr}")
        # accepts an iterator of (name, param) like model.named_parameters()
        self.params = OrderedDict(parameters)
        for name, p in self.params.items():
            if not isinstance(p, torch.nn.Parameter):
                raise TypeError(f"{name} is not an nn.Parameter")
        self.lr = lr
        self.t = 0  # global step count (advances once per step(), not per
        # parameter — the reference advances it per tensor, a bias-correction
        # bug we do not replicate: SURVEY.md 2.11.1 / adamw.py:59)

    # --- hooks for distributed subclasses --------------------------------
    def pre_step(self):
        pass

    def post_step(self):
        pass

    def _should_update(self, name, param):
        return param.grad is not None

    @torch.no_grad()
    def step(self):
        self.t += 1
        with trace_range("tdsa:optimizer.step"):
            self._run_step()

    @torch.no_grad()
    def _run_step(self):
        self.pre_step()
        self._apply_updates(
            [(n, p) for n, p in self.params.items()
             if self._should_update(n, p)]
        )
        self.post_step()
        for param in self.params.values():
            param.grad = None

    def _apply_updates(self, items):
        """Default: per-parameter updates. AdamW overrides this with one
        fused multi-tensor kernel launch on GPU (csrc/kernels/optim.hip)."""
        for name, param in items:
            self.one_step(name, param)

    @torch.no_grad()
    def zero_grad(self):
        for param in self.params.values():
            param.grad = None

    def one_step(self, name, param):
        raise NotImplementedError

    # --- checkpointing ----------------------------------------------------
    def _state_tensors(self):
        """Subclasses return {key: {name: tensor}} of optimizer state."""
        return {}

    def state_dict(self):
        sd = {"t": self.t, "lr": self.lr, "state": {}}
        for key, per_param in self._state_tensors().items():
            sd["state"][key] = {n: t for n, t in per_param.items() if t is not None}
        return sd

    def load_state_dict(self, sd):
        self.t = sd["t"]
        self.lr = sd["lr"]
        own = self._state_tensors()
        for key, per_param in sd["state"].items():
            if key not in own:
                continue
            for n, t in per_param.items():
                if n in own[key] and own[key][n] is not None:
                    own[key][n].copy_(t.to(own[key][n].device))
