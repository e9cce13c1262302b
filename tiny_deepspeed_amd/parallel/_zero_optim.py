"""Shared machinery for the ZeRO sharded optimizers.

Optimizer state lives ONLY on the owning rank (ZeRO-1's point:
``/root/reference/tiny_deepspeed/core/zero/zero1/optim.py:75-107``).
Step protocol (one correct rule, fixing the reference's divergent skip
logic — SURVEY.md 2.11.10): every rank enters the parameter broadcast for
every parameter; only the owner updates, and only the owner checks grad
presence.
"""

from .comm import default_comm


class _ZeroOptimMixin:
    broadcast_params_after_step = True  # ZeRO-1/2; ZeRO-3 gathers JIT instead

    def _setup_zero(self, param_part_table, ranks_map, comm):
        if param_part_table is None:
            raise ValueError("param_part_table (partition) is required")
        self.comm = comm if comm is not None else default_comm()
        self.parts = dict(param_part_table)
        self.ranks_map = ranks_map

    def _owner(self, name):
        return int(self.parts[name])

    def _owns(self, name):
        return self._owner(name) == self.comm.rank

    # state is allocated only on the owner
    def _owns_state(self, name, param):
        return param.numel() > 0 and self._owns(name)

    # only the owner updates; it alone checks grad presence
    def _should_update(self, name, param):
        return self._owns(name) and param.numel() > 0 and param.grad is not None

    def pre_step(self):
        # device-side wait for in-flight grad reduces
        self.comm.sync()

    def post_step(self):
        if self.broadcast_params_after_step and not self.comm._inactive():
            # refresh replicas: bucketed async broadcasts from each owner,
            # identical order on all ranks, then one stream wait
            self.comm.broadcast_bucketed(
                [(p.data, self._owner(n)) for n, p in self.params.items()]
            )
            self.comm.sync()
