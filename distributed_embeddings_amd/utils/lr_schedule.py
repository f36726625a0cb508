"""Warmup + polynomial-decay learning-rate schedule.

Parity: reference ``examples/dlrm/utils.py:45-88`` (LearningRateScheduler):
linear warmup to ``base_lr`` over ``warmup_steps``, constant until
``decay_start``, then quadratic polynomial decay to ~0 over ``decay_steps``.
"""


class WarmupPolyDecay:
    """``fused_modules``: modules with ``set_fused_lr`` (e.g.
    ``DistributedEmbedding`` with the in-backward fused optimizer enabled)
    whose device-resident lr follows the schedule too."""

    def __init__(self, optimizer, base_lr: float, warmup_steps: int = 0,
                 decay_start: int = 0, decay_steps: int = 0, power: float = 2.0,
                 end_lr: float = 0.0, fused_modules=()):
        self.opt = optimizer
        self.base_lr = base_lr
        self.warmup_steps = warmup_steps
        self.decay_start = decay_start
        self.decay_steps = decay_steps
        self.power = power
        self.end_lr = end_lr
        self.fused_modules = list(fused_modules)
        self._step = 0

    def lr_at(self, step: int) -> float:
        if self.warmup_steps and step < self.warmup_steps:
            return self.base_lr * (step + 1) / self.warmup_steps
        if self.decay_steps and step >= self.decay_start:
            t = min(step - self.decay_start, self.decay_steps) / self.decay_steps
            return (self.base_lr - self.end_lr) * (1 - t) ** self.power + self.end_lr
        return self.base_lr

    def step(self):
        lr = self.lr_at(self._step)
        groups = getattr(self.opt, "param_groups", None)
        if groups is None:
            groups = self.opt.optimizer.param_groups
        for g in groups:
            g["lr"] = lr
        for m in self.fused_modules:
            m.set_fused_lr(lr)
        self._step += 1
        return lr
