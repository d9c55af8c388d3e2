"""Hyperbolic-tangent decay with restarts (reference timm/scheduler/tanh_lr.py)."""

import math

from .scheduler import Scheduler


class TanhLRScheduler(Scheduler):
    def __init__(self, optimizer, t_initial, lb=-6., ub=4., t_mul=1.,
                 lr_min=0., decay_rate=1., warmup_t=0, warmup_lr_init=0,
                 warmup_prefix=False, cycle_limit=0, t_in_epochs=True,
                 noise_range_t=None, noise_pct=0.67, noise_std=1.0,
                 noise_seed=42, initialize=True):
        super().__init__(optimizer, 'lr', noise_range_t=noise_range_t,
                         noise_pct=noise_pct, noise_std=noise_std,
                         noise_seed=noise_seed, initialize=initialize)
        assert t_initial > 0 and lb < ub
        self.lb = lb
        self.ub = ub
        self.t_initial = t_initial
        self.t_mul = t_mul
        self.lr_min = lr_min
        self.decay_rate = decay_rate
        self.cycle_limit = cycle_limit
        self.warmup_t = warmup_t
        self.warmup_lr_init = warmup_lr_init
        self.warmup_prefix = warmup_prefix
        self.t_in_epochs = t_in_epochs
        if self.warmup_t:
            t_v = self.base_values if self.warmup_prefix else self._get_lr(self.warmup_t)
            self.warmup_steps = [(v - warmup_lr_init) / self.warmup_t for v in t_v]
            super().update_groups(self.warmup_lr_init)
        else:
            self.warmup_steps = [1 for _ in self.base_values]

    def _get_lr(self, t):
        if t < self.warmup_t:
            lrs = [self.warmup_lr_init + t * s for s in self.warmup_steps]
        else:
            if self.warmup_prefix:
                t = t - self.warmup_t
            if self.t_mul != 1:
                i = math.floor(math.log(1 - t / self.t_initial * (1 - self.t_mul),
                                        self.t_mul))
                t_i = self.t_mul ** i * self.t_initial
                t_curr = t - (1 - self.t_mul ** i) / (1 - self.t_mul) * self.t_initial
            else:
                i = t // self.t_initial
                t_i = self.t_initial
                t_curr = t - (self.t_initial * i)
            if self.cycle_limit == 0 or (self.cycle_limit > 0 and i < self.cycle_limit):
                gamma = self.decay_rate ** i
                lr_min = self.lr_min * gamma
                lr_max_values = [v * gamma for v in self.base_values]
                tr = t_curr / t_i
                lrs = [lr_min + 0.5 * (lr_max - lr_min)
                       * (1 - math.tanh(self.lb * (1. - tr) + self.ub * tr))
                       for lr_max in lr_max_values]
            else:
                lrs = [self.lr_min * (self.decay_rate ** self.cycle_limit)
                       for _ in self.base_values]
        return lrs

    def get_epoch_values(self, epoch):
        return self._get_lr(epoch) if self.t_in_epochs else None

    def get_update_values(self, num_updates):
        return self._get_lr(num_updates) if not self.t_in_epochs else None

    def get_cycle_length(self, cycles=0):
        if not cycles:
            cycles = self.cycle_limit
        cycles = max(1, cycles)
        if self.t_mul == 1.0:
            return self.t_initial * cycles
        return int(math.floor(-self.t_initial * (self.t_mul ** cycles - 1)
                              / (1 - self.t_mul)))
