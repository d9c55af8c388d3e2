"""Lookahead optimizer wrapper (reference timm/optim/lookahead.py:10)."""

from collections import defaultdict

import torch


class Lookahead:
    """Wraps a base optimizer; not an Optimizer subclass (modern torch
    Optimizer internals require going through Optimizer.__init__, which a
    wrapper sharing the base's param_groups cannot do)."""

    def __init__(self, base_optimizer, alpha=0.5, k=6):
        if not 0.0 <= alpha <= 1.0:
            raise ValueError('Invalid slow update rate: %f' % alpha)
        if not 1 <= k:
            raise ValueError('Invalid lookahead steps: %d' % k)
        defaults = dict(lookahead_alpha=alpha, lookahead_k=k,
                        lookahead_step=0)
        self.base_optimizer = base_optimizer
        self.param_groups = self.base_optimizer.param_groups
        self.defaults = base_optimizer.defaults
        self.defaults.update(defaults)
        self.state = defaultdict(dict)
        for name, default in defaults.items():
            for group in self.param_groups:
                group.setdefault(name, default)

    def update_slow(self, group):
        for fast_p in group['params']:
            if fast_p.grad is None:
                continue
            param_state = self.state[fast_p]
            if 'slow_buffer' not in param_state:
                param_state['slow_buffer'] = torch.empty_like(fast_p)
                param_state['slow_buffer'].copy_(fast_p)
            slow = param_state['slow_buffer']
            slow.add_(fast_p - slow, alpha=group['lookahead_alpha'])
            fast_p.data.copy_(slow)

    def sync_lookahead(self):
        for group in self.param_groups:
            self.update_slow(group)

    def step(self, closure=None):
        loss = self.base_optimizer.step(closure)
        for group in self.param_groups:
            group['lookahead_step'] += 1
            if group['lookahead_step'] % group['lookahead_k'] == 0:
                self.update_slow(group)
        return loss

    def state_dict(self):
        fast_state_dict = self.base_optimizer.state_dict()
        slow_state = {(id(k) if isinstance(k, torch.Tensor) else k): v
                      for k, v in self.state.items()}
        return {'state': fast_state_dict['state'],
                'slow_state': slow_state,
                'param_groups': fast_state_dict['param_groups']}

    def load_state_dict(self, state_dict):
        self.base_optimizer.load_state_dict(
            {'state': state_dict['state'],
             'param_groups': state_dict['param_groups']})
        self.param_groups = self.base_optimizer.param_groups

    def zero_grad(self, set_to_none=True):
        self.base_optimizer.zero_grad(set_to_none=set_to_none)

    def add_param_group(self, param_group):
        self.base_optimizer.add_param_group(param_group)
