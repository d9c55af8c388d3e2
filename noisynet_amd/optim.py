"""Optimizers backed by the fused HIP update kernels.

torch.optim-compatible SGD / Adam / AdamW whose per-parameter update is one
fused kernel call (grad + L2 + momentum + step + optional weight clamp in a
single pass over the parameter, csrc/optimizer.hip) -- the reference runs
these as a chain of eager elementwise ops plus a separate post-step
``p.data.clamp_(-w_max, w_max)`` (noisynet.py:1520-1542).

Per-param-group ``weight_decay``/``lr`` (per-layer groups,
noisynet.py:1135-1161) are honoured; a group may carry ``clamp`` =
(min, max) to fold the post-step weight clip into the update.
"""

import torch

from . import ops


class SGD(torch.optim.Optimizer):
    def __init__(self, params, lr, momentum=0.0, weight_decay=0.0,
                 nesterov=False):
        defaults = dict(lr=lr, momentum=momentum, weight_decay=weight_decay,
                        nesterov=nesterov, clamp=None)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            clamp = group.get('clamp') or (0.0, 0.0)
            for p in group['params']:
                if p.grad is None:
                    continue
                state = self.state[p]
                if group['momentum'] != 0 and 'momentum_buffer' not in state:
                    state['momentum_buffer'] = torch.zeros_like(p)
                buf = state.get('momentum_buffer',
                                torch.empty(0, device=p.device, dtype=p.dtype))
                ops.sgd_step(p, p.grad, buf, group['lr'], group['momentum'],
                             group['weight_decay'], group['nesterov'],
                             clamp[0], clamp[1])
        return loss


class AdamW(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=1e-2, amsgrad=False):
        if amsgrad:
            # amsgrad path kept eager (rarely used; --amsgrad)
            pass
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay, amsgrad=amsgrad, clamp=None)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            clamp = group.get('clamp') or (0.0, 0.0)
            beta1, beta2 = group['betas']
            for p in group['params']:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state['step'] = 0
                    state['exp_avg'] = torch.zeros_like(p)
                    state['exp_avg_sq'] = torch.zeros_like(p)
                    if group['amsgrad']:
                        state['max_exp_avg_sq'] = torch.zeros_like(p)
                state['step'] += 1
                if group['amsgrad']:
                    # eager amsgrad fallback (reference --amsgrad)
                    p.mul_(1 - group['lr'] * group['weight_decay'])
                    ea, eas = state['exp_avg'], state['exp_avg_sq']
                    ea.mul_(beta1).add_(p.grad, alpha=1 - beta1)
                    eas.mul_(beta2).addcmul_(p.grad, p.grad, value=1 - beta2)
                    torch.maximum(state['max_exp_avg_sq'], eas,
                                  out=state['max_exp_avg_sq'])
                    bc1 = 1 - beta1 ** state['step']
                    bc2 = 1 - beta2 ** state['step']
                    denom = (state['max_exp_avg_sq'] / bc2).sqrt().add_(group['eps'])
                    p.addcdiv_(ea, denom, value=-group['lr'] / bc1)
                    if clamp[1] > clamp[0]:
                        p.clamp_(clamp[0], clamp[1])
                else:
                    ops.adamw_step(p, p.grad, state['exp_avg'],
                                   state['exp_avg_sq'], state['step'],
                                   group['lr'], beta1, beta2, group['eps'],
                                   group['weight_decay'], clamp[0], clamp[1])
        return loss


class Adam(AdamW):
    """Adam = AdamW with coupled L2: we emulate by adding wd*p to grad."""

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            wd = group['weight_decay']
            if wd != 0:
                for p in group['params']:
                    if p.grad is not None:
                        p.grad = p.grad.add(p, alpha=wd)
            saved_wd = group['weight_decay']
            group['weight_decay'] = 0.0
            try:
                clamp = group.get('clamp') or (0.0, 0.0)
                beta1, beta2 = group['betas']
                for p in group['params']:
                    if p.grad is None:
                        continue
                    state = self.state[p]
                    if len(state) == 0:
                        state['step'] = 0
                        state['exp_avg'] = torch.zeros_like(p)
                        state['exp_avg_sq'] = torch.zeros_like(p)
                    state['step'] += 1
                    ops.adamw_step(p, p.grad, state['exp_avg'],
                                   state['exp_avg_sq'], state['step'],
                                   group['lr'], beta1, beta2, group['eps'],
                                   0.0, clamp[0], clamp[1])
            finally:
                group['weight_decay'] = saved_wd
        return loss
