"""NovoGrad: layer-wise second moment (reference timm/optim/novograd.py /
nvnovograd.py)."""

import torch
from torch.optim.optimizer import Optimizer


class NovoGrad(Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.95, 0.98), eps=1e-8,
                 weight_decay=0, grad_averaging=False, amsgrad=False):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay,
                        grad_averaging=grad_averaging, amsgrad=amsgrad)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            for p in group['params']:
                if p.grad is None:
                    continue
                grad = p.grad
                state = self.state[p]
                g_2 = torch.sum(grad.float() ** 2)
                if len(state) == 0:
                    state['step'] = 0
                    state['moments'] = grad.div(g_2.sqrt() + group['eps']) + \
                        group['weight_decay'] * p
                    state['grads_ema'] = g_2
                moments = state['moments']
                grads_ema = state['grads_ema']
                beta1, beta2 = group['betas']
                state['step'] += 1
                grads_ema.mul_(beta2).add_(g_2, alpha=1. - beta2)
                denom = grads_ema.sqrt() + group['eps']
                grad_u = grad / denom
                if group['weight_decay'] != 0:
                    grad_u = grad_u.add(p, alpha=group['weight_decay'])
                if group['grad_averaging']:
                    grad_u = grad_u * (1. - beta1)
                moments.mul_(beta1).add_(grad_u)
                p.add_(moments, alpha=-group['lr'])
        return loss
