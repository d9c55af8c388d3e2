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


class NvNovoGrad(Optimizer):
    """Nvidia-variant NovoGrad (reference timm/optim/nvnovograd.py:13-120):
    zero-initialised first moment (vs NovoGrad's bias-corrected first step)
    and an optional AMSGrad max on the layer-wise second moment."""

    def __init__(self, params, lr=1e-3, betas=(0.95, 0.98), eps=1e-8,
                 weight_decay=0, grad_averaging=False, amsgrad=False):
        if lr < 0.0 or eps < 0.0:
            raise ValueError('NvNovoGrad: bad lr/eps')
        if not (0.0 <= betas[0] < 1.0 and 0.0 <= betas[1] < 1.0):
            raise ValueError('NvNovoGrad: bad betas')
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay,
                        grad_averaging=grad_averaging, amsgrad=amsgrad)
        super().__init__(params, defaults)

    def __setstate__(self, state):
        super().__setstate__(state)
        for group in self.param_groups:
            group.setdefault('amsgrad', False)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            beta1, beta2 = group['betas']
            for p in group['params']:
                if p.grad is None:
                    continue
                if p.grad.is_sparse:
                    raise RuntimeError('NvNovoGrad: sparse grads unsupported')
                grad = p.grad.clone()
                state = self.state[p]
                if len(state) == 0:
                    state['step'] = 0
                    state['exp_avg'] = torch.zeros_like(p)
                    state['exp_avg_sq'] = torch.zeros(
                        [], device=p.device, dtype=torch.float32)
                    if group['amsgrad']:
                        state['max_exp_avg_sq'] = torch.zeros(
                            [], device=p.device, dtype=torch.float32)
                state['step'] += 1
                norm = grad.float().pow(2).sum()
                sq = state['exp_avg_sq']
                if float(sq) == 0.0:
                    sq.copy_(norm)
                else:
                    sq.mul_(beta2).add_(norm, alpha=1.0 - beta2)
                if group['amsgrad']:
                    torch.maximum(state['max_exp_avg_sq'], sq,
                                  out=state['max_exp_avg_sq'])
                    denom = state['max_exp_avg_sq'].sqrt().add_(group['eps'])
                else:
                    denom = sq.sqrt().add_(group['eps'])
                grad.div_(denom)
                if group['weight_decay'] != 0:
                    grad.add_(p, alpha=group['weight_decay'])
                if group['grad_averaging']:
                    grad.mul_(1.0 - beta1)
                ea = state['exp_avg']
                ea.mul_(beta1).add_(grad)
                p.add_(ea, alpha=-group['lr'])
        return loss
