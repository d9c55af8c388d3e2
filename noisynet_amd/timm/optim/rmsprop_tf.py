"""RMSprop with Tensorflow semantics: eps inside the sqrt, one-based rho
initialization (reference timm/optim/rmsprop_tf.py:5)."""

import torch
from torch.optim.optimizer import Optimizer


class RMSpropTF(Optimizer):
    def __init__(self, params, lr=1e-2, alpha=0.9, eps=1e-10, weight_decay=0,
                 momentum=0., centered=False, decoupled_decay=False,
                 lr_in_momentum=True):
        defaults = dict(lr=lr, momentum=momentum, alpha=alpha, eps=eps,
                        centered=centered, weight_decay=weight_decay,
                        decoupled_decay=decoupled_decay,
                        lr_in_momentum=lr_in_momentum)
        super().__init__(params, defaults)

    def __setstate__(self, state):
        super().__setstate__(state)
        for group in self.param_groups:
            group.setdefault('momentum', 0)
            group.setdefault('centered', False)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            for p in group['params']:
                if p.grad is None:
                    continue
                grad = p.grad
                state = self.state[p]
                if len(state) == 0:
                    state['step'] = 0
                    state['square_avg'] = torch.ones_like(p)  # TF inits to 1
                    if group['momentum'] > 0:
                        state['momentum_buffer'] = torch.zeros_like(p)
                    if group['centered']:
                        state['grad_avg'] = torch.zeros_like(p)
                square_avg = state['square_avg']
                one_minus_alpha = 1. - group['alpha']
                state['step'] += 1
                if group['weight_decay'] != 0:
                    if group['decoupled_decay']:
                        p.mul_(1. - group['lr'] * group['weight_decay'])
                    else:
                        grad = grad.add(p, alpha=group['weight_decay'])
                square_avg.add_(grad.pow(2) - square_avg, alpha=one_minus_alpha)
                if group['centered']:
                    grad_avg = state['grad_avg']
                    grad_avg.add_(grad - grad_avg, alpha=one_minus_alpha)
                    avg = square_avg.addcmul(grad_avg, grad_avg, value=-1) \
                        .add(group['eps']).sqrt_()  # eps inside sqrt
                else:
                    avg = square_avg.add(group['eps']).sqrt_()
                if group['momentum'] > 0:
                    buf = state['momentum_buffer']
                    if group['lr_in_momentum']:
                        buf.mul_(group['momentum']).addcdiv_(grad, avg,
                                                             value=group['lr'])
                        p.add_(-buf)
                    else:
                        buf.mul_(group['momentum']).addcdiv_(grad, avg)
                        p.add_(buf, alpha=-group['lr'])
                else:
                    p.addcdiv_(grad, avg, value=-group['lr'])
        return loss
