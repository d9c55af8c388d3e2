"""Optimizer factory (reference timm/optim/optim_factory.py:11-97).

The sgd/adamw paths use the fused HIP-update optimizers
(noisynet_amd.optim); the exotic ones (radam/novograd/rmsproptf/...) use
their own implementations below.
"""

import torch.optim as optim

from ... import optim as native_optim
from .adamw import AdamW
from .lookahead import Lookahead
from .nadam import Nadam
from .novograd import NovoGrad, NvNovoGrad
from .radam import RAdam
from .rmsprop_tf import RMSpropTF


def add_weight_decay(model, weight_decay=1e-5, skip_list=()):
    """No weight decay for 1-dim params and biases (:11-23)."""
    decay, no_decay = [], []
    for name, param in model.named_parameters():
        if not param.requires_grad:
            continue
        if len(param.shape) == 1 or name.endswith(".bias") or name in skip_list:
            no_decay.append(param)
        else:
            decay.append(param)
    return [{'params': no_decay, 'weight_decay': 0.},
            {'params': decay, 'weight_decay': weight_decay}]


def create_optimizer(args, model, filter_bias_and_bn=True):
    opt_lower = args.opt.lower()
    weight_decay = args.weight_decay
    if weight_decay and filter_bias_and_bn:
        parameters = add_weight_decay(model, weight_decay)
        weight_decay = 0.
    else:
        parameters = model.parameters()

    opt_split = opt_lower.split('_')
    opt_lower = opt_split[-1]
    if opt_lower in ('sgd', 'nesterov', 'fusedsgd'):
        optimizer = native_optim.SGD(parameters, lr=args.lr,
                                     momentum=args.momentum,
                                     weight_decay=weight_decay, nesterov=True)
    elif opt_lower == 'momentum':
        optimizer = native_optim.SGD(parameters, lr=args.lr,
                                     momentum=args.momentum,
                                     weight_decay=weight_decay, nesterov=False)
    elif opt_lower in ('adam', 'fusedadam'):
        optimizer = native_optim.Adam(parameters, lr=args.lr,
                                      weight_decay=weight_decay,
                                      eps=args.opt_eps)
    elif opt_lower in ('adamw', 'fusedadamw'):
        optimizer = native_optim.AdamW(parameters, lr=args.lr,
                                       weight_decay=weight_decay,
                                       eps=args.opt_eps)
    elif opt_lower == 'nadam':
        optimizer = Nadam(parameters, lr=args.lr, weight_decay=weight_decay,
                          eps=args.opt_eps)
    elif opt_lower == 'radam':
        optimizer = RAdam(parameters, lr=args.lr, weight_decay=weight_decay,
                          eps=args.opt_eps)
    elif opt_lower == 'adadelta':
        optimizer = optim.Adadelta(parameters, lr=args.lr,
                                   weight_decay=weight_decay, eps=args.opt_eps)
    elif opt_lower == 'rmsprop':
        optimizer = optim.RMSprop(parameters, lr=args.lr, alpha=0.9,
                                  eps=args.opt_eps, momentum=args.momentum,
                                  weight_decay=weight_decay)
    elif opt_lower == 'rmsproptf':
        optimizer = RMSpropTF(parameters, lr=args.lr, alpha=0.9,
                              eps=args.opt_eps, momentum=args.momentum,
                              weight_decay=weight_decay)
    elif opt_lower == 'novograd':
        optimizer = NovoGrad(parameters, lr=args.lr,
                             weight_decay=weight_decay, eps=args.opt_eps)
    elif opt_lower == 'nvnovograd':
        optimizer = NvNovoGrad(parameters, lr=args.lr,
                               weight_decay=weight_decay, eps=args.opt_eps)
    # apex Fused* names (reference optim_factory.py:75-97): this framework's
    # SGD/Adam/AdamW ARE fused-kernel updates (csrc/optimizer.hip), so the
    # names resolve to them instead of requiring apex
    elif opt_lower == 'fusedsgd':
        optimizer = native_optim.SGD(parameters, lr=args.lr,
                                     momentum=args.momentum,
                                     weight_decay=weight_decay, nesterov=True)
    elif opt_lower == 'fusedadam':
        optimizer = native_optim.Adam(parameters, lr=args.lr,
                                      weight_decay=weight_decay,
                                      eps=args.opt_eps)
    elif opt_lower == 'fusedadamw':
        optimizer = native_optim.AdamW(parameters, lr=args.lr,
                                       weight_decay=weight_decay,
                                       eps=args.opt_eps)
    elif opt_lower == 'fusednovograd':
        optimizer = NvNovoGrad(parameters, lr=args.lr,
                               weight_decay=weight_decay, eps=args.opt_eps)
    else:
        raise ValueError("Invalid optimizer %s" % args.opt)

    if len(opt_split) > 1 and opt_split[0] == 'lookahead':
        optimizer = Lookahead(optimizer)
    return optimizer
