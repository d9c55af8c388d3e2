"""Training utilities: init, logging, LR schedules, accuracy.

Parity with reference utils.py:10-47 (adjust_learning_rate, accuracy),
:179-331 (saveargs, weights_init, init_model, print_model, act_fn).
"""

import math
import os
from math import cos, pi

import torch
import torch.nn as nn


def adjust_learning_rate(args, optimizer, epoch, iteration, num_iter):
    """step/cos/linear/schedule decay with warmup (utils.py:10-39)."""
    warmup_epoch = 5 if getattr(args, 'warmup', False) else 0
    warmup_iter = warmup_epoch * num_iter
    current_iter = iteration + epoch * num_iter
    max_iter = args.epochs * num_iter

    if args.lr_decay == 'step':
        lr = args.lr * (args.gamma ** ((current_iter - warmup_iter) // (max_iter - warmup_iter)))
    elif args.lr_decay == 'cos':
        lr = args.lr * (1 + cos(pi * (current_iter - warmup_iter) / (max_iter - warmup_iter))) / 2
    elif args.lr_decay == 'linear':
        lr = args.lr * (1 - (current_iter - warmup_iter) / (max_iter - warmup_iter))
    elif args.lr_decay == 'schedule':
        count = sum(1 for s in args.schedule if s <= epoch)
        lr = args.lr * pow(args.gamma, count)
    else:
        raise ValueError('Unknown lr mode {}'.format(args.lr_decay))

    if epoch < warmup_epoch:
        lr = args.lr * current_iter / warmup_iter

    for param_group in optimizer.param_groups:
        param_group['lr'] = lr


def accuracy(output, target):
    with torch.no_grad():
        batch_size = target.size(0)
        pred = output.data.max(1)[1]
        return pred.eq(target.data).sum().item() * 100.0 / batch_size


def saveargs(args):
    path = args.checkpoint_dir
    if not os.path.isdir(path):
        os.makedirs(path)
    with open(os.path.join(path, 'args.txt'), 'w') as f:
        for arg in vars(args):
            f.write(arg + ' ' + str(getattr(args, arg)) + '\n')


def weights_init(m):
    """Default init (utils.py:204-218): conv ~ N(0, sqrt(2/fan_out)),
    BN gamma=1 beta=0, linear kaiming fan_in."""
    if isinstance(m, nn.Conv2d):
        n = m.kernel_size[0] * m.kernel_size[1] * m.out_channels
        m.weight.data.normal_(0, math.sqrt(2. / n))
        if m.bias is not None:
            nn.init.constant_(m.bias, 0)
    elif isinstance(m, nn.BatchNorm2d):
        m.weight.data.fill_(1)
        m.bias.data.zero_()
    elif isinstance(m, nn.Linear):
        nn.init.kaiming_normal_(m.weight, mode='fan_in', nonlinearity='relu')
        if m.bias is not None:
            nn.init.constant_(m.bias, 0)


def init_model(model, args, s=0):
    """utils.py:245-303: apply weights_init, then the optional conv-weight
    init override (--weight_init kn/xn/ku/xu/ortho) + scaling."""
    model.apply(weights_init)
    for n, p in model.named_parameters():
        if 'weight' in n and 'conv' in n:
            if args.weight_init == 'kn':
                torch.nn.init.kaiming_normal_(p, mode='fan_out', nonlinearity='relu')
            elif args.weight_init == 'xn':
                torch.nn.init.xavier_normal_(p, gain=nn.init.calculate_gain('relu'))
            elif args.weight_init == 'ku':
                torch.nn.init.kaiming_uniform_(p, mode='fan_out', nonlinearity='relu')
            elif args.weight_init == 'xu':
                torch.nn.init.xavier_uniform_(p, gain=nn.init.calculate_gain('relu'))
            elif args.weight_init == 'ortho':
                torch.nn.init.orthogonal_(p, gain=args.weight_init_scale_conv)
            if args.weight_init_scale_conv != 1.0 and args.weight_init != 'ortho':
                p.data = p.data * args.weight_init_scale_conv
        elif 'linear' in n and 'weight' in n:
            nn.init.kaiming_normal_(p, mode='fan_in', nonlinearity='relu')


def print_model(model, args, full=False):
    print('\n\n****** Model Configuration ******\n')
    for arg in vars(args):
        print(arg, getattr(args, arg))
    if full:
        print('\n\n****** Model Graph ******\n')
        for arg in vars(model):
            print(arg, getattr(model, arg))
    print('\nModel parameters:\n')
    total = 0
    for name, param in model.named_parameters():
        size = param.numel() / 1000.
        print('{}  {}  {:.2f}k'.format(name, list(param.size()), size))
        total += size
    print('\nModel size: {:.2f}k parameters\n'.format(total))


def act_fn(act):
    table = {
        'relu': nn.ReLU(inplace=False), 'lrelu': nn.LeakyReLU(inplace=True),
        'prelu': nn.PReLU(), 'rrelu': nn.RReLU(inplace=True),
        'elu': nn.ELU(inplace=True), 'selu': nn.SELU(inplace=True),
        'tanh': nn.Tanh(), 'sigmoid': nn.Sigmoid(),
    }
    if act not in table:
        print('Activation function {} is not supported'.format(act))
        return None
    return table[act]


def print_batchnorm(model, i):
    print('\nIteration', i)
    for name in ('bn1', 'bn2', 'bn3', 'bn4'):
        bn = getattr(model, name, None)
        if bn is None:
            continue
        print('\n%s.weight\n' % name, bn.weight.detach().cpu().numpy())
        print('%s.bias\n' % name, bn.bias.detach().cpu().numpy())
        if bn.running_var is not None:
            print('%s run_var\n' % name, bn.running_var.detach().cpu().numpy())
            print('%s run_mean\n' % name, bn.running_mean.detach().cpu().numpy())
