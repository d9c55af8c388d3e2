"""Robustness-evaluation harness.

Rebuilds main.py:278-654: gradient-based weight selection, multiplicative
weight distortion, stuck-at fault injection, temperature drift, pruning,
weight scaling, activation distortion, and BN folding -- over any model that
names its layers conv*/linear*/fc*/classifier* like the reference.
"""

import copy

import numpy as np
import torch
import torch.nn as nn

from . import ops


def _weight_params(model):
    params = []
    for n, p in model.named_parameters():
        if (('conv' in n or 'fc' in n or 'classifier' in n or 'linear' in n)
                and 'weight' in n):
            params.append(p)
    return params


def _iter_val(val_loader, args):
    """Yield (input, label) whether val_loader is a GPU-resident tuple
    (CIFAR path) or an iterable of batches."""
    if isinstance(val_loader, tuple):
        inputs, labels = val_loader
        n = inputs.shape[0] // args.batch_size
        for i in range(n):
            yield (inputs[i * args.batch_size:(i + 1) * args.batch_size],
                   labels[i * args.batch_size:(i + 1) * args.batch_size])
    else:
        for images, target in val_loader:
            yield images, target


def get_gradients(model, args, val_loader):
    """Accumulate |dL/dW| over validation batches (main.py:278-322)."""
    params = _weight_params(model)
    grads = [torch.zeros_like(p) for p in params]
    criterion = nn.CrossEntropyLoss()
    count = 0
    for input, label in _iter_val(val_loader, args):
        output = model(input)
        loss = criterion(output, label)
        batch_grads = torch.autograd.grad(loss, params)
        for bg, grad in zip(batch_grads, grads):
            grad += torch.abs(bg)
        count += 1
        if not isinstance(val_loader, tuple) and count > 10:
            break
    return grads


def select_values(args, params, grads):
    """Top-K% selection by weight/grad/taylor product (main.py:325-348)."""
    pctls, values_list = [], []
    for p, g in zip(params, grads):
        if args.selection_criteria == 'grad_magnitude':
            k = int(g.numel() * (100 - args.selected_weights) / 100.0)
            pctl, _ = torch.kthvalue(torch.abs(g.view(-1)), k)
            values = g.data
        elif args.selection_criteria == 'weight_magnitude':
            k = int(p.numel() * (100 - args.selected_weights) / 100.0)
            pctl, _ = torch.kthvalue(torch.abs(p.view(-1)), k)
            values = p.clone().data
        elif args.selection_criteria == 'combined':
            k = int(g.numel() * (100 - args.selected_weights) / 100.0)
            pctl, _ = torch.kthvalue(torch.abs((g * p).view(-1)), k)
            values = g.data * p.clone().data
        else:
            raise SystemExit('Unknown selection criteria: {}'.format(args.selection_criteria))
        pctls.append(pctl)
        values_list.append(values)
    return pctls, values_list


def distort_weights(args, params, grads=None, values=None, pctls=None, noise=0.0):
    """Multiplicative uniform noise, reduced for selected weights
    (main.py:351-377)."""
    with torch.no_grad():
        if values is None:
            values = [0] * len(params)
        if pctls is None:
            pctls = [0] * len(params)
        for p, v, pctl in zip(params, values, pctls):
            p_noise = p * torch.empty_like(p).uniform_(-noise, noise)
            if args.selected_weights > 0:
                p.data = torch.where(torch.abs(v) < pctl, p.data + p_noise,
                                     p.data + p_noise * args.selected_weights_noise_scale)
            else:
                p.data.add_(p_noise)


def _stuck_at(args, p, noise):
    """Stuck-at fault modes (main.py:448-493)."""
    mode = args.stuck_at_weights
    if mode == 'random_zero':
        mask = torch.empty_like(p).uniform_() > noise
        p.data = p.data * mask
    elif mode == 'largest_zero':
        thr = 1 - noise
        pos = p[p > 0]
        neg = p[p < 0].abs()
        pctl_pos, _ = torch.kthvalue(pos.flatten(), max(1, int(pos.numel() * thr)))
        pctl_neg, _ = torch.kthvalue(neg.flatten(), max(1, int(neg.numel() * thr)))
        p.data[p.data > pctl_pos] = 0
        p.data[p.data < -pctl_neg] = 0
    elif mode == 'smallest_zero':
        pos = p[p > 0]
        neg = p[p < 0].abs()
        pctl_pos, _ = torch.kthvalue(pos.flatten(), max(1, int(pos.numel() * noise)))
        pctl_neg, _ = torch.kthvalue(neg.flatten(), max(1, int(neg.numel() * noise)))
        p_copy_pos = p.data.clone()
        p_copy_neg = p.data.clone()
        p_copy_pos[p.data < pctl_pos] = 0
        p_copy_neg[p.data > -pctl_neg] = 0
        p.data = p_copy_pos + p_copy_neg
    elif mode == 'random_one':
        mask = torch.empty_like(p).uniform_() > noise
        p_copy = p.data.clone()
        p.data = torch.where(mask, p_copy, p_copy.sign() * p_copy.abs().max())
    else:
        raise ValueError(mode)


def test_distortion(model, args, val_loader=None, mode='weights', vars=None,
                    exit_after=False):
    """Sweep noise levels x num_sims, restoring state each time
    (main.py:380-538). Returns the list of mean accuracies per level.

    Unlike the reference (which hard-exits via raise SystemExit at :529),
    this returns results; pass exit_after=True for reference behaviour.
    """
    model.eval()
    if mode == 'weights':
        orig_m = copy.deepcopy(model.state_dict())
    if mode == 'acts':
        args.distort_act = True

    acc_d, error_bars = [], []
    if args.noise > 0:
        vars = [args.noise]

    params = _weight_params(model)
    if args.selected_weights > 0:
        grads = get_gradients(model, args, val_loader)
        pctls, values = select_values(args, params, grads)
    else:
        pctls = values = None

    for noise in vars:
        te_acc_dist = []
        for s in range(args.num_sims):
            if mode == 'weights':
                if args.scale_weights > 0:
                    with torch.no_grad():
                        for p in params:
                            p.data = args.scale_weights * p.data
                elif getattr(args, 'test_temp', 0) > 0:
                    with torch.no_grad():
                        for p in params:
                            p.data = (p.data.sign() * p.data.abs().max()
                                      * (p.data.abs() / p.data.abs().max())
                                      ** ((args.test_temp + 273.) / (args.temperature + 273.)))
                elif getattr(args, 'stuck_at_weights', None) is not None:
                    with torch.no_grad():
                        for p in params:
                            _stuck_at(args, p, noise)
                else:
                    distort_weights(args, params, values=values, pctls=pctls,
                                    noise=noise)

            te_accs = []
            with torch.no_grad():
                for input, label in _iter_val(val_loader, args):
                    output = model(input)
                    pred = output.data.max(1)[1]
                    te_accs.append(pred.eq(label.data).float().mean().item() * 100.0)
            te_acc_d = float(np.mean(te_accs, dtype=np.float64))
            te_acc_dist.append(te_acc_d)

            if mode == 'weights':
                model.load_state_dict(orig_m)

        avg = float(np.mean(te_acc_dist, dtype=np.float64))
        error_bars.append(te_acc_dist)
        acc_d.append(avg)
        print('Noise {:>5.2f}: {}  avg acc {:>5.2f}'.format(
            noise, [float('{:.2f}'.format(v)) for v in te_acc_dist], avg))

    if mode == 'acts':
        args.distort_act = False
    if exit_after:
        raise SystemExit
    if args.distort_w_test and getattr(args, 'var_name', None):
        return [float('{0:.2f}'.format(x)) for x in acc_d]
    return float(np.mean(acc_d, dtype=np.float64)) if acc_d else 0.0


def merge_batchnorm(model, args):
    """Fold BN gamma/sqrt(var) into the preceding conv/linear weights in
    place (main.py:540-654). The bias term is then added at forward time
    from BN stats by the models' merge_bn paths."""
    print('\nMerging batchnorm into weights...\n')
    eps = getattr(args, 'eps', 1e-7)
    arch = args.arch

    def scale_of(bn, conv_dims):
        view = (-1, 1, 1, 1) if conv_dims == 4 else (-1, 1)
        return (bn.weight.data.view(*view)
                / torch.sqrt(bn.running_var.data.view(*view) + eps))

    if arch == 'noisynet':
        model.conv1.weight.data *= scale_of(model.bn1, 4)
        model.conv2.weight.data *= scale_of(model.bn2, 4)
        model.linear1.weight.data *= scale_of(model.bn3, 2)
        model.linear2.weight.data *= scale_of(model.bn4, 2)
    elif arch == 'resnet18':
        m = model.module if hasattr(model, 'module') else model
        m.conv1.weight.data *= scale_of(m.bn1, 4)
        for layer in (m.layer1, m.layer2, m.layer3, m.layer4):
            for block in layer:
                block.conv1.weight.data *= scale_of(block.bn1, 4)
                block.conv2.weight.data *= scale_of(block.bn2, 4)
                if block.downsample is not None:
                    block.conv3.weight.data *= scale_of(block.bn3, 4)
    elif arch == 'mobilenet_v2':
        m = model.module if hasattr(model, 'module') else model
        for mod in m.modules():
            conv = getattr(mod, 'conv', None)
            bn = getattr(mod, 'bn', None)
            if isinstance(conv, nn.Conv2d) and isinstance(bn, nn.BatchNorm2d):
                conv.weight.data *= scale_of(bn, 4)
    else:
        raise ValueError('merge_batchnorm: unknown arch %s' % arch)
