"""ImageNet training driver (the `python main.py -a resnet18 ...` entrypoint).

Reproduces the reference main.py flow: build_model with
arch select + distributed wiring, DALI-style GPU-resident data (here: the
synthetic ImageNet-shaped generator -- no datasets in this environment),
the step/cos/linear LR schedules with warmup, L1/L3 penalties, calibration
stop at batch 5, post-step weight clamp/percentile clip, per-epoch validate
+ rank-0 checkpointing ({epoch, arch, state_dict, best_acc, optimizer}),
resume with name-matched partial copy, and the distortion/fault-injection
sweeps on --evaluate/--distort_w_test.

MI355X: one process per GPU over RCCL (torchrun env contract), our bucketed
overlap all-reduce (noisynet_amd.distributed.DataParallel), channels_last.
"""

import os
import random

import numpy as np
import torch
import torch.nn as nn

from .. import data as data_mod
from .. import distributed as dist_mod
from .. import ops
from .. import optim as native_optim
from .. import utils
from ..config import build_main_parser
from ..harness import merge_batchnorm, test_distortion
from ..quant import QuantMeasure, finish_calibration
from ..models.resnet import ResNet18


def build_model(args):
    if args.arch == 'resnet18':
        model = ResNet18(args)
    elif args.arch == 'mobilenet_v2':
        from ..models.mobilenet import mobilenet_v2
        model = mobilenet_v2(args)
    elif args.arch == 'efficientnet_b0':
        from ..models.efficientnet import efficientnet_b0
        model = efficientnet_b0(args)
    else:
        raise ValueError('unknown arch %s' % args.arch)

    distributed = dist_mod.init_distributed()
    device = torch.device('cuda', dist_mod.env_local_rank()) \
        if torch.cuda.is_available() else torch.device('cpu')
    model = model.to(device)
    if args.fp16:
        model = model.half()
    if getattr(args, 'bf16', False):
        model = model.bfloat16()
        for m in model.modules():
            if isinstance(m, (nn.BatchNorm1d, nn.BatchNorm2d)):
                m.float()
    if device.type == 'cuda':
        model = model.to(memory_format=torch.channels_last)

    # --sync-bn uses the FUSED SyncBN: models pass sync=args.sync_bn into
    # ops.bn_act, which all-reduces (mean, E[x^2]) across ranks inside the
    # fused BN kernel path (SURVEY.md §5 distributed row).

    dp = dist_mod.DataParallel(model) if distributed else None
    return model, dp, device, distributed


def setup_data(args, device):
    """Synthetic ImageNet-shaped loaders, rank-sharded (the DALI stand-in)."""
    rank = dist_mod.env_rank()
    ws = dist_mod.env_world_size()
    dtype = torch.bfloat16 if getattr(args, 'bf16', False) else (
        torch.float16 if args.fp16 else torch.float32)
    train_loader = data_mod.SyntheticImageNet(
        args.batch_size, num_batches=args.synthetic_batches, device=device,
        dtype=dtype, seed=1234, rank=rank, world_size=ws)
    val_loader = data_mod.SyntheticImageNet(
        args.batch_size, num_batches=max(4, args.synthetic_batches // 10),
        device=device, dtype=dtype, seed=999, rank=rank, world_size=ws)
    return train_loader, val_loader


def validate(val_loader, model, args, epoch=0):
    model.eval()
    accs = []
    with torch.no_grad():
        for images, target in val_loader:
            if images.is_cuda:
                images = images.contiguous(memory_format=torch.channels_last)
            output = model(images)
            accs.append(utils.accuracy(output, target))
    return float(np.mean(accs)) if accs else 0.0


def save_checkpoint(state, path):
    os.makedirs(os.path.dirname(path) or '.', exist_ok=True)
    torch.save(state, path)


def load_from_checkpoint(args, model, optimizer=None, device='cpu'):
    """Name-matched partial copy with DataParallel-prefix detection
    (main.py:212-275)."""
    print('loading checkpoint', args.resume)
    ckpt = torch.load(args.resume, map_location=device, weights_only=False)
    sd = ckpt.get('state_dict', ckpt) if isinstance(ckpt, dict) else ckpt
    own = model.state_dict()
    loaded = 0
    for name, param in sd.items():
        if name.startswith('module.'):
            name = name[len('module.'):]
        if name in own and own[name].shape == torch.as_tensor(param).shape:
            own[name].copy_(param)
            loaded += 1
    print('restored %d/%d tensors' % (loaded, len(own)))
    start_epoch = ckpt.get('epoch', 0) if isinstance(ckpt, dict) else 0
    best_acc = ckpt.get('best_acc', 0) if isinstance(ckpt, dict) else 0
    if args.reset_start_epoch:
        start_epoch = 0
    if optimizer is not None and isinstance(ckpt, dict) and 'optimizer' in ckpt:
        try:
            optimizer.load_state_dict(ckpt['optimizer'])
        except Exception as exc:
            print('optimizer state not restored:', exc)
    return start_epoch, best_acc


def train(model, dp, args, train_loader, val_loader, optimizer, device,
          start_epoch=0):
    # fused softmax-xent HIP kernel unless the L3 gradient penalty needs a
    # twice-differentiable loss (main.py:894-904)
    if args.L3 > 0:
        criterion = nn.CrossEntropyLoss()
    else:
        criterion = lambda out, tgt: ops.cross_entropy(out, tgt)  # noqa: E731
    best_acc = 0.0
    num_iter = len(train_loader)
    for epoch in range(start_epoch, args.epochs):
        model.train()
        for i, (images, target) in enumerate(train_loader):
            utils.adjust_learning_rate(args, optimizer, epoch, i, num_iter)
            if images.is_cuda:
                images = images.contiguous(memory_format=torch.channels_last)
            output = model(images, epoch, i)
            loss = criterion(output.float(), target)

            if args.L1 > 0:
                for n_, p in model.named_parameters():
                    if 'weight' in n_ and ('conv' in n_ or 'fc' in n_):
                        loss = loss + args.L1 * p.norm(p=1)

            optimizer.zero_grad(set_to_none=False)
            if args.L3 > 0:
                params = [p for n_, p in model.named_parameters()
                          if 'weight' in n_ and ('conv' in n_ or 'fc' in n_)]
                grads = torch.autograd.grad(loss, params, create_graph=True)
                gn = sum(g.pow(2).sum() for g in grads)
                loss = loss + args.L3 * gn
            loss.backward()
            if dp is not None:
                dp.finish()

            if args.grad_clip > 0:
                for p in model.parameters():
                    if p.grad is not None:
                        p.grad.data.clamp_(-args.grad_clip, args.grad_clip)
            optimizer.step()

            # calibration stop at batch 5 (main.py:944-951)
            if args.q_a > 0 and args.calculate_running and epoch == start_epoch and i == 5:
                finish_calibration(model, device)

            with torch.no_grad():
                if args.w_max > 0:
                    for n_, p in model.named_parameters():
                        if ('conv' in n_ or 'fc' in n_) and 'weight' in n_:
                            p.data.clamp_(-args.w_max, args.w_max)
                if args.w_pctl > 0:
                    for n_, p in model.named_parameters():
                        if ('conv' in n_ or 'fc' in n_) and 'weight' in n_:
                            thr = torch.quantile(p.abs().float().flatten(),
                                                 args.w_pctl / 100.0)
                            p.data.clamp_(-thr, thr)

            if i % args.print_freq == 0 and dist_mod.env_rank() == 0:
                print('epoch {} it {}/{} loss {:.4f}'.format(
                    epoch, i, num_iter, loss.item()))

        acc = validate(val_loader, model, args, epoch)
        if dist_mod.env_world_size() > 1:
            acc_t = torch.tensor(acc, device=device)
            acc = float(dist_mod.reduce_tensor(acc_t).item())
        if dist_mod.env_rank() == 0:
            print('Epoch {} val acc {:.2f}'.format(epoch, acc))
            is_best = acc > best_acc
            best_acc = max(acc, best_acc)
            save_checkpoint({'epoch': epoch + 1, 'arch': args.arch,
                             'state_dict': model.state_dict(),
                             'best_acc': best_acc,
                             'optimizer': optimizer.state_dict()},
                            os.path.join('results', args.tag + args.arch,
                                         'checkpoint.pth.tar'))
    return best_acc


def main(argv=None):
    args = build_main_parser().parse_args(argv)
    if args.seed is not None:
        random.seed(args.seed)
        np.random.seed(args.seed)
        torch.manual_seed(args.seed)
    if args.gpu is not None:
        os.environ['CUDA_VISIBLE_DEVICES'] = args.gpu

    model, dp, device, distributed = build_model(args)
    train_loader, val_loader = setup_data(args, device)

    params = [p for p in model.parameters() if p.requires_grad]
    optimizer = native_optim.SGD(params, lr=args.lr, momentum=args.momentum,
                                 weight_decay=args.weight_decay, nesterov=False)

    # --var_name eval sweep (reference main.py:1011-1087; gated dead there,
    # functional here): re-validate a restored model over a hyperparam grid.
    if args.var_name and args.resume:
        grids = {
            'pctl': [99.99, 99.992, 99.994, 99.996, 99.998, 99.999, 99.9995],
            'q_scale': [0.87, 0.88, 0.89, 0.90, 0.91, 0.92, 0.93, 0.94,
                        0.95, 0.96],
            'selected_weights': [1, 2, 5, 10],
        }
        var_list = grids.get(args.var_name, [getattr(args, args.var_name)])
        load_from_checkpoint(args, model, optimizer, device)
        total_list = []
        for var in var_list:
            setattr(args, args.var_name, var)
            accs = [validate(val_loader, model, args)
                    for _ in range(args.num_sims)]
            total_list.append((float(np.mean(accs)), float(np.min(accs)),
                               float(np.max(accs))))
            print('{:d} runs:  {} {} {:.2f} ({:.2f}/{:.2f})'.format(
                args.num_sims, args.var_name, var, *total_list[-1]))
        for var, (mean_, min_, max_) in zip(var_list, total_list):
            print('{} {} acc {:.2f} ({:.2f}/{:.2f})'.format(
                args.var_name, var, mean_, min_, max_))
        return total_list

    start_epoch, best_acc = 0, 0.0
    if args.resume:
        start_epoch, best_acc = load_from_checkpoint(args, model, optimizer,
                                                     device)
        if args.w_max > 0:
            with torch.no_grad():
                for n_, p in model.named_parameters():
                    if ('conv' in n_ or 'fc' in n_) and 'weight' in n_:
                        p.data.clamp_(-args.w_max, args.w_max)
        if args.merge_bn:
            merge_batchnorm(model, args)
        if args.distort_w_test:
            noise_levels = [0, 0.05, 0.1, 0.2, 0.3, 0.4, 0.5]
            test_distortion(model, args, val_loader=val_loader, mode='weights',
                            vars=noise_levels)
            return
        if args.evaluate:
            acc = validate(val_loader, model, args)
            print('Restored Model Accuracy (epoch {}): {:.2f}'.format(
                start_epoch, acc))
            return acc

    best = train(model, dp, args, train_loader, val_loader, optimizer, device,
                 start_epoch)
    if distributed:
        torch.distributed.destroy_process_group()
    return best
