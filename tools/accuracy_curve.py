#!/usr/bin/env python3
"""Accuracy-vs-current curve on the learnable synthetic CIFAR.

Trains the flagship NoisyNet config at several I_max currents (plus the
noise-free baseline) and reports held-out top-1 per epoch -- the
synthetic-data analogue of the reference README's ~88% clean / ~78% @1nA
table (README.md:6-13). Writes a markdown summary for profiles/.

Usage (GPU box):
    python tools/accuracy_curve.py --epochs 40 --out gpurun_out/accuracy_curve.md
"""

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from noisynet_amd import data as data_mod  # noqa: E402
from noisynet_amd import ops, utils  # noqa: E402
from noisynet_amd import optim as native_optim  # noqa: E402
from noisynet_amd.config import broadcast_per_layer, build_noisynet_parser  # noqa: E402
from noisynet_amd.models.noisynet import Net  # noqa: E402
from noisynet_amd.quant import finish_calibration, start_calibration  # noqa: E402


def train_one(current, epochs, X, y, Xt, yt, device, seed=11, batch=64):
    # exact README flagship flags (README.md:6-9); the reference's default
    # optimizer is AdamW (noisynet.py:249) -- forcing SGD here was measured
    # ~50 points worse at 1 nA
    argv = ['--q_a', '4', '--act_max', '5', '--w_max1', '0.3', '--LR',
            '0.005', '--L2_1', '0.0005', '--L2_2', '0.0002', '--batch_size',
            str(batch), '--calculate_running']
    if current > 0:
        argv = ['--current', str(current)] + argv
    args = build_noisynet_parser().parse_args(argv)
    broadcast_per_layer(args)
    torch.manual_seed(seed)
    model = Net(args)
    utils.init_model(model, args)
    model = model.to(device)
    if device.type == 'cuda':
        model = model.to(memory_format=torch.channels_last)
    groups = [
        {'params': model.conv1.parameters(), 'weight_decay': args.L2_1,
         'lr': args.LR, 'clamp': (-args.w_max1, args.w_max1)},
        {'params': model.conv2.parameters(), 'weight_decay': args.L2_2,
         'lr': args.LR},
        {'params': model.linear1.parameters(), 'weight_decay': 0.0,
         'lr': args.LR},
        {'params': model.linear2.parameters(), 'weight_decay': 0.0,
         'lr': args.LR},
        {'params': [p for m in (model.bn1, model.bn2, model.bn3, model.bn4)
                    for p in m.parameters()], 'weight_decay': 0.0,
         'lr': args.LR}]
    if args.optim == 'SGD':
        opt = native_optim.SGD(groups, lr=args.LR, momentum=args.momentum,
                               nesterov=True)
    else:  # AdamW: the reference's (and this driver's) default
        opt = native_optim.AdamW(groups, lr=args.LR)
    n = X.shape[0]
    start_calibration(model)
    curve = []
    step = 0
    for epoch in range(epochs):
        # step-decay schedule (the reference trains with LR decay,
        # noisynet.py:1176-1231; constant LR left late epochs unstable)
        if epoch in (int(epochs * 0.5), int(epochs * 0.75),
                     int(epochs * 0.9)):
            for group in opt.param_groups:
                group['lr'] *= 0.2
        model.train()
        perm = torch.randperm(n, device=device)
        for i in range(n // batch):
            idx = perm[i * batch:(i + 1) * batch]
            xb = data_mod.gpu_augment(X[idx])
            if step == 5:
                finish_calibration(model, device)
            out = model(xb, epoch, i)
            loss = ops.cross_entropy(out, y[idx])
            opt.zero_grad(set_to_none=False)
            loss.backward()
            opt.step()
            step += 1
        model.eval()
        correct = 0
        with torch.no_grad():
            for i in range(0, Xt.shape[0], 2000):
                out = model(Xt[i:i + 2000], epoch, 100)
                correct += int((out.argmax(1) == yt[i:i + 2000]).sum())
        acc = 100.0 * correct / Xt.shape[0]
        curve.append(acc)
        print('current=%s epoch=%d top1=%.2f loss=%.3f'
              % (current, epoch, acc, float(loss)), flush=True)
    return curve


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--epochs', type=int, default=40)
    p.add_argument('--currents', type=str, default='0,30,10,1',
                   help='comma-separated I_max nA values; 0 = noise-free')
    p.add_argument('--out', type=str, default='gpurun_out/accuracy_curve.md')
    a = p.parse_args()

    device = torch.device('cuda' if torch.cuda.is_available() else 'cpu')
    tr, trl, te, tel = data_mod.synthesize_cifar4bit(50000, 10000)
    X = torch.from_numpy(tr).to(device)
    X = torch.nn.functional.pad(X, (4, 4, 4, 4))
    y = torch.from_numpy(trl).to(device)
    Xt = torch.from_numpy(te).to(device)
    yt = torch.from_numpy(tel).to(device)
    if device.type == 'cuda':
        X = X.contiguous(memory_format=torch.channels_last)
        Xt = Xt.contiguous(memory_format=torch.channels_last)

    results = {}
    for cur in [float(c) for c in a.currents.split(',')]:
        results[cur] = train_one(cur, a.epochs, X, y, Xt, yt, device)

    os.makedirs(os.path.dirname(a.out) or '.', exist_ok=True)
    with open(a.out, 'w') as f:
        f.write('# Accuracy vs analog current (learnable synthetic CIFAR)\n\n')
        f.write('Flagship config (`--q_a 4 --act_max 5 --w_max1 0.3 --LR '
                '0.005 --L2_1 0.0005 --L2_2 0.0002`, batch 64, GPU crop/flip '
                'augment), %d epochs, 50k train / 10k test synthetic 4-bit '
                'CIFAR (data.py synthesize_cifar4bit). Reference semantics: '
                'README.md:6-13 (~88%% clean / ~78%% @1nA on real CIFAR).\n\n'
                % a.epochs)
        f.write('| I_max (nA) | best top-1 | final top-1 |\n|---|---|---|\n')
        for cur, curve in results.items():
            label = 'noise-free' if cur == 0 else ('%g' % cur)
            f.write('| %s | %.2f%% | %.2f%% |\n'
                    % (label, max(curve), curve[-1]))
        f.write('\n## Per-epoch curves\n\n')
        for cur, curve in results.items():
            f.write('- I_max=%s: %s\n'
                    % (('clean' if cur == 0 else '%g nA' % cur),
                       ' '.join('%.1f' % v for v in curve)))
    print('wrote', a.out)


if __name__ == '__main__':
    main()
