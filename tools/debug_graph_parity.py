#!/usr/bin/env python3
"""Bisect the eager-vs-graphed trajectory divergence.

Runs the deterministic flagship config three ways with identical seeds:
  A. eager, twice  -> is the eager path itself bit-reproducible?
  B. eager vs graphed -> per-step loss comparison, first divergent step.
"""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from noisynet_amd import ops, utils  # noqa: E402
from noisynet_amd import optim as native_optim  # noqa: E402
from noisynet_amd.config import broadcast_per_layer, build_noisynet_parser  # noqa: E402
from noisynet_amd.graphs import GraphedTrainStep  # noqa: E402
from noisynet_amd.models.noisynet import Net  # noqa: E402
from noisynet_amd.quant import finish_calibration, start_calibration  # noqa: E402

ARGV = ['--q_a', '4', '--act_max', '5', '--batch_size', '64',
        '--stochastic', '0', '--calculate_running', '--no-augment']
STEPS = 6


def build():
    args = build_noisynet_parser().parse_args(ARGV)
    broadcast_per_layer(args)
    torch.manual_seed(3)
    model = Net(args)
    utils.init_model(model, args)
    model = model.cuda().to(memory_format=torch.channels_last)
    opt = native_optim.SGD(model.parameters(), lr=0.01, momentum=0.9,
                           nesterov=True)
    return model, opt


def calibrate(model, x):
    start_calibration(model)
    with torch.no_grad():
        for i in range(6):
            model(x, 0, i)
    finish_calibration(model, torch.device('cuda'))


def data():
    torch.manual_seed(0)
    x = torch.rand(64, 3, 32, 32).cuda().contiguous(
        memory_format=torch.channels_last)
    y = torch.randint(0, 10, (64,)).cuda()
    return x, y


def run_eager(x, y, steps=STEPS):
    model, opt = build()
    calibrate(model, x)
    model.train()
    losses = []
    for i in range(steps):
        loss = ops.cross_entropy(model(x, 0, 1000), y)
        opt.zero_grad(set_to_none=False)
        loss.backward()
        opt.step()
        losses.append(float(loss))
    return losses, {n: p.detach().clone() for n, p in model.named_parameters()}


def run_graphed(x, y, steps=STEPS):
    model, opt = build()
    calibrate(model, x)
    model.train()
    sx, sy = x.clone(), y.clone()

    def body():
        loss = ops.cross_entropy(model(sx, 0, 1000), sy)
        opt.zero_grad(set_to_none=False)
        loss.backward()
        opt.step()
        return loss

    g = GraphedTrainStep(body, warmup=1)
    losses = [float('nan')]  # warmup step loss not tracked
    for _ in range(steps - 1):
        losses.append(float(g.replay()))
    g.close()
    return losses, {n: p.detach().clone() for n, p in model.named_parameters()}


def main():
    x, y = data()
    la1, pa1 = run_eager(x, y)
    la2, pa2 = run_eager(x, y)
    same = all(torch.equal(pa1[n], pa2[n]) for n in pa1)
    print("eager-vs-eager params identical:", same)
    print("eager run1 losses:", ["%.6f" % v for v in la1])
    print("eager run2 losses:", ["%.6f" % v for v in la2])

    lg, pg = run_graphed(x, y)
    print("graphed   losses:", ["%.6f" % v for v in lg])
    for n in pa1:
        d = (pa1[n].float() - pg[n].float()).abs().max().item()
        print("param %-20s max|eager-graph| = %.3e" % (n, d))


if __name__ == '__main__':
    main()
