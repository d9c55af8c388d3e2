"""MNIST MLP chip driver (the `python chip_mnist.py` entrypoint).

Rebuilds reference chip_mnist.py: the 784->390->10 MLP with input/activation
quantization (including the triple_input 4/3/2-bit concat trick), the L3
gradient penalty (double backward), magnitude pruning with separate pos/neg
thresholds at --prune_epoch, the L1_1/L1_2/L3/L2 sweep grids, and the
chip_plots .mat/.npy export.
"""

import argparse
import os

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import data as data_mod
from .. import ops
from .. import optim as native_optim
from .. import utils
from ..quant import QuantMeasure


class Net(nn.Module):
    def __init__(self, args):
        super().__init__()
        self.debug = args.debug
        self.q_a = args.q_a
        self.triple_input = args.triple_input
        self.batchnorm1 = args.bn1
        self.batchnorm2 = args.bn2

        input_size = 3 if self.triple_input else 1
        self.fc1 = nn.Linear(784 * input_size, 390, bias=args.use_bias)
        self.fc2 = nn.Linear(390, 10, bias=args.use_bias)
        self.quantize = QuantMeasure(args.q_a, stochastic=args.stochastic,
                                     max_value=1, debug=args.debug)
        if args.triple_input:
            self.quantize1 = QuantMeasure(4, stochastic=args.stochastic,
                                          max_value=1, debug=args.debug)
            self.quantize2 = QuantMeasure(3, stochastic=args.stochastic,
                                          max_value=1, debug=args.debug)
            self.quantize3 = QuantMeasure(2, stochastic=args.stochastic,
                                          max_value=1, debug=args.debug)
        if args.bn1:
            self.bn1 = nn.BatchNorm1d(390, track_running_stats=args.track_running_stats)
        if args.bn2:
            self.bn2 = nn.BatchNorm1d(10, track_running_stats=args.track_running_stats)

        self.drop_p_input = args.dropout_input
        self.drop_p_act = args.dropout_act

    def forward(self, x):
        self.input = x
        if self.q_a > 0:
            if self.triple_input:
                x = torch.cat([self.quantize1(x), self.quantize2(x),
                               self.quantize3(x)], dim=1)
            else:
                x = self.quantize(x)
            self.quantized_input = x
        if self.drop_p_input > 0:
            x = ops.dropout(x, self.drop_p_input, self.training)
        self.preact = ops.linear(x, self.fc1.weight, self.fc1.bias)
        x = F.relu(self.preact)
        if self.batchnorm1:
            x = self.bn1(x)
        self.act = x
        if self.drop_p_act > 0:
            x = ops.dropout(x, self.drop_p_act, self.training)
        self.output = ops.linear(x, self.fc2.weight, self.fc2.bias)
        if self.batchnorm2:
            self.output = self.bn2(self.output)
        if self.training:
            return F.log_softmax(self.output, dim=1)
        return self.output


def train(args, model, num_train_batches, images, labels, optimizer):
    model.train()
    correct = 0
    for i in range(num_train_batches):
        batch = images[i * args.batch_size:(i + 1) * args.batch_size]
        batch_labels = labels[i * args.batch_size:(i + 1) * args.batch_size]
        optimizer.zero_grad(set_to_none=False)
        output = model(batch)
        loss = F.nll_loss(output, batch_labels)
        if args.L3 > 0:
            param_grads = torch.autograd.grad(loss, list(model.parameters()),
                                              create_graph=True)
            grad_norm = sum(g.pow(2).sum() for g in param_grads)
            loss = loss + args.L3 * grad_norm
        if args.L1_1 > 0:
            loss = loss + args.L1_1 * model.fc1.weight.norm(p=1)
        if args.L1_2 > 0:
            loss = loss + args.L1_2 * model.fc2.weight.norm(p=1)
        loss.backward()
        optimizer.step()
        if args.w_max > 0:
            for n, p in model.named_parameters():
                if 'weight' in n:
                    p.data.clamp_(-args.w_max, args.w_max)
        pred = output.argmax(dim=1, keepdim=True)
        correct += pred.eq(batch_labels.view_as(pred)).sum().item()
    return 100. * correct / (num_train_batches * args.batch_size)


def test(model, images, labels):
    model.eval()
    with torch.no_grad():
        output = model(images)
        pred = output.argmax(dim=1, keepdim=True)
        correct = pred.eq(labels.view_as(pred)).sum().item()
    return 100. * correct / len(images)


def prune_weights(args, model):
    """Magnitude pruning with separate pos/neg thresholds
    (chip_mnist.py:132-157)."""
    sparsities = []
    with torch.no_grad():
        for n, p in model.named_parameters():
            prune_pct = args.prune_weights1 if 'fc1' in n else args.prune_weights2
            if prune_pct <= 0 or 'weight' not in n:
                sparsities.append(0.0)
                continue
            w = p.clone()
            w_pos = w.data[w.data >= 0]
            w_neg = w.data[w.data < 0]
            pos_thr, _ = torch.kthvalue(torch.abs(w_pos.view(-1)),
                                        max(1, int(w_pos.numel() * prune_pct / 100.0)))
            neg_thr, _ = torch.kthvalue(torch.abs(w_neg.view(-1)),
                                        max(1, int(w_neg.numel() * prune_pct / 100.0)))
            w_pos[w_pos < pos_thr] = 0
            w_neg[w_neg > -neg_thr] = 0
            p.data[w.data < 0] = w_neg
            p.data[w.data >= 0] = w_pos
            sparsity = p.data[torch.abs(p.data) < 0.01 * p.data.max()].numel() \
                / p.data.numel() * 100.0
            sparsities.append(sparsity)
    return sparsities


def build_parser():
    parser = argparse.ArgumentParser(
        description='NoisyNet-MI355X MNIST chip MLP',
        formatter_class=argparse.ArgumentDefaultsHelpFormatter)
    parser.add_argument('--dataset', type=str, default='data/mnist.npy')
    parser.add_argument('--batch-size', '--batch_size', type=int, default=100,
                        metavar='N', dest='batch_size')
    parser.add_argument('--epochs', type=int, default=101, metavar='N')
    parser.add_argument('--LR', type=float, default=0.01, metavar='LR')
    parser.add_argument('--L2', type=float, default=0.0001, metavar='L2')
    parser.add_argument('--L1_1', type=float, default=5e-4, metavar='L2')
    parser.add_argument('--L1_2', type=float, default=1e-5, metavar='L2')
    parser.add_argument('--L3', type=float, default=0.05, metavar='L3')
    parser.add_argument('--momentum', type=float, default=0.9, metavar='M')
    parser.add_argument('--seed', type=int, default=1, metavar='S')
    parser.add_argument('--use_bias', dest='use_bias', action='store_true')
    parser.add_argument('--q_a', type=int, default=4, metavar='S')
    parser.add_argument('--act_max', type=float, default=1.0)
    parser.add_argument('--w_max', type=float, default=0.)
    parser.add_argument('--stochastic', type=float, default=0.5)
    parser.add_argument('--debug', dest='debug', action='store_true')
    parser.add_argument('--calculate_running', dest='calculate_running',
                        action='store_true')
    parser.add_argument('--plot', dest='plot', action='store_true')
    parser.add_argument('--save', dest='save', action='store_true')
    parser.add_argument('--bn1', dest='bn1', action='store_true')
    parser.add_argument('--bn2', dest='bn2', action='store_true')
    parser.add_argument('--track_running_stats', dest='track_running_stats',
                        action='store_true')
    parser.add_argument('--augment', dest='augment', action='store_true')
    parser.add_argument('--triple_input', dest='triple_input', action='store_true')
    parser.add_argument('--dropout_input', type=float, default=0.2)
    parser.add_argument('--dropout_act', type=float, default=0.4)
    parser.add_argument('--prune_weights1', type=float, default=0.0)
    parser.add_argument('--prune_weights2', type=float, default=0.0)
    parser.add_argument('--prune_epoch', type=float, default=90)
    parser.add_argument('--var_name', type=str, default='')
    parser.add_argument('--gpu', type=str, default=None)
    parser.add_argument('--num_sims', type=int, default=1)
    parser.add_argument('--n_train', type=int, default=60000)
    parser.add_argument('--n_test', type=int, default=10000)
    return parser


def load_mnist(args, device):
    if args.dataset and os.path.exists(args.dataset):
        data = np.load(args.dataset, allow_pickle=True)
        (train_inputs, train_labels), (test_inputs, test_labels) = data
    else:
        train_inputs, train_labels, test_inputs, test_labels = \
            data_mod.synthesize_mnist(args.n_train, args.n_test)
        print('[data] %s not found -> synthetic MNIST-shaped data' % args.dataset)
    to = lambda a: torch.from_numpy(np.asarray(a)).to(device)
    return to(train_inputs).float(), to(train_labels).long(), \
        to(test_inputs).float(), to(test_labels).long()


def main(argv=None):
    args = build_parser().parse_args(argv)
    if args.gpu is not None:
        os.environ['CUDA_VISIBLE_DEVICES'] = args.gpu
    np.set_printoptions(precision=4, linewidth=200, suppress=True)
    device = 'cuda' if torch.cuda.is_available() else 'cpu'
    torch.manual_seed(args.seed)

    train_inputs, train_labels, test_inputs, test_labels = load_mnist(args, device)

    grids = {
        'L1_1': [0, 1e-6, 2e-6, 3e-6, 5e-6, 7e-6, 1e-5, 2e-5, 3e-5, 4e-5,
                 5e-5, 7e-5, 1e-4, 2e-4],
        'L1_2': [0, 1e-6, 2e-6, 3e-6, 5e-6, 7e-6, 1e-5, 2e-5, 3e-5, 4e-5,
                 5e-5, 7e-5, 1e-4, 2e-4],
        'L3': [0, 0.001, 0.002, 0.003, 0.005, 0.007, 0.01, 0.02, 0.03, 0.04,
               0.05, 0.06, 0.08, 0.1, 0.2],
        'L2': [0, 5e-6, 1e-5, 2e-5, 3e-5, 4e-5, 5e-5, 7e-5, 1e-4, 2e-4, 3e-4,
               4e-4, 5e-4, 0.001],
    }
    var_list = grids.get(args.var_name, [' '])

    total_list = []
    for var in var_list:
        if args.var_name:
            print('\n********** Setting {} to {} **********\n'.format(args.var_name, var))
            setattr(args, args.var_name, var)
        best_accs = []
        for s in range(args.num_sims):
            model = Net(args).to(device)
            optimizer = native_optim.SGD(model.parameters(), lr=args.LR,
                                         momentum=args.momentum,
                                         weight_decay=args.L2)
            num_train_batches = len(train_inputs) // args.batch_size
            best_acc = 0
            if s == 0:
                utils.print_model(model, args)
            for epoch in range(args.epochs):
                rnd_idx = torch.randperm(len(train_inputs), device=train_inputs.device)
                train_inputs = train_inputs[rnd_idx]
                train_labels = train_labels[rnd_idx]
                if epoch % 70 == 0 and epoch != 0:
                    for pg in optimizer.param_groups:
                        pg['lr'] = pg['lr'] / 10.
                train_acc = train(args, model, num_train_batches, train_inputs,
                                  train_labels, optimizer)
                val_acc = test(model, test_inputs, test_labels)
                if ((args.prune_weights1 > 0 or args.prune_weights2 > 0)
                        and epoch % args.prune_epoch == 0 and epoch != 0):
                    print('Accuracy before pruning: {:.2f}'.format(val_acc))
                    prune_weights(args, model)
                    val_acc = test(model, test_inputs, test_labels)
                    print('Accuracy after pruning: {:.2f}'.format(val_acc))
                print('Epoch {:>2d} train acc {:>.2f} test acc {:>.2f}  LR {:.4f}'.format(
                    epoch, train_acc, val_acc, optimizer.param_groups[0]['lr']))
                if val_acc > best_acc:
                    best_acc = val_acc
                    if epoch > 80 and args.save:
                        os.makedirs('chip_plots', exist_ok=True)
                        tensors = {
                            'fc1_weights': model.fc1.weight,
                            'preact': model.preact,
                            'act': model.act,
                            'fc2_weights': model.fc2.weight,
                            'output': model.output,
                        }
                        try:
                            import scipy.io
                            scipy.io.savemat('chip_plots/mlp.mat', mdict={
                                k: v.detach().cpu().half().numpy()
                                for k, v in tensors.items()})
                        except ImportError:
                            np.save('chip_plots/mlp.npy',
                                    {k: v.detach().cpu().numpy()
                                     for k, v in tensors.items()})
            print('Simulation {:d}  Best Accuracy: {:.2f}'.format(s, best_acc))
            best_accs.append(best_acc)
        total_list.append((np.mean(best_accs), np.min(best_accs), np.max(best_accs)))
        print('{:d} runs:  {} {} {:.2f} ({:.2f}/{:.2f})'.format(
            args.num_sims, args.var_name, var, *total_list[-1]))
    for var, (mean_, min_, max_) in zip(var_list, total_list):
        print('{} {:>5} acc {:.2f} ({:.2f}/{:.2f})'.format(
            args.var_name, var, mean_, min_, max_))
    return total_list
