"""CIFAR-10 training driver (the `python noisynet.py ...` entrypoint).

Reproduces the reference noisynet.py script flow: the
current x var_list x num_sims sweep loops, per-layer param groups, the
manual/step/exp/triangle LR schedules, L1/L2-family penalties, the
L3/L3_new/L3_act/L4 gradient penalties (double backward), trainable
w_max/act_max, post-step weight clamping, calibration stop at (epoch 0,
batch 5), per-epoch eval, best-checkpoint save/delete (state_dict with the
accuracy-encoded filename), early stopping and the results txt dump.

MI355X specifics: device-resident dataset + on-GPU augmentation, bf16
compute via --bf16, fused optimizer kernels (noisynet_amd.optim), and
channels-last layout on GPU for the MFMA conv kernels.
"""

import os
import random
from datetime import datetime

import numpy as np
import torch
from torch import nn
from torch.optim import lr_scheduler

from .. import data as data_mod
from .. import ops
from .. import optim as native_optim
from .. import utils
from ..config import broadcast_per_layer, build_noisynet_parser, var_list_for
from ..harness import merge_batchnorm, test_distortion
from ..models.noisynet import Net
from ..quant import QuantMeasure, finish_calibration, start_calibration


def make_optimizer(model, args):
    param_groups = [
        {'params': model.conv1.parameters(), 'weight_decay': args.L2_1,
         'lr': args.LR_1, 'clamp': ((-args.w_max1, args.w_max1)
                                    if args.w_max1 > 0 and not args.train_w_max else None)},
        {'params': model.conv2.parameters(), 'weight_decay': args.L2_2,
         'lr': args.LR_2, 'clamp': ((-args.w_max2, args.w_max2) if args.w_max2 > 0 else None)},
        {'params': model.linear1.parameters(), 'weight_decay': args.L2_3,
         'lr': args.LR_3, 'clamp': ((-args.w_max3, args.w_max3) if args.w_max3 > 0 else None)},
        {'params': model.linear2.parameters(), 'weight_decay': args.L2_4,
         'lr': args.LR_4, 'clamp': ((-args.w_max4, args.w_max4) if args.w_max4 > 0 else None)},
    ]
    if args.train_act_max:
        param_groups += [
            {'params': [model.act_max1], 'weight_decay': 0, 'lr': args.LR_act_max},
            {'params': [model.act_max2], 'weight_decay': 0, 'lr': args.LR_act_max},
            {'params': [model.act_max3], 'weight_decay': 0, 'lr': args.LR_act_max}]
    if args.train_w_max:
        param_groups += [
            {'params': [model.w_min1], 'weight_decay': 0, 'lr': args.LR_w_max},
            {'params': [model.w_max1], 'weight_decay': 0, 'lr': args.LR_w_max}]
    if args.batchnorm:
        param_groups += [
            {'params': model.bn1.parameters(), 'weight_decay': args.L2_bn},
            {'params': model.bn2.parameters(), 'weight_decay': args.L2_bn}]
        if args.bn3:
            param_groups += [{'params': model.bn3.parameters(), 'weight_decay': args.L2_bn}]
        if args.bn4:
            param_groups += [{'params': model.bn4.parameters(), 'weight_decay': args.L2_bn}]

    if args.optim == 'SGD':
        return native_optim.SGD(param_groups, lr=args.LR, momentum=args.momentum,
                                nesterov=args.nesterov)
    if args.optim == 'Adam':
        return native_optim.Adam(param_groups, lr=args.LR)
    return native_optim.AdamW(param_groups, lr=args.LR)


def apply_regularizers(model, args, loss, epoch, i):
    """L2_act* / L1_* / L2_act_max / L2_w_max / L2_bn penalties
    (noisynet.py:1298-1344)."""
    if args.L2_act1 > 0:
        loss = loss + args.L2_act1 * model.conv1_.pow(2).sum()
    if args.L2_act2 > 0:
        loss = loss + args.L2_act2 * model.conv2_.pow(2).sum()
    if args.L2_act3 > 0:
        loss = loss + args.L2_act3 * model.linear1_.pow(2).sum()
    if args.L2_act4 > 0:
        loss = loss + args.L2_act4 * model.linear2_.pow(2).sum()
    if args.L1_1 > 0:
        loss = loss + args.L1_1 * model.conv1.weight.norm(p=1)
    if args.L1_2 > 0:
        loss = loss + args.L1_2 * model.conv2.weight.norm(p=1)
    if args.L1_3 > 0:
        loss = loss + args.L1_3 * model.linear1.weight.norm(p=1)
    if args.L1_4 > 0:
        loss = loss + args.L1_4 * model.linear2.weight.norm(p=1)
    if args.train_act_max and args.L2_act_max > 0:
        if args.current1 == 0:
            loss = loss + args.L2_act_max * (model.act_max1 ** 2 + model.act_max2 ** 2
                                             + model.act_max3 ** 2)
        else:
            loss = loss + args.L2_act_max * ((model.act_max1 ** 2) / args.current2
                                             + (model.act_max2 ** 2) / args.current3
                                             + (model.act_max3 ** 2) / args.current4)
    if args.train_w_max and args.L2_w_max > 0:
        loss = loss + args.L2_w_max * (model.w_min1 ** 2 + model.w_max1 ** 2)
    if args.batchnorm:
        if args.L2_bn_weight > 0:
            loss = loss + args.L2_bn_weight * (torch.sum(model.bn1.weight ** 2)
                                               + torch.sum(model.bn2.weight ** 2))
        if args.L2_bn_bias > 0:
            loss = loss + args.L2_bn_bias * (torch.sum(model.bn1.bias ** 2)
                                             + torch.sum(model.bn2.bias ** 2))
    return loss


def gradient_penalties(model, args, loss):
    """L3_new / L3 / L3_act / L4 double-backward penalties
    (noisynet.py:1348-1476). Returns (loss, needs_retain_graph)."""
    params = [model.conv1.weight, model.conv2.weight,
              model.linear1.weight, model.linear2.weight]

    if args.L3_new > 0:
        param_grads = torch.autograd.grad(loss, params, create_graph=True,
                                          only_inputs=True)
        grad_norm = 0
        for grad in param_grads:
            if args.L3_L2:
                grad_norm = grad_norm + args.L3_new * grad.pow(2).sum()
            elif args.L3_L1:
                grad_norm = grad_norm + args.L3_new * grad.norm(p=1)
        loss = loss + grad_norm

    retain = args.L3 > 0 or args.L4 > 0 or args.print_stats or args.L3_act > 0
    return loss, retain


def post_backward_penalties(model, args, loss):
    """Penalties applied AFTER the main backward (accumulate into .grad)."""
    params = [model.conv1.weight, model.conv2.weight,
              model.linear1.weight, model.linear2.weight]

    if args.L3_act > 0:
        acts = [model.conv1_, model.conv2_, model.linear1_, model.linear2_]
        acts_grad = torch.autograd.grad(loss, acts, create_graph=True)
        act_grad_norm = args.L3_act * torch.stack(
            [g.pow(2).sum() for g in acts_grad]).sum()
        act_grad_norm.backward(retain_graph=args.L3 > 0 or args.L4 > 0)

    if args.L3 > 0:
        param_grads = torch.autograd.grad(loss, params, create_graph=True,
                                          only_inputs=True)
        grad_sum = 0
        grad_norm = 0
        for grad in param_grads:
            if args.L4 > 0:
                grad_sum = grad_sum + grad.pow(2).sum()
            grad_norm = grad_norm + args.L3 * grad.pow(2).sum()
        grad_norm.backward(retain_graph=args.L4 > 0)
        if args.L4 > 0:
            grads2 = torch.autograd.grad(grad_sum, params, create_graph=False)
            g2_norm = 0
            for g2 in grads2:
                g2_norm = g2_norm + g2.pow(2).sum()
            (args.L4 * g2_norm).backward(retain_graph=True)
    elif args.L4 > 0:
        grads = torch.autograd.grad(loss, params, create_graph=True)
        grads_sum = 0
        for g in grads:
            grads_sum = grads_sum + g.pow(2).sum()
        grads2 = torch.autograd.grad(grads_sum, params, create_graph=True)
        g2_norm = 0
        for g2 in grads2:
            g2_norm = g2_norm + g2.pow(2).sum()
        (args.L4 * g2_norm).backward(retain_graph=False)


def evaluate(model, args, test_inputs, test_labels, epoch=0):
    model.eval()
    te_accuracies = []
    num_test_batches = max(1, len(test_inputs) // args.batch_size)
    with torch.no_grad():
        for i in range(num_test_batches):
            input = test_inputs[i * args.batch_size:(i + 1) * args.batch_size]
            label = test_labels[i * args.batch_size:(i + 1) * args.batch_size]
            output = model(input, epoch, i)
            pred = output.data.max(1)[1]
            te_accuracies.append(pred.eq(label.data).float().mean().item() * 100.0)
    return float(np.mean(te_accuracies, dtype=np.float64))


def train_one(args, model, train_inputs, train_labels, test_inputs, test_labels,
              device, s=0, max_batches=None):
    """One simulation: full training run. Returns (best_acc, best_epoch)."""
    num_train_batches = len(train_labels) // args.batch_size
    if max_batches:
        num_train_batches = min(num_train_batches, max_batches)

    optimizer = make_optimizer(model, args)

    scheduler = None
    lr = args.LR
    if args.LR_scheduler == 'step':
        scheduler = lr_scheduler.StepLR(optimizer, args.LR_step_after, gamma=args.LR_step)
    elif args.LR_scheduler == 'exp':
        scheduler = lr_scheduler.ExponentialLR(optimizer, gamma=args.LR_decay)
    elif args.LR_scheduler == 'triangle':
        lr_increment = args.LR / ((args.LR_max_epoch + 1) * num_train_batches)
        mom_decrement = args.momentum / ((args.LR_max_epoch + 1) * num_train_batches)
        lr_decrement = (args.LR - 0.05 * args.LR) / (
            (args.nepochs - args.LR_max_epoch - args.LR_finetune_epochs) * num_train_batches)
        lr_decrement2 = (0.05 * args.LR) / (args.LR_finetune_epochs * num_train_batches)
        mom_increment = lr_decrement
        mom_increment2 = lr_decrement2
        lr = 0
        mom = args.momentum

    if args.q_a > 0 and args.calculate_running:
        start_calibration(model)

    best_accuracy, best_epoch, prev_best_acc = 0.0, 0, 15
    saved, saved_accuracy = False, 0.0
    create_dir = True
    # Fused softmax-xent HIP kernel on the hot path; the gradient-penalty
    # configs (L3/L4 double backward) need a twice-differentiable loss, so
    # they keep the eager composition.
    needs_double_backward = (args.L3 > 0 or args.L3_new > 0
                             or args.L3_act > 0 or args.L4 > 0
                             or args.print_stats)
    criterion = nn.CrossEntropyLoss() if needs_double_backward \
        else (lambda out, tgt: ops.cross_entropy(out, tgt))

    for epoch in range(args.nepochs):
        model.power = [[] for _ in range(args.num_layers)]
        model.nsr = [[] for _ in range(args.num_layers)]
        model.input_sparsity = [[] for _ in range(args.num_layers)]
        model.train()
        tr_accuracies = []

        if args.LR_scheduler == 'manual':
            lr = args.LR * args.LR_step ** (epoch // args.LR_step_after)
            for pg in optimizer.param_groups:
                pg['lr'] = lr
        elif args.LR_scheduler not in ('triangle',) and scheduler is not None:
            scheduler.step()
            lr = scheduler.get_last_lr()[0]

        rnd_idx = torch.randperm(len(train_inputs), device=train_inputs.device)
        train_inputs = train_inputs[rnd_idx]
        train_labels = train_labels[rnd_idx]

        for i in range(num_train_batches):
            if args.q_a > 0 and args.calculate_running and epoch == 0 and i == 5:
                finish_calibration(model, device)

            input = train_inputs[i * args.batch_size:(i + 1) * args.batch_size]
            label = train_labels[i * args.batch_size:(i + 1) * args.batch_size]

            if args.augment:
                input = data_mod.gpu_augment(input)

            output = model(input, epoch, i, s)
            loss = criterion(output, label)

            if args.LR_scheduler == 'triangle':
                if epoch <= args.LR_max_epoch:
                    lr += lr_increment
                    mom -= mom_decrement
                elif epoch <= args.nepochs - args.LR_finetune_epochs:
                    lr -= lr_decrement
                    mom += mom_increment
                else:
                    lr -= lr_decrement2
                    mom += mom_increment2
                for pg in optimizer.param_groups:
                    pg['lr'] = lr / args.batch_size
                    pg['momentum'] = mom

            loss = apply_regularizers(model, args, loss, epoch, i)
            optimizer.zero_grad(set_to_none=False)
            loss, retain = gradient_penalties(model, args, loss)
            loss.backward(retain_graph=retain)
            if retain:
                post_backward_penalties(model, args, loss)

            if args.grad_clip > 0:
                for n, p in model.named_parameters():
                    if p.grad is not None:
                        p.grad.data.clamp_(-args.grad_clip, args.grad_clip)

            if args.train_w_max:
                w_max1_grad = torch.sum(
                    model.conv1.weight.grad[model.conv1.weight >= model.w_max1])
                w_min1_grad = torch.sum(
                    model.conv1.weight.grad[model.conv1.weight <= model.w_min1])
                model.w_min1.data = model.w_min1.data - args.LR_w_max * w_min1_grad
                model.w_max1.data = model.w_max1.data - args.LR_w_max * w_max1_grad
                if args.L2_w_max > 0 and model.w_max1.grad is not None:
                    model.w_min1.data = model.w_min1.data - args.LR_w_max * model.w_min1.grad.data
                    model.w_max1.data = model.w_max1.data - args.LR_w_max * model.w_max1.grad.data
                    model.w_max1.grad.data.zero_()
                    model.w_min1.grad.data.zero_()

            optimizer.step()

            # post-step weight clamp for cases the fused clamp doesn't cover
            if args.w_max1 > 0 and args.train_w_max:
                model.conv1.weight.data = torch.where(
                    model.conv1.weight > model.w_max1, model.w_max1, model.conv1.weight)
                model.conv1.weight.data = torch.where(
                    model.conv1.weight < model.w_min1, model.w_min1, model.conv1.weight)

            pred = output.data.max(1)[1]
            tr_accuracies.append(pred.eq(label.data).float().mean().item() * 100.0)

        tr_acc = float(np.mean(tr_accuracies, dtype=np.float64))
        te_acc = evaluate(model, args, test_inputs, test_labels, epoch)

        print('{}\tEpoch {:>3d}  Train {:.2f}  Test {:.2f}  LR {:.4f}'.format(
            str(datetime.now())[:-7], epoch, tr_acc, te_acc, lr))

        if te_acc > best_accuracy:
            if saved:
                old = os.path.join(args.checkpoint_dir,
                                   'model_epoch_{:d}_acc_{:.2f}.pth'.format(best_epoch, saved_accuracy))
                if os.path.exists(old):
                    os.remove(old)
            if epoch > 10:
                if create_dir:
                    utils.saveargs(args)
                    create_dir = False
                if s == 0:
                    saved_accuracy = te_acc
                    torch.save(model.state_dict(), os.path.join(
                        args.checkpoint_dir,
                        'model_epoch_{:d}_acc_{:.2f}.pth'.format(epoch, te_acc)))
                    saved = True
            best_accuracy = te_acc
            best_epoch = epoch

        if epoch != 0 and epoch % args.early_stop_after == 0:
            if best_accuracy <= prev_best_acc:
                break
            prev_best_acc = best_accuracy

    return best_accuracy, best_epoch


def restore_model(args, device):
    """Name-matched partial state_dict restore (noisynet.py:979-1002)."""
    print('\nLoading model from checkpoint {}\n'.format(args.resume))
    args.checkpoint_dir = '/'.join(args.resume.split('/')[:-1]) + '/'
    model = Net(args=args).to(device)
    saved_model = torch.load(args.resume, map_location=device, weights_only=False)
    own_state = model.state_dict()
    for saved_name, saved_param in saved_model.items():
        for name, param in model.named_parameters():
            if name == saved_name:
                param.data = saved_param.data
        if 'running_min' in saved_name or 'running_max' in saved_name:
            continue
        elif 'running' in saved_name and args.track_running_stats:
            if saved_name in own_state:
                own_state[saved_name].copy_(saved_param)
    if args.w_max > 0:
        for n, p in model.named_parameters():
            if ('conv' in n or 'fc' in n or 'linear' in n) and 'weight' in n:
                p.data.clamp_(-args.w_max, args.w_max)
    if args.merge_bn:
        merge_batchnorm(model, args)
    return model


def main(argv=None):
    parser = build_noisynet_parser()
    args = parser.parse_args(argv)

    if args.seed is not None:
        random.seed(args.seed)
        np.random.seed(args.seed)
        torch.manual_seed(args.seed)
    if args.gpu is not None:
        os.environ['CUDA_VISIBLE_DEVICES'] = args.gpu

    device = 'cuda' if torch.cuda.is_available() else 'cpu'
    np.set_printoptions(precision=4, linewidth=120, suppress=True)

    train_inputs, train_labels, test_inputs, test_labels = \
        data_mod.load_cifar(args, device)

    if args.var_name == 'current':
        current_vars = [1, 3, 5, 10, 20, 50, 100]
    else:
        current_vars = [args.current]

    currents = {}
    for current in current_vars:
        print('\n****************** Current {} ********************\n'.format(current))
        currents[current] = []
        args.current = current
        results = {}

        var_list = var_list_for(args.var_name, current) if args.var_name else [' ']
        for var in var_list:
            if args.var_name:
                print('\n********** Setting {} to {} **********\n'.format(args.var_name, var))
                setattr(args, args.var_name, var)
                if args.var_name == 'LR':
                    args.LR_1 = args.LR_2 = args.LR_3 = args.LR_4 = args.LR
            broadcast_per_layer(args)

            tag = (args.tag + args.var_name + '-' + str(var) + '_') if args.var_name else args.tag
            args.checkpoint_dir = os.path.join(
                'results/', tag + 'current-' + '-'.join(
                    str(c) for c in args.layer_currents)
                + '_L3-' + str(args.L3) + '_L2-' + str(args.L2_1)
                + '_actmax-' + str(args.act_max1)
                + '_w_max1-' + str(args.w_max1) + '_bn-' + str(args.batchnorm)
                + '_LR-' + str(args.LR) + '_'
                + datetime.now().strftime('%Y-%m-%d_%H-%M-%S/'))

            best_accuracies = []
            for s in range(args.num_sims):
                if args.resume is None:
                    model = Net(args=args)
                    utils.init_model(model, args, s)
                    model = model.to(device)
                    if args.fp16:
                        model = model.half()
                    if args.bf16:
                        model = model.bfloat16()
                        if args.keep_bn_fp32:
                            for layer in model.modules():
                                if isinstance(layer, (nn.BatchNorm2d, nn.BatchNorm1d)):
                                    layer.float()
                else:
                    model = restore_model(args, device)
                    te_acc = evaluate(model, args, test_inputs, test_labels)
                    print('\nRestored Model Accuracy: {:.2f}\n'.format(te_acc))
                    if args.distort_w_test:
                        noise_levels = [0, 0.05, 0.1, 0.2, 0.3, 0.4, 0.5]
                        test_distortion(model, args,
                                        val_loader=(test_inputs, test_labels),
                                        mode='weights', vars=noise_levels)
                    best_accuracies.append(te_acc)
                    continue

                if s == 0:
                    utils.print_model(model, args, full=args.debug)

                best_acc, best_epoch = train_one(
                    args, model, train_inputs, train_labels, test_inputs,
                    test_labels, device, s)
                print('\nSimulation {:d}  {} {}  Best Accuracy: {:.2f} (epoch {})\n'.format(
                    s, args.tag + args.var_name, var, best_acc, best_epoch))
                best_accuracies.append(best_acc)

            results[var] = best_accuracies
            if best_accuracies:
                fmt = '{} {:<8}  {} mean {:>4.2f}  max {:>4.2f}  min {:>4.2f}'.format(
                    args.var_name, str(var),
                    [float('{:.2f}'.format(x)) for x in best_accuracies],
                    np.mean(best_accuracies), np.max(best_accuracies),
                    np.min(best_accuracies))
                print(fmt)
                currents[current].append(fmt)

            if not os.path.isdir(args.checkpoint_dir):
                utils.saveargs(args)
            output_file = os.path.join(
                args.checkpoint_dir,
                'results_current_{}_{}.txt'.format(args.current, args.var_name))
            with open(output_file, 'w') as f:
                for cur in currents:
                    f.write('\nCurrent {}nA\n'.format(cur))
                    for res in currents[cur]:
                        f.write(res + '\n')

    return currents
