"""Integration tests: Net forward/backward, driver plumbing, checkpoints,
gradient penalties (double backward), distortion harness."""

import os

import numpy as np
import pytest
import torch

from noisynet_amd import utils
from noisynet_amd.config import broadcast_per_layer, build_noisynet_parser
from noisynet_amd.drivers import cifar
from noisynet_amd.harness import merge_batchnorm
from noisynet_amd.harness import test_distortion as run_distortion
from noisynet_amd.models.noisynet import Net


def make_args(extra=None):
    argv = ['--n_train', '256', '--n_test', '128', '--batch_size', '32',
            '--nepochs', '1']
    if extra:
        argv += extra
    args = build_noisynet_parser().parse_args(argv)
    broadcast_per_layer(args)
    return args


def test_flag_parse_reference_commands():
    """The two README flagship commands must parse (README.md:6-13)."""
    p = build_noisynet_parser()
    a1 = p.parse_args(['--current', '1', '--act_max', '5', '--w_max1', '0.3',
                       '--LR', '0.005', '--L2_1', '0.0005', '--L2_2', '0.0002'])
    assert a1.current == 1 and a1.act_max == 5 and a1.w_max1 == 0.3
    a2 = p.parse_args(['--L2', '0.0005', '--dropout', '0.1', '--nepochs', '450'])
    assert a2.L2 == 0.0005 and a2.dropout == 0.1 and a2.nepochs == 450
    # --no-X forms
    a3 = p.parse_args(['--no-batchnorm', '--no-merged_dac', '--no-augment'])
    assert not a3.batchnorm and not a3.merged_dac and not a3.augment


def test_net_noisefree_forward_shapes():
    args = make_args()
    m = Net(args)
    x = torch.rand(4, 3, 32, 32)
    out = m(x, 0, 0)
    assert out.shape == (4, 10)


def test_net_noisy_quantized_backward():
    args = make_args(['--current', '1', '--act_max', '5', '--q_a', '4',
                      '--q_w', '4', '--w_max1', '0.3'])
    m = Net(args)
    m.train()
    x = torch.rand(4, 3, 32, 32)
    y = torch.randint(0, 10, (4,))
    out = m(x, 0, 0)
    loss = torch.nn.CrossEntropyLoss()(out, y)
    loss.backward()
    for name in ('conv1', 'conv2', 'linear1', 'linear2'):
        g = getattr(m, name).weight.grad
        assert g is not None and torch.isfinite(g).all(), name


def test_telemetry_recorded_first_batches():
    args = make_args(['--current', '1'])
    m = Net(args)
    m.train()
    x = torch.rand(4, 3, 32, 32)
    m(x, 0, 0)
    assert len(m.power[0]) == 1
    assert len(m.nsr[0]) == 1
    assert len(m.input_sparsity[0]) == 1
    m(x, 0, 25)  # i >= 20: no telemetry
    assert len(m.power[0]) == 1


def test_l3_gradient_penalty_double_backward():
    args = make_args(['--L3', '0.01'])
    m = Net(args)
    m.train()
    x = torch.rand(4, 3, 32, 32)
    y = torch.randint(0, 10, (4,))
    out = m(x, 0, 0)
    loss = torch.nn.CrossEntropyLoss()(out, y)
    loss, retain = cifar.gradient_penalties(m, args, loss)
    loss.backward(retain_graph=retain)
    cifar.post_backward_penalties(m, args, loss)
    assert torch.isfinite(m.conv1.weight.grad).all()


def test_l3_new_penalty():
    args = make_args(['--L3_new', '0.01', '--L3_L2'])
    m = Net(args)
    m.train()
    x = torch.rand(4, 3, 32, 32)
    y = torch.randint(0, 10, (4,))
    out = m(x, 0, 0)
    base = torch.nn.CrossEntropyLoss()(out, y)
    loss, _ = cifar.gradient_penalties(m, args, base)
    assert loss.item() > base.item()
    loss.backward()


def test_checkpoint_save_restore_roundtrip(tmp_path):
    args = make_args(['--q_a', '4'])
    m = Net(args)
    utils.init_model(m, args)
    path = tmp_path / 'model_epoch_5_acc_55.00.pth'
    torch.save(m.state_dict(), str(path))

    args.resume = str(path)
    m2 = cifar.restore_model(args, 'cpu')
    for (n1, p1), (n2, p2) in zip(m.named_parameters(), m2.named_parameters()):
        assert n1 == n2
        assert torch.equal(p1, p2), n1

    # restored model gives identical eval outputs
    m.eval()
    m2.eval()
    x = torch.rand(4, 3, 32, 32)
    assert torch.allclose(m(x), m2(x))


def test_checkpoint_tolerates_extra_and_missing_keys(tmp_path):
    """Name-matched partial copy (noisynet.py:987-1002)."""
    args = make_args()
    m = Net(args)
    sd = m.state_dict()
    sd['bogus_extra_key'] = torch.zeros(3)
    del sd['linear2.weight']
    path = tmp_path / 'model_epoch_1_acc_10.00.pth'
    torch.save(sd, str(path))
    args.resume = str(path)
    m2 = cifar.restore_model(args, 'cpu')  # must not raise
    assert torch.equal(m2.conv1.weight, m.conv1.weight)


def test_driver_end_to_end(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    out = cifar.main(['--nepochs', '2', '--n_train', '256', '--n_test', '128',
                      '--batch_size', '32', '--current', '1', '--act_max', '5',
                      '--q_a', '4', '--calculate_running', '--LR', '0.005'])
    assert out  # sweep results returned


def test_driver_noisefree_baseline(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    out = cifar.main(['--nepochs', '1', '--n_train', '128', '--n_test', '64',
                      '--batch_size', '32', '--L2', '0.0005', '--dropout', '0.1'])
    assert out


def test_merge_batchnorm_noisynet():
    args = make_args()
    args.arch = 'noisynet'
    m = Net(args)
    utils.init_model(m, args)
    # give BN non-trivial stats
    m.train()
    for _ in range(3):
        m(torch.rand(16, 3, 32, 32))
    w_before = m.conv1.weight.data.clone()
    merge_batchnorm(m, args)
    scale = m.bn1.weight.data.view(-1, 1, 1, 1) / torch.sqrt(
        m.bn1.running_var.data.view(-1, 1, 1, 1) + 1e-7)
    assert torch.allclose(m.conv1.weight.data, w_before * scale, atol=1e-6)


def test_merged_bn_eval_path_close_to_unmerged():
    """After BN folding, the merge_bn eval forward must match the unmerged
    eval forward (the reference restore-check mechanism)."""
    args = make_args()
    m = Net(args)
    utils.init_model(m, args)
    m.train()
    for _ in range(5):
        m(torch.rand(32, 3, 32, 32))
    m.eval()
    x = torch.rand(8, 3, 32, 32)
    out_ref = m(x)
    merge_batchnorm(m, args)
    args.merge_bn = True
    out_merged = m(x)
    assert torch.allclose(out_ref, out_merged, atol=1e-3, rtol=1e-3)


def test_distortion_harness():
    args = make_args(['--num_sims', '2'])
    args.stuck_at_weights = None
    args.test_temp = 0
    m = Net(args)
    utils.init_model(m, args)
    m.eval()
    inputs = torch.rand(64, 3, 32, 32)
    labels = torch.randint(0, 10, (64,))
    w_before = m.conv1.weight.data.clone()
    res = run_distortion(m, args, val_loader=(inputs, labels),
                          mode='weights', vars=[0.1, 0.5])
    # weights restored after sweep
    assert torch.equal(m.conv1.weight.data, w_before)


def test_stuck_at_faults():
    args = make_args()
    args.test_temp = 0
    m = Net(args)
    utils.init_model(m, args)
    m.eval()
    inputs = torch.rand(32, 3, 32, 32)
    labels = torch.randint(0, 10, (32,))
    for mode in ('random_zero', 'random_one', 'largest_zero', 'smallest_zero'):
        args.stuck_at_weights = mode
        run_distortion(m, args, val_loader=(inputs, labels), mode='weights',
                        vars=[0.2])
    args.stuck_at_weights = None


def test_train_w_max_and_act_max():
    args = make_args(['--train_act_max', '--train_w_max', '--act_max', '1',
                      '--w_max1', '0.3'])
    m = Net(args)
    with torch.no_grad():
        m.act_max1.fill_(1.0)
        m.act_max2.fill_(1.0)
        m.act_max3.fill_(1.0)
        m.w_max1.fill_(0.3)
        m.w_min1.fill_(-0.3)
    m.train()
    out = m(torch.rand(4, 3, 32, 32), 0, 0)
    loss = out.sum()
    loss.backward()
    assert m.act_max1.grad is not None


def test_temperature_and_scale_weights_modes():
    """The power-law temperature drift (main.py:430-446) and the
    scale_weights mode both perturb weights and restore after the sweep."""
    args = make_args(['--num_sims', '1'])
    args.stuck_at_weights = None
    m = Net(args)
    utils.init_model(m, args)
    m.eval()
    inputs = torch.rand(32, 3, 32, 32)
    labels = torch.randint(0, 10, (32,))

    inputs = torch.rand(args.batch_size * 2, 3, 32, 32)
    labels = torch.randint(0, 10, (args.batch_size * 2,))
    args.test_temp = 77
    args.temperature = 25
    w0 = m.conv1.weight.data.clone()
    run_distortion(m, args, val_loader=(inputs, labels), mode='weights',
                   vars=[0.0])
    assert torch.allclose(m.conv1.weight.data, w0)
    args.test_temp = 0

    args.scale_weights = 0.5
    run_distortion(m, args, val_loader=(inputs, labels), mode='weights',
                   vars=[0.0])
    assert torch.allclose(m.conv1.weight.data, w0)
    args.scale_weights = 0


def test_act_distortion_mode():
    """mode='acts' flips args.distort_act for the duration of the sweep."""
    args = make_args(['--num_sims', '1'])
    args.stuck_at_weights = None
    args.test_temp = 0
    m = Net(args)
    utils.init_model(m, args)
    m.eval()
    inputs = torch.rand(args.batch_size * 2, 3, 32, 32)
    labels = torch.randint(0, 10, (args.batch_size * 2,))
    res = run_distortion(m, args, val_loader=(inputs, labels), mode='acts',
                         vars=[0.1])
    assert isinstance(res, float) and res == res  # finite accuracy
    assert not args.distort_act


def test_selected_weights_reduced_distortion():
    """--selected_weights protects the top-K% weights: distortion of the
    selected set is reduced (main.py:351-377)."""
    args = make_args(['--num_sims', '1'])
    args.stuck_at_weights = None
    args.test_temp = 0
    args.selected_weights = 5
    args.selection_criteria = 'weight_magnitude'
    m = Net(args)
    utils.init_model(m, args)
    m.eval()
    inputs = torch.rand(args.batch_size * 2, 3, 32, 32)
    labels = torch.randint(0, 10, (args.batch_size * 2,))
    res = run_distortion(m, args, val_loader=(inputs, labels), mode='weights',
                         vars=[0.3])
    assert isinstance(res, float) and res == res
    args.selected_weights = 0
