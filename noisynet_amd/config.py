"""Flag-compatible CLI builders (table-driven).

The reference drivers use ~100 argparse flags each, including the repetitive
``--X`` / ``--no-X`` mutually-exclusive-group idiom and per-layer suffixed
flags with global->per-layer broadcast (noisynet.py:20-311, :861-903;
main.py:40-192). These tables reproduce every flag and default.
"""

import argparse


def _bool_pair(parser, name, default):
    grp = parser.add_mutually_exclusive_group(required=False)
    grp.add_argument('--' + name, dest=name, action='store_true')
    grp.add_argument('--no-' + name, dest=name, action='store_false')
    parser.set_defaults(**{name: default})


# (name, default) -- noisynet.py bool-pair flags in reference order
_NOISYNET_BOOL_PAIRS = [
    ('generate_input', False), ('use_bias', False), ('fp16', False),
    ('keep_bn_fp32', False), ('augment', True), ('normalize', False),
    ('whiten_cifar10', False), ('train_act_max', False), ('train_w_max', False),
    ('batchnorm', True), ('bn3', True), ('bn4', True), ('biprecision', False),
    ('amsgrad', False), ('debug', False), ('nesterov', True), ('split', False),
    ('debug_quant', False), ('distort_w_test', False), ('write', False),
    ('plot', False), ('plot_basic', False), ('plot_noise', False),
    ('plot_power', False), ('weightnorm', False), ('print_clip', False),
    ('track_running_stats', True), ('noise_test', False), ('merged_dac', True),
    ('merge_bn', False), ('blocked', False), ('print_stats', False),
    ('calculate_running', False), ('debug_noise', False),
]

_NOISYNET_FLOATS = {
    'current': 0.0, 'current1': 0.0, 'current2': 0.0, 'current3': 0.0,
    'current4': 0.0, 'noise': 0.0, 'train_current': 0.0, 'test_current': 0.0,
    'act_max': 0.0, 'act_max1': 0.0, 'act_max2': 0.0, 'act_max3': 0.0,
    'w_min1': 0.0, 'w_max': 0.0, 'w_max1': 0.0, 'w_max2': 0.0, 'w_max3': 0.0,
    'w_max4': 0.0, 'grad_clip': 0.0, 'dropout': 0.0, 'dropout_conv': 0.0,
    'LR_act_max': 0.001, 'LR_w_max': 0.001, 'LR_1': 0.0, 'LR_2': 0.0,
    'LR_3': 0.0, 'LR_4': 0.0, 'LR': 0.001, 'LR_decay': 0.95, 'LR_step': 0.1,
    'momentum': 0.9, 'L1_1': 0.0, 'L1_2': 0.0, 'L1_3': 0.0, 'L1_4': 0.0,
    'L1': 0.0, 'L2_w_max': 0.0, 'L2_act_max': 0.0, 'L2_bn': 0.0, 'L2': 0.0,
    'L3': 0.0, 'L3_new': 0.0, 'L3_act': 0.0, 'L4': 0.0, 'L2_1': 0.0,
    'L2_2': 0.0, 'L2_3': 0.0, 'L2_4': 0.0, 'L2_act1': 0.0, 'L2_act2': 0.0,
    'L2_act3': 0.0, 'L2_act4': 0.0, 'L2_bn_weight': 0.0, 'L2_bn_bias': 0.0,
    'weight_init_scale_conv': 1.0, 'weight_init_scale_fc': 1.0, 'w_scale': 1.0,
    'n_w': 0.0, 'n_w1': 0.0, 'n_w2': 0.0, 'n_w3': 0.0, 'n_w4': 0.0,
    'n_w_test': 0.0, 'selected_weights': 0.0, 'noise_values': 0.0,
    'selected_weights_noise_scale': 0.0, 'scale_weights': 0.0,
    'stochastic': 0.5, 'pctl': 99.98, 'uniform_ind': 0.0, 'uniform_dep': 0.0,
    'normal_ind': 0.0, 'normal_dep': 0.0,
}

_NOISYNET_INTS = {
    'nepochs': 250, 'num_sims': 1, 'num_layers': 4, 'fs': 5, 'fm1': 65,
    'fm2': 120, 'fm3': 256, 'fm4': 512, 'fc': 390, 'width': 1,
    'LR_step_after': 100, 'LR_max_epoch': 10, 'LR_finetune_epochs': 20,
    'early_stop_after': 100,
}

_NOISYNET_STRS = {
    'dataset': 'data/cifar_RGB_4bit.npz', 'tag': '', 'optim': 'AdamW',
    'LR_scheduler': 'manual', 'weight_init': 'default', 'var_name': '',
}


def build_noisynet_parser():
    parser = argparse.ArgumentParser(
        description='NoisyNet-MI355X CIFAR-10 training',
        formatter_class=argparse.ArgumentDefaultsHelpFormatter)
    for name, default in _NOISYNET_BOOL_PAIRS:
        _bool_pair(parser, name, default)
    parser.add_argument('--resume', type=str, default=None, metavar='')
    parser.add_argument('-a', '--arch', metavar='ARCH', default='noisynet')
    for name, default in _NOISYNET_STRS.items():
        parser.add_argument('--' + name, type=str, default=default, metavar='')
    for name, default in _NOISYNET_FLOATS.items():
        parser.add_argument('--' + name, type=float, default=default, metavar='')
    for name, default in _NOISYNET_INTS.items():
        parser.add_argument('--' + name, type=int, default=default, metavar='')
    parser.add_argument('--batch_size', '--batchsize', '--batch-size', '--bs',
                        type=int, default=64, metavar='')
    parser.add_argument('--block_size', type=int, default=None, metavar='')
    parser.add_argument('--q_a', type=int, default=0, metavar='')
    parser.add_argument('--q_w', type=int, default=0, metavar='')
    for i in (1, 2, 3, 4):
        parser.add_argument('--q_a%d' % i, type=int, default=0, metavar='')
        parser.add_argument('--q_w%d' % i, type=int, default=0, metavar='')
    parser.add_argument('--seed', type=int, default=None, metavar='')
    parser.add_argument('--distort_act', dest='distort_act', action='store_true')
    parser.add_argument('--L3_L2', dest='L3_L2', action='store_true')
    parser.add_argument('--L3_L1', dest='L3_L1', action='store_true')
    parser.add_argument('--selection_criteria', type=str, default=None, metavar='')
    parser.add_argument('--gpu', default=None, type=str)
    # MI355X additions (do not collide with reference flags)
    parser.add_argument('--bf16', dest='bf16', action='store_true',
                        help='run compute in bfloat16 (MI355X native)')
    parser.add_argument('--local_rank', '--local-rank', type=int, default=0)
    parser.add_argument('--n_train', type=int, default=50000,
                        help='synthetic-data train set size (no npz present)')
    parser.add_argument('--n_test', type=int, default=10000,
                        help='synthetic-data test set size (no npz present)')
    return parser


def broadcast_per_layer(args):
    """Global -> per-layer flag broadcast (noisynet.py:725-734, :861-903)."""
    if args.current > 0:
        args.current1 = args.current2 = args.current3 = args.current4 = args.current
    args.layer_currents = [args.current1, args.current2, args.current3, args.current4]
    if args.q_a > 0:
        args.q_a1 = args.q_a2 = args.q_a3 = args.q_a4 = args.q_a
    if args.q_w > 0:
        args.q_w1 = args.q_w2 = args.q_w3 = args.q_w4 = args.q_w
    if args.L2 > 0:
        if args.q_a2 == 1:
            args.L2_1 = args.L2_2 = args.L2_3 = args.L2_4 = args.L2 * args.width
        else:
            args.L2_1 = args.L2_2 = args.L2_3 = args.L2_4 = args.L2
    if args.L1 > 0:
        args.L1_1 = args.L1_2 = args.L1_3 = args.L1_4 = args.L1
    if args.act_max > 0:
        args.act_max1 = args.act_max2 = args.act_max3 = args.act_max
    if args.w_max > 0:
        args.w_max1 = args.w_max2 = args.w_max3 = args.w_max4 = args.w_max
    if args.n_w > 0:
        args.n_w1 = args.n_w2 = args.n_w3 = args.n_w4 = args.n_w
    if args.LR_1 == 0:
        args.LR_1 = args.LR
    if args.LR_2 == 0:
        args.LR_2 = args.LR
    if args.LR_3 == 0:
        args.LR_3 = args.LR
    if args.LR_4 == 0:
        args.LR_4 = args.LR
    return args


def var_list_for(var_name, current=0.0):
    """Hyperparameter sweep grids (--var_name); last-assignment-wins values
    from noisynet.py:755-854."""
    grids = {
        'current': [1, 3, 5, 10, 20, 50, 100],
        'w_max1': [0.1, 0.2, 0.3, 0.4, 0.5, 0.6, 0.8, 1],
        'act_max': [0.25, 1, 2, 4, 10, 0],
        'act_max1': [0.5, 1, 1.5, 2, 2.5, 3, 4, 5],
        'act_max2': [0.5, 1, 2, 3, 4, 5, 10],
        'act_max3': [0.5, 1, 2, 3, 4, 5, 10],
        'LR': [0.0001, 0.0002, 0.0003, 0.0005, 0.001, 0.002, 0.003, 0.004,
               0.006, 0.008, 0.01],
        'L2_act_max': [0.0001, 0.0002, 0.0005, 0.001, 0.002, 0.005, 0.01,
                       0.02, 0.03, 0.05],
        'uniform_ind': [x / current for x in [0.12, 0.14, 0.16]] if current else [],
        'uniform_dep': [0.2, 0.3, 0.4, 0.5, 0.6, 0.7, 0.8, 0.9, 1],
        'normal_ind': [x / current for x in [0.05, 0.07, 0.09]] if current else [],
        'normal_dep': [x / current for x in [0.3, 0.4, 0.5]] if current else [],
        'L2_1': [0.0, 0.0002, 0.0005, 0.001, 0.002, 0.003, 0.005],
        'L2': [0, 0.0005, 0.001, 0.002, 0.005, 0.01, 0.02, 0.03, 0.04, 0.05,
               0.07, 0.1, 0.15, 0.2, 0.25, 0.3, 0.4],
        'L1': [2e-6, 4e-6, 6e-6, 8e-6, 1e-5, 2e-5, 3e-5],
        'L2_2': [0.0, 0.00001, 0.00002, 0.00003, 0.00005, 0.0001],
        'L3': [0, 0.0005, 0.001, 0.002, 0.003, 0.005, 0.007, 0.01, 0.02, 0.03,
               0.04, 0.06, 0.08, 0.1, 0.2, 0.3, 0.5, 1],
        'L3_new': [0, 0.005, 0.01, 0.02, 0.05, 0.1, 0.2, 0.3, 0.4, 0.5, 1],
        'L3_act': [0.001, 0.005, 0.01, 0.05, 0.1, 0.2, 0.5, 1, 2],
        'L4': [0.00002, 0.00005, 0.0001, 0.0002, 0.0005, 0.001, 0.002, 0.005,
               0.01, 0.02, 0.05, 0.1, 0.2, 0.5, 1, 2, 5],
        'momentum': [0., 0.5, 0.7, 0.8, 0.85, 0.9, 0.95, 0.97, 0.99],
        'grad_clip': [0.005, 0.05, 0.5, 2, 0],
        'dropout': [0, 0.1, 0.15, 0.2, 0.25, 0.3, 0.35, 0.4, 0.5],
        'width': [1, 2, 4],
        'noise': [0, 0.02, 0.05, 0.1, 0.15, 0.2, 0.25, 0.3, 0.4, 0.5],
        'n_w': [0, 0.02, 0.05, 0.1, 0.2, 0.3, 0.4, 0.5],
        'selected_weights': [2, 5, 10],
        'L2_w_max': [0.1],
    }
    return grids.get(var_name, [' '])


# ---------------------------------------------------------------------------
# main.py (ImageNet) parser -- reference main.py:40-192
# ---------------------------------------------------------------------------

_MAIN_BOOL_PAIRS = [
    ('pretrained', False), ('debug_quant', False), ('normalize', False),
    ('dali', True), ('amp', False), ('dali_cpu', True), ('merge_bn', False),
    ('bn_out', False), ('fp16', False), ('track_running_stats', True),
    ('plot', False), ('print_shapes', False), ('plot_basic', False),
    ('calculate_running', False), ('q_inplace', False),
    ('ignore_best_acc', True), ('reset_start_epoch', False),
]


def build_main_parser():
    parser = argparse.ArgumentParser(
        description='NoisyNet-MI355X ImageNet Training',
        formatter_class=argparse.ArgumentDefaultsHelpFormatter)
    parser.add_argument('--data', default='/data/imagenet/', metavar='DIR')
    parser.add_argument('-a', '--arch', metavar='ARCH', default='resnet18')
    parser.add_argument('-j', '--workers', default=10, type=int, metavar='N')
    parser.add_argument('--epochs', default=150, type=int, metavar='N')
    parser.add_argument('--start-epoch', default=0, type=int, metavar='N',
                        dest='start_epoch')
    parser.add_argument('-b', '--batch_size', '--batchsize', '--batch-size',
                        '--bs', default=256, type=int, metavar='N')
    parser.add_argument('--lr', '--LR', '--learning-rate', default=0.1,
                        type=float, metavar='LR', dest='lr')
    parser.add_argument('--gamma', type=float, default=0.1)
    parser.add_argument('--momentum', default=0.9, type=float, metavar='M')
    parser.add_argument('--L1', type=float, default=0.000, metavar='')
    parser.add_argument('--wd', '--L2', '--weight-decay', default=1e-4,
                        type=float, metavar='W', dest='weight_decay')
    parser.add_argument('--L3', type=float, default=0.000, metavar='')
    parser.add_argument('-p', '--print-freq', default=1000, type=int,
                        metavar='N', dest='print_freq')
    parser.add_argument('--resume', default='', type=str, metavar='PATH')
    parser.add_argument('--tag', default='', type=str, metavar='PATH')
    parser.add_argument('-e', '--evaluate', dest='evaluate', action='store_true')
    parser.add_argument('--debug', dest='debug', action='store_true')
    parser.add_argument('--distort_w_test', dest='distort_w_test', action='store_true')
    parser.add_argument('--distort_act', dest='distort_act', action='store_true')
    parser.add_argument('--distort_pre_act', dest='distort_pre_act', action='store_true')
    parser.add_argument('--distort_act_test', dest='distort_act_test', action='store_true')
    parser.add_argument('--noise', default=0, type=float)
    parser.add_argument('--stochastic', default=0.5, type=float)
    parser.add_argument('--step-after', default=30, type=int, dest='step_after')
    parser.add_argument('--seed', default=None, type=int)
    parser.add_argument('--num_sims', default=1, type=int)
    parser.add_argument('--var_name', default=None, type=str)
    parser.add_argument('--q_a', default=4, type=int)
    parser.add_argument('--q_a_first', default=0, type=int)
    parser.add_argument('--q_w', default=0, type=int)
    parser.add_argument('--n_w', type=float, default=0, metavar='')
    parser.add_argument('--n_w_test', type=float, default=0, metavar='')
    parser.add_argument('--local_rank', '--local-rank', default=0, type=int)
    parser.add_argument('--world_size', default=1, type=int)
    parser.add_argument('--block_size', type=int, default=None, metavar='')
    parser.add_argument('--act_max', default=0, type=float)
    parser.add_argument('--w_max', default=0, type=float)
    parser.add_argument('--eps', default=1e-7, type=float)
    parser.add_argument('--grad_clip', default=0, type=float)
    parser.add_argument('--q_scale', default=1, type=float)
    parser.add_argument('--scale_bias', default=0, type=float)
    parser.add_argument('--pctl', default=99.98, type=float)
    parser.add_argument('--w_pctl', default=0, type=float)
    parser.add_argument('--offset', default=0, type=float)
    parser.add_argument('--offset_input', default=0, type=float)
    parser.add_argument('--gpu', default=None, type=str)
    parser.add_argument('--amp_level', default='O1', type=str)
    parser.add_argument('--loss_scale', default=128.0, type=float)
    parser.add_argument('--keep-batchnorm-fp32', type=str, default=None,
                        dest='keep_batchnorm_fp32')
    parser.add_argument('--selected_weights', type=float, default=0, metavar='')
    parser.add_argument('--selection_criteria', type=str, default=None, metavar='')
    parser.add_argument('--selected_weights_noise_scale', type=float, default=0, metavar='')
    parser.add_argument('--scale_weights', type=float, default=0, metavar='')
    parser.add_argument('--test_temp', type=float, default=0, metavar='')
    parser.add_argument('--temperature', type=float, default=0, metavar='')
    parser.add_argument('--debug_noise', dest='debug_noise', action='store_true')
    parser.add_argument('--old_checkpoint', dest='old_checkpoint', action='store_true')
    parser.add_argument('--warmup', action='store_true')
    parser.add_argument('--lr-decay', type=str, default='step', dest='lr_decay')
    parser.add_argument('--stuck_at_weights', type=str, default=None, metavar='')
    parser.add_argument('--sync-bn', action='store_true', dest='sync_bn')
    for name, default in _MAIN_BOOL_PAIRS:
        _bool_pair(parser, name, default)
    # MI355X additions
    parser.add_argument('--bf16', dest='bf16', action='store_true')
    parser.add_argument('--synthetic_batches', type=int, default=50,
                        help='batches per epoch for the synthetic loader')
    return parser
