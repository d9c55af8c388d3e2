"""timm-style training driver (`python train_efficientnet.py ...`).

Reproduces the reference train_efficientnet.py flow:
YAML-config-over-argparse, registry create_model, prefetcher loaders,
mixup, EMA, distributed metric reduction, CheckpointSaver with top-k +
recovery files, cosine scheduler stepped per-iteration, and the
images/sec rate log line (the reference's headline throughput metric,
train_efficientnet.py:514-530).
"""

import argparse
import logging
import os
import time
from collections import OrderedDict
from datetime import datetime

import torch
import torch.nn as nn

try:
    import yaml
except ImportError:  # pragma: no cover
    yaml = None

from .. import distributed as dist_mod
from ..timm.data import (Dataset, FastCollateMixup, create_loader,
                         resolve_data_config)
from ..timm.loss import (FusedCrossEntropy, LabelSmoothingCrossEntropy,
                         SoftTargetCrossEntropy)
from ..timm.models import create_model, resume_checkpoint
from ..timm.optim import create_optimizer
from ..timm.scheduler import create_scheduler
from ..timm.utils import (AverageMeter, CheckpointSaver, ModelEma, accuracy,
                          get_state_dict, reduce_tensor,
                          setup_default_logging, update_summary)

config_parser = parser = argparse.ArgumentParser(description='Training Config',
                                                 add_help=False)
parser.add_argument('-c', '--config', default='', type=str, metavar='FILE',
                    help='YAML config file specifying default arguments')

parser = argparse.ArgumentParser(description='NoisyNet-MI355X timm-style training')
parser.add_argument('data', nargs='?', default='', metavar='DIR')
parser.add_argument('--model', default='efficientnet_b0', type=str)
parser.add_argument('--pretrained', action='store_true', default=False)
parser.add_argument('--initial-checkpoint', default='', type=str)
parser.add_argument('--resume', default='', type=str)
parser.add_argument('--num-classes', type=int, default=1000)
parser.add_argument('--img-size', type=int, default=None)
parser.add_argument('--mean', type=float, nargs='+', default=None)
parser.add_argument('--std', type=float, nargs='+', default=None)
parser.add_argument('--interpolation', default='', type=str)
parser.add_argument('-b', '--batch-size', type=int, default=32)
parser.add_argument('-vb', '--validation-batch-size-multiplier', type=int, default=1)
parser.add_argument('--drop', type=float, default=0.0)
parser.add_argument('--drop-connect', type=float, default=0.0)
parser.add_argument('--opt', default='sgd', type=str)
parser.add_argument('--opt-eps', default=1e-8, type=float)
parser.add_argument('--momentum', type=float, default=0.9)
parser.add_argument('--weight-decay', type=float, default=0.0001)
parser.add_argument('--sched', default='step', type=str)
parser.add_argument('--lr', type=float, default=0.01)
parser.add_argument('--warmup-lr', type=float, default=0.0001)
parser.add_argument('--min-lr', type=float, default=1e-5)
parser.add_argument('--epochs', type=int, default=200)
parser.add_argument('--start-epoch', default=None, type=int)
parser.add_argument('--decay-epochs', type=int, default=30)
parser.add_argument('--warmup-epochs', type=int, default=3)
parser.add_argument('--cooldown-epochs', type=int, default=10)
parser.add_argument('--patience-epochs', type=int, default=10)
parser.add_argument('--decay-rate', '--dr', type=float, default=0.1)
parser.add_argument('--bn-momentum', type=float, default=None)
parser.add_argument('--bn-eps', type=float, default=None)
parser.add_argument('--reprob', type=float, default=0.)
parser.add_argument('--remode', type=str, default='const')
parser.add_argument('--recount', type=int, default=1)
parser.add_argument('--mixup', type=float, default=0.0)
parser.add_argument('--mixup-off-epoch', default=0, type=int)
parser.add_argument('--smoothing', type=float, default=0.1)
parser.add_argument('--bn-tf', action='store_true', default=False)
parser.add_argument('--model-ema', action='store_true', default=False)
parser.add_argument('--model-ema-force-cpu', action='store_true', default=False)
parser.add_argument('--model-ema-decay', type=float, default=0.9998)
parser.add_argument('--seed', type=int, default=42)
parser.add_argument('--log-interval', type=int, default=50)
parser.add_argument('--recovery-interval', type=int, default=0)
parser.add_argument('-j', '--workers', type=int, default=4)
parser.add_argument('--num-gpu', type=int, default=1)
parser.add_argument('--save-images', action='store_true', default=False)
parser.add_argument('--amp', action='store_true', default=False)
parser.add_argument('--bf16', action='store_true', default=False,
                    help='bf16 compute (MI355X native)')
parser.add_argument('--sync-bn', action='store_true')
parser.add_argument('--no-prefetcher', action='store_true', default=False)
parser.add_argument('--output', default='', type=str)
parser.add_argument('--eval-metric', default='prec1', type=str)
parser.add_argument('--local_rank', '--local-rank', default=0, type=int)
parser.add_argument('--synthetic-batches', default=20, type=int,
                    help='synthetic batches per epoch when no data dir')


def _parse_args(args=None):
    """YAML config overlay (reference train_efficientnet.py:38-41,164-178)."""
    args_config, remaining = config_parser.parse_known_args(args)
    if args_config.config and yaml is not None:
        with open(args_config.config, 'r') as f:
            cfg = yaml.safe_load(f)
            parser.set_defaults(**cfg)
    parsed = parser.parse_args(remaining)
    args_text = yaml.safe_dump(parsed.__dict__, default_flow_style=False) \
        if yaml is not None else str(parsed.__dict__)
    return parsed, args_text


def main(argv=None):
    setup_default_logging()
    args, args_text = _parse_args(argv)
    args.prefetcher = not args.no_prefetcher and torch.cuda.is_available()
    args.distributed = dist_mod.env_world_size() > 1
    args.device = 'cuda:0' if torch.cuda.is_available() else 'cpu'
    args.world_size = 1
    args.rank = 0
    if args.distributed:
        dist_mod.init_distributed()
        args.world_size = dist_mod.env_world_size()
        args.rank = dist_mod.env_rank()
        args.device = 'cuda:%d' % dist_mod.env_local_rank() \
            if torch.cuda.is_available() else 'cpu'
    torch.manual_seed(args.seed + args.rank)

    model = create_model(
        args.model, pretrained=args.pretrained,
        num_classes=args.num_classes,
        drop_rate=args.drop, drop_connect_rate=args.drop_connect,
        checkpoint_path=args.initial_checkpoint)

    data_config = resolve_data_config(vars(args), model=model,
                                      verbose=args.rank == 0)

    model = model.to(args.device)
    if args.bf16:
        model = model.bfloat16()
        for m in model.modules():
            if isinstance(m, (nn.BatchNorm1d, nn.BatchNorm2d)):
                m.float()
    if torch.cuda.is_available():
        model = model.to(memory_format=torch.channels_last)

    optimizer = create_optimizer(args, model)

    resume_state = {}
    resume_epoch = None
    if args.resume:
        resume_state, resume_epoch = resume_checkpoint(model, args.resume)
        if resume_state and 'optimizer' in resume_state:
            optimizer.load_state_dict(resume_state['optimizer'])

    model_ema = None
    if args.model_ema:
        model_ema = ModelEma(model, decay=args.model_ema_decay,
                             device='cpu' if args.model_ema_force_cpu else '',
                             resume=args.resume)

    if args.sync_bn and args.distributed:
        model = nn.SyncBatchNorm.convert_sync_batchnorm(model)
    dp = dist_mod.DataParallel(model) if args.distributed else None

    lr_scheduler, num_epochs = create_scheduler(args, optimizer)
    start_epoch = 0
    if args.start_epoch is not None:
        start_epoch = args.start_epoch
    elif resume_epoch is not None:
        start_epoch = resume_epoch
    if lr_scheduler is not None and start_epoch > 0:
        lr_scheduler.step(start_epoch)

    # data
    train_dir = os.path.join(args.data, 'train') if args.data else ''
    dataset_train = Dataset(train_dir)
    if not (args.data and os.path.isdir(train_dir)):
        dataset_train = Dataset('')
        dataset_train._synthetic.num_samples = \
            args.synthetic_batches * args.batch_size
        dataset_train.samples = [('synthetic', 0)] * len(dataset_train._synthetic)

    collate_fn = None
    if args.prefetcher and args.mixup > 0:
        collate_fn = FastCollateMixup(args.mixup, args.smoothing,
                                      args.num_classes)

    loader_train = create_loader(
        dataset_train, input_size=data_config['input_size'],
        batch_size=args.batch_size, is_training=True,
        use_prefetcher=args.prefetcher, re_prob=args.reprob,
        re_mode=args.remode, re_count=args.recount,
        mean=data_config['mean'], std=data_config['std'],
        num_workers=args.workers, distributed=args.distributed,
        collate_fn=collate_fn, bf16=args.bf16)

    eval_dir = os.path.join(args.data, 'val') if args.data else ''
    dataset_eval = Dataset(eval_dir)
    if not (args.data and os.path.isdir(eval_dir)):
        dataset_eval = Dataset('')
        dataset_eval._synthetic.num_samples = \
            max(2, args.synthetic_batches // 4) * args.batch_size
        dataset_eval.samples = [('synthetic', 0)] * len(dataset_eval._synthetic)

    loader_eval = create_loader(
        dataset_eval, input_size=data_config['input_size'],
        batch_size=args.validation_batch_size_multiplier * args.batch_size,
        is_training=False, use_prefetcher=args.prefetcher,
        mean=data_config['mean'], std=data_config['std'],
        num_workers=args.workers, distributed=args.distributed,
        bf16=args.bf16)

    if args.mixup > 0.:
        train_loss_fn = SoftTargetCrossEntropy()
        validate_loss_fn = FusedCrossEntropy()
    elif args.smoothing:
        train_loss_fn = LabelSmoothingCrossEntropy(smoothing=args.smoothing)
        validate_loss_fn = FusedCrossEntropy()
    else:
        train_loss_fn = validate_loss_fn = FusedCrossEntropy()

    eval_metric = args.eval_metric
    best_metric = None
    best_epoch = None
    saver = None
    output_dir = ''
    if args.rank == 0:
        output_base = args.output if args.output else './output'
        exp_name = '-'.join([datetime.now().strftime("%Y%m%d-%H%M%S"),
                             args.model, str(data_config['input_size'][-1])])
        output_dir = os.path.join(output_base, 'train', exp_name)
        os.makedirs(output_dir, exist_ok=True)
        saver = CheckpointSaver(checkpoint_dir=output_dir,
                                recovery_dir=output_dir,
                                decreasing=eval_metric == 'loss')
        with open(os.path.join(output_dir, 'args.yaml'), 'w') as f:
            f.write(args_text)

    try:
        for epoch in range(start_epoch, num_epochs):
            if args.distributed and hasattr(loader_train.sampler, 'set_epoch'):
                loader_train.sampler.set_epoch(epoch)

            train_metrics = train_epoch(
                epoch, model, loader_train, optimizer, train_loss_fn, args,
                lr_scheduler=lr_scheduler, saver=saver,
                output_dir=output_dir, model_ema=model_ema, dp=dp)

            eval_metrics = validate(model, loader_eval, validate_loss_fn, args)
            if model_ema is not None and not args.model_ema_force_cpu:
                ema_eval_metrics = validate(model_ema.ema, loader_eval,
                                            validate_loss_fn, args,
                                            log_suffix=' (EMA)')
                eval_metrics = ema_eval_metrics

            if lr_scheduler is not None:
                lr_scheduler.step(epoch + 1, eval_metrics[eval_metric])

            if output_dir:
                update_summary(epoch, train_metrics, eval_metrics,
                               os.path.join(output_dir, 'summary.csv'),
                               write_header=best_metric is None)
            if saver is not None:
                save_metric = eval_metrics[eval_metric]
                best_metric, best_epoch = saver.save_checkpoint(
                    model, optimizer, args, epoch=epoch, model_ema=model_ema,
                    metric=save_metric)
    except KeyboardInterrupt:
        pass
    if best_metric is not None:
        logging.info('*** Best metric: %s (epoch %s)', best_metric, best_epoch)
    return best_metric


def train_epoch(epoch, model, loader, optimizer, loss_fn, args,
                lr_scheduler=None, saver=None, output_dir='', model_ema=None,
                dp=None):
    batch_time_m = AverageMeter()
    data_time_m = AverageMeter()
    losses_m = AverageMeter()
    model.train()
    end = time.time()
    last_idx = len(loader) - 1
    num_updates = epoch * len(loader)
    for batch_idx, (input, target) in enumerate(loader):
        last_batch = batch_idx == last_idx
        data_time_m.update(time.time() - end)
        if not args.prefetcher and torch.cuda.is_available():
            input, target = input.cuda(), target.cuda()
        if input.dtype == torch.uint8:
            input = (input.bfloat16() if args.bf16 else input.float()) / 255.
        if input.is_cuda:
            input = input.contiguous(memory_format=torch.channels_last)

        output = model(input)
        loss = loss_fn(output.float(), target)
        losses_m.update(loss.item(), input.size(0))

        optimizer.zero_grad(set_to_none=False)
        loss.backward()
        if dp is not None:
            dp.finish()
        optimizer.step()

        if torch.cuda.is_available():
            torch.cuda.synchronize()
        if model_ema is not None:
            model_ema.update(model)
        num_updates += 1

        batch_time_m.update(time.time() - end)
        if last_batch or batch_idx % args.log_interval == 0:
            lrl = [pg['lr'] for pg in optimizer.param_groups]
            lr = sum(lrl) / len(lrl)
            if args.rank == 0:
                logging.info(
                    'Train: {} [{:>4d}/{}]  Loss: {:.3g}  '
                    'Time: {:.3f}s, {:>7.2f}/s  LR: {:.3e}  Data: {:.3f}'.format(
                        epoch, batch_idx, len(loader), losses_m.avg,
                        batch_time_m.val,
                        input.size(0) * args.world_size / batch_time_m.val,
                        lr, data_time_m.val))
        if saver is not None and args.recovery_interval and (
                last_batch or (batch_idx + 1) % args.recovery_interval == 0):
            saver.save_recovery(model, optimizer, args, epoch,
                                model_ema=model_ema, batch_idx=batch_idx)
        if lr_scheduler is not None:
            lr_scheduler.step_update(num_updates=num_updates,
                                     metric=losses_m.avg)
        end = time.time()
    return OrderedDict([('loss', losses_m.avg)])


def validate(model, loader, loss_fn, args, log_suffix=''):
    losses_m = AverageMeter()
    prec1_m = AverageMeter()
    prec5_m = AverageMeter()
    model.eval()
    with torch.no_grad():
        for batch_idx, (input, target) in enumerate(loader):
            if not args.prefetcher and torch.cuda.is_available():
                input, target = input.cuda(), target.cuda()
            if input.dtype == torch.uint8:
                input = (input.bfloat16() if args.bf16 else input.float()) / 255.
            if input.is_cuda:
                input = input.contiguous(memory_format=torch.channels_last)
            output = model(input)
            if isinstance(output, (tuple, list)):
                output = output[0]
            loss = loss_fn(output.float(), target)
            prec1, prec5 = accuracy(output, target, topk=(1, 5))
            if args.distributed:
                loss = reduce_tensor(loss.data, args.world_size)
                prec1 = reduce_tensor(prec1, args.world_size)
                prec5 = reduce_tensor(prec5, args.world_size)
            losses_m.update(loss.item(), input.size(0))
            prec1_m.update(prec1.item(), output.size(0))
            prec5_m.update(prec5.item(), output.size(0))
    return OrderedDict([('loss', losses_m.avg), ('prec1', prec1_m.avg),
                        ('prec5', prec5_m.avg)])
