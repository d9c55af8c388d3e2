"""create_scheduler dispatch (reference timm/scheduler/scheduler_factory.py:6-43)."""

from .cosine_lr import CosineLRScheduler
from .plateau_lr import PlateauLRScheduler
from .step_lr import StepLRScheduler
from .tanh_lr import TanhLRScheduler


def create_scheduler(args, optimizer):
    num_epochs = args.epochs
    lr_scheduler = None
    if args.sched == 'cosine':
        lr_scheduler = CosineLRScheduler(
            optimizer, t_initial=num_epochs,
            t_mul=getattr(args, 'lr_cycle_mul', 1.),
            lr_min=args.min_lr, decay_rate=args.decay_rate,
            warmup_lr_init=args.warmup_lr, warmup_t=args.warmup_epochs,
            cycle_limit=getattr(args, 'lr_cycle_limit', 1),
            t_in_epochs=True)
        num_epochs = lr_scheduler.get_cycle_length() + args.cooldown_epochs
    elif args.sched == 'tanh':
        lr_scheduler = TanhLRScheduler(
            optimizer, t_initial=num_epochs,
            t_mul=getattr(args, 'lr_cycle_mul', 1.),
            lr_min=args.min_lr, warmup_lr_init=args.warmup_lr,
            warmup_t=args.warmup_epochs,
            cycle_limit=getattr(args, 'lr_cycle_limit', 1),
            t_in_epochs=True)
        num_epochs = lr_scheduler.get_cycle_length() + args.cooldown_epochs
    elif args.sched == 'step':
        lr_scheduler = StepLRScheduler(
            optimizer, decay_t=args.decay_epochs, decay_rate=args.decay_rate,
            warmup_lr_init=args.warmup_lr, warmup_t=args.warmup_epochs)
    elif args.sched == 'plateau':
        lr_scheduler = PlateauLRScheduler(
            optimizer, decay_rate=args.decay_rate,
            patience_t=args.patience_epochs, lr_min=args.min_lr,
            warmup_lr_init=args.warmup_lr, warmup_t=args.warmup_epochs,
            cooldown_t=args.cooldown_epochs)
    return lr_scheduler, num_epochs
