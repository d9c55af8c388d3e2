"""Loaders: fast uint8 collate + PrefetchLoader with a side HIP stream
overlapping H2D copy + normalization (+fp16/bf16 cast, RandomErasing) with
compute (reference timm/data/loader.py:7-198 -- the cuda.Stream prefetch
structure maps 1:1 onto HIP streams on ROCm)."""

import numpy as np
import torch
import torch.utils.data

from .distributed_sampler import OrderedDistributedSampler
from .random_erasing import RandomErasing


def fast_collate(batch):
    """uint8 HWC/CHW numpy -> stacked uint8 tensor (loader.py:7-14)."""
    targets = torch.tensor([b[1] for b in batch], dtype=torch.int64)
    batch_size = len(targets)
    first = batch[0][0]
    if isinstance(first, np.ndarray):
        tensor = torch.zeros((batch_size, *first.shape), dtype=torch.uint8)
        for i in range(batch_size):
            tensor[i] += torch.from_numpy(batch[i][0])
    elif isinstance(first, torch.Tensor):
        tensor = torch.stack([b[0] for b in batch])
    else:
        raise TypeError(type(first))
    return tensor, targets


class PrefetchLoader:
    """H2D + normalize (+cast, +RandomErasing) on a side stream
    (loader.py:17-87)."""

    def __init__(self, loader, mean=(0., 0., 0.), std=(255., 255., 255.),
                 fp16=False, bf16=False, re_prob=0., re_mode='const',
                 re_count=1, re_num_splits=0):
        self.loader = loader
        self.mean = torch.tensor([x * 1.0 for x in mean]).cuda().view(1, 3, 1, 1)
        self.std = torch.tensor([x * 1.0 for x in std]).cuda().view(1, 3, 1, 1)
        self.fp16 = fp16
        self.bf16 = bf16
        if fp16:
            self.mean = self.mean.half()
            self.std = self.std.half()
        if bf16:
            self.mean = self.mean.bfloat16()
            self.std = self.std.bfloat16()
        if re_prob > 0.:
            self.random_erasing = RandomErasing(
                probability=re_prob, mode=re_mode, min_count=re_count,
                num_splits=re_num_splits)
        else:
            self.random_erasing = None

    def __iter__(self):
        stream = torch.cuda.Stream()
        first = True
        next_input = next_target = None
        for input, target in self.loader:
            with torch.cuda.stream(stream):
                staged_input = input.cuda(non_blocking=True)
                staged_target = target.cuda(non_blocking=True)
                if self.fp16:
                    staged_input = staged_input.half()
                elif self.bf16:
                    staged_input = staged_input.bfloat16()
                else:
                    staged_input = staged_input.float()
                staged_input = staged_input.sub_(self.mean).div_(self.std)
                if self.random_erasing is not None:
                    staged_input = self.random_erasing(staged_input)
            if not first:
                yield next_input, next_target
            else:
                first = False
            torch.cuda.current_stream().wait_stream(stream)
            next_input = staged_input
            next_target = staged_target
        yield next_input, next_target

    def __len__(self):
        return len(self.loader)

    @property
    def sampler(self):
        return self.loader.sampler

    @property
    def dataset(self):
        return self.loader.dataset

    @property
    def mixup_enabled(self):
        if isinstance(self.loader.collate_fn, object) and \
                hasattr(self.loader.collate_fn, 'mixup_enabled'):
            return self.loader.collate_fn.mixup_enabled
        return False

    @mixup_enabled.setter
    def mixup_enabled(self, x):
        if hasattr(self.loader.collate_fn, 'mixup_enabled'):
            self.loader.collate_fn.mixup_enabled = x


def create_loader(dataset, input_size, batch_size, is_training=False,
                  use_prefetcher=True, re_prob=0., re_mode='const',
                  re_count=1, re_split=False, mean=(0., 0., 0.),
                  std=(255., 255., 255.), num_workers=1, distributed=False,
                  collate_fn=None, fp16=False, bf16=False, tf_preprocessing=False):
    from .transforms import transforms_imagenet_eval, transforms_imagenet_train
    size = input_size[-1] if isinstance(input_size, (tuple, list)) else input_size
    if dataset.transform is None:
        if tf_preprocessing:
            from .tf_preprocessing import TfPreprocessTransform
            dataset.transform = TfPreprocessTransform(
                is_training=is_training, size=size)
        elif is_training:
            dataset.transform = transforms_imagenet_train(img_size=size)
        else:
            dataset.transform = transforms_imagenet_eval(img_size=size)

    sampler = None
    if distributed:
        if is_training:
            sampler = torch.utils.data.distributed.DistributedSampler(dataset)
        else:
            sampler = OrderedDistributedSampler(dataset)

    if collate_fn is None:
        collate_fn = fast_collate if use_prefetcher else \
            torch.utils.data.dataloader.default_collate

    loader = torch.utils.data.DataLoader(
        dataset, batch_size=batch_size,
        shuffle=sampler is None and is_training, num_workers=num_workers,
        sampler=sampler, collate_fn=collate_fn, drop_last=is_training)
    if use_prefetcher and torch.cuda.is_available():
        loader = PrefetchLoader(loader, mean=mean, std=std, fp16=fp16,
                                bf16=bf16, re_prob=re_prob if is_training else 0.,
                                re_mode=re_mode, re_count=re_count)
    return loader
