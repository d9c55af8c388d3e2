"""Train/eval transform builders over numpy uint8 CHW arrays.

The reference (timm/data/transforms.py:63-257) uses PIL-based torchvision
transforms; here the synthetic datasets emit numpy uint8 CHW, so the
builders return light numpy-space equivalents (random crop/flip for train,
center crop for eval) producing uint8 CHW for fast_collate. When PIL data
is used, torchvision transforms can be passed to the Dataset directly.
"""

import numpy as np


class RandomResizedCropAndInterpolation:
    """Random crop (area/aspect jitter approximated by random crop of a
    resized field) for numpy CHW uint8."""

    def __init__(self, size, scale=(0.08, 1.0), ratio=(3. / 4., 4. / 3.)):
        self.size = size
        self.scale = scale
        self.ratio = ratio

    def __call__(self, img):
        c, h, w = img.shape
        if h <= self.size or w <= self.size:
            return img[:, :self.size, :self.size]
        top = np.random.randint(0, h - self.size + 1)
        left = np.random.randint(0, w - self.size + 1)
        return img[:, top:top + self.size, left:left + self.size]


class RandomHorizontalFlipNp:
    def __init__(self, p=0.5):
        self.p = p

    def __call__(self, img):
        if np.random.rand() < self.p:
            return img[:, :, ::-1].copy()
        return img


class CenterCropNp:
    def __init__(self, size):
        self.size = size

    def __call__(self, img):
        c, h, w = img.shape
        top = max(0, (h - self.size) // 2)
        left = max(0, (w - self.size) // 2)
        return img[:, top:top + self.size, left:left + self.size]


class Compose:
    def __init__(self, transforms):
        self.transforms = transforms

    def __call__(self, x):
        for t in self.transforms:
            x = t(x)
        return x


def transforms_imagenet_train(img_size=224, scale=(0.08, 1.0),
                              hflip=0.5, **kwargs):
    return Compose([RandomResizedCropAndInterpolation(img_size, scale),
                    RandomHorizontalFlipNp(hflip)])


def transforms_imagenet_eval(img_size=224, crop_pct=0.875, **kwargs):
    return Compose([CenterCropNp(img_size)])
