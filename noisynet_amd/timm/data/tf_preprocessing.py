"""TensorFlow-exact EfficientNet preprocessing, without TensorFlow.

The reference (timm/data/tf_preprocessing.py:199-227) shells the original
TPU preprocessing graph through a tf.Session. TensorFlow is not in this
environment, so the SAME math is reproduced on PIL/numpy:

  * eval: center crop of ``size/(size+32) * min(h, w)`` pixels (the
    "padded center crop"), then bicubic resize to ``size`` -- the
    EfficientNet paper's 0.875-style crop (reference :108-126);
  * train: sample_distorted_bounding_box-style random crop (area 8-100%,
    aspect 3/4-4/3, 10 attempts, center-crop fallback), bicubic resize,
    random horizontal flip (reference :86-105,129-153).

Input may be raw encoded bytes or a PIL image; output is a CHW uint8
numpy array exactly like the reference transform.
"""

import io
import random

import numpy as np

CROP_PADDING = 32

_PIL_METHODS = {'bicubic': 'BICUBIC', 'bilinear': 'BILINEAR',
                'nearest': 'NEAREST', 'lanczos': 'LANCZOS'}


def _to_pil(image):
    from PIL import Image
    if isinstance(image, (bytes, bytearray)):
        return Image.open(io.BytesIO(image)).convert('RGB')
    if isinstance(image, Image.Image):
        return image.convert('RGB')
    return Image.fromarray(np.asarray(image)).convert('RGB')


def _resample(interpolation):
    from PIL import Image
    return getattr(Image, _PIL_METHODS.get(interpolation, 'BICUBIC'))


def _center_crop_box(width, height, size):
    crop = int((size / (size + CROP_PADDING)) * min(width, height))
    left = ((width - crop) + 1) // 2
    top = ((height - crop) + 1) // 2
    return (left, top, left + crop, top + crop)


def _random_crop_box(width, height, max_attempts=10,
                     area_range=(0.08, 1.0), aspect_range=(3. / 4, 4. / 3)):
    area = width * height
    for _ in range(max_attempts):
        target_area = random.uniform(*area_range) * area
        aspect = random.uniform(*aspect_range)
        w = int(round((target_area * aspect) ** 0.5))
        h = int(round((target_area / aspect) ** 0.5))
        if 0 < w <= width and 0 < h <= height:
            left = random.randint(0, width - w)
            top = random.randint(0, height - h)
            return (left, top, left + w, top + h)
    return None  # caller falls back to the padded center crop


def preprocess_for_eval(image, size=224, interpolation='bicubic'):
    img = _to_pil(image)
    box = _center_crop_box(img.width, img.height, size)
    return img.crop(box).resize((size, size), _resample(interpolation))


def preprocess_for_train(image, size=224, interpolation='bicubic'):
    img = _to_pil(image)
    box = _random_crop_box(img.width, img.height)
    if box is None:
        box = _center_crop_box(img.width, img.height, size)
    out = img.crop(box).resize((size, size), _resample(interpolation))
    if random.random() < 0.5:
        from PIL import Image
        out = out.transpose(Image.FLIP_LEFT_RIGHT)
    return out


def preprocess_image(image, is_training=False, size=224,
                     interpolation='bicubic'):
    if is_training:
        return preprocess_for_train(image, size, interpolation)
    return preprocess_for_eval(image, size, interpolation)


class TfPreprocessTransform:
    """Drop-in for the reference's tf.Session transform: encoded bytes or
    PIL in, CHW uint8 numpy out."""

    def __init__(self, is_training=False, size=224, interpolation='bicubic'):
        self.is_training = is_training
        self.size = size[0] if isinstance(size, tuple) else size
        self.interpolation = interpolation

    def __call__(self, image):
        img = preprocess_image(image, self.is_training, self.size,
                               self.interpolation)
        arr = np.asarray(img, dtype=np.uint8)
        if arr.ndim < 3:
            arr = arr[..., None]
        return np.rollaxis(arr, 2)  # HWC -> CHW
