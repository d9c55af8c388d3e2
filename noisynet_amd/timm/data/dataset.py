"""Datasets: folder scanner + synthetic fallback.

Reference timm/data/dataset.py scans class-subfolder image trees (and
tarfiles). This environment has no image datasets, so ``Dataset`` falls
back to an in-memory synthetic ImageNet-shaped dataset of the same
interface when the path does not exist; the folder-scanning path is kept
for real data.
"""

import os
import re

import numpy as np
import torch.utils.data as data

IMG_EXTENSIONS = ['.png', '.jpg', '.jpeg']


def natural_key(string_):
    return [int(s) if s.isdigit() else s
            for s in re.split(r'(\d+)', string_.lower())]


def find_images_and_targets(folder, types=IMG_EXTENSIONS, class_to_idx=None,
                            leaf_name_only=True, sort=True):
    labels = []
    filenames = []
    for root, _, files in os.walk(folder, topdown=False):
        rel_path = os.path.relpath(root, folder) if root != folder else ''
        label = os.path.basename(rel_path) if leaf_name_only \
            else rel_path.replace(os.path.sep, '_')
        for f in files:
            base, ext = os.path.splitext(f)
            if ext.lower() in types:
                filenames.append(os.path.join(root, f))
                labels.append(label)
    if class_to_idx is None:
        unique_labels = set(labels)
        sorted_labels = list(sorted(unique_labels, key=natural_key))
        class_to_idx = {c: idx for idx, c in enumerate(sorted_labels)}
    images_and_targets = [(f, class_to_idx[l])
                          for f, l in zip(filenames, labels)
                          if l in class_to_idx]
    if sort:
        images_and_targets = sorted(images_and_targets,
                                    key=lambda k: natural_key(k[0]))
    return images_and_targets, class_to_idx


class SyntheticImageDataset(data.Dataset):
    """ImageNet-shaped random images as PIL-free numpy uint8 HWC arrays."""

    def __init__(self, num_samples=1000, size=224, num_classes=1000, seed=42):
        self.num_samples = num_samples
        self.size = size
        self.num_classes = num_classes
        self.seed = seed
        self.transform = None

    def __len__(self):
        return self.num_samples

    def __getitem__(self, index):
        rng = np.random.RandomState(self.seed + index)
        img = rng.randint(0, 256, (3, self.size, self.size), dtype=np.uint8)
        target = int(rng.randint(0, self.num_classes))
        if self.transform is not None:
            img = self.transform(img)
        return img, target


class Dataset(data.Dataset):
    """Folder dataset with synthetic fallback when the path is absent."""

    def __init__(self, root, load_bytes=False, transform=None):
        self.root = root
        self.transform = transform
        if root and os.path.isdir(root):
            images, class_to_idx = find_images_and_targets(root)
            if len(images) == 0:
                raise RuntimeError('Found 0 images in %s' % root)
            self.samples = images
            self.class_to_idx = class_to_idx
            self._synthetic = None
        else:
            self._synthetic = SyntheticImageDataset()
            self.samples = [('synthetic', 0)] * len(self._synthetic)
            self.class_to_idx = {str(i): i for i in range(1000)}

    def __len__(self):
        return len(self.samples)

    def __getitem__(self, index):
        if self._synthetic is not None:
            img, target = self._synthetic[index]
            if self.transform is not None:
                img = self.transform(img)
            return img, target
        from PIL import Image
        path, target = self.samples[index]
        img = Image.open(path).convert('RGB')
        if self.transform is not None:
            img = self.transform(img)
        return img, target

    def filenames(self, indices=None, basename=False):
        if indices:
            fns = [self.samples[i][0] for i in indices]
        else:
            fns = [s[0] for s in self.samples]
        if basename:
            fns = [os.path.basename(f) for f in fns]
        return fns


def scan_tar_members(tf, types=IMG_EXTENSIONS):
    """Index a class-per-directory image tarball: returns [(TarInfo, class
    index)] in natural-key order (capability parity with the reference's
    tar scanner, timm/data/dataset.py:92-110)."""
    entries = []
    class_names = set()
    for member in tf.getmembers():
        if not member.isfile():
            continue
        dirname, fname = os.path.split(member.path)
        if os.path.splitext(fname)[1].lower() not in types:
            continue
        label = os.path.basename(dirname)
        class_names.add(label)
        entries.append((member, label))
    class_to_idx = {c: i for i, c in
                    enumerate(sorted(class_names, key=natural_key))}
    entries.sort(key=lambda e: natural_key(e[0].path))
    return [(m, class_to_idx[l]) for m, l in entries], class_to_idx


class DatasetTar(data.Dataset):
    """Image dataset backed by a single tarball of class subdirectories
    (reference timm/data/dataset.py:115-142).

    The tar is indexed once at construction; each worker process lazily
    reopens its own file handle (tarfile handles cannot be shared across
    fork/spawn boundaries).
    """

    def __init__(self, root, load_bytes=False, transform=None):
        import tarfile
        if not os.path.isfile(root):
            raise FileNotFoundError('DatasetTar: %s is not a file' % root)
        self.root = root
        with tarfile.open(root) as tf:
            self.samples, self.class_to_idx = scan_tar_members(tf)
        self._tf = None  # per-process lazy handle
        self.load_bytes = load_bytes
        self.transform = transform

    def __len__(self):
        return len(self.samples)

    def __getitem__(self, index):
        import tarfile
        if self._tf is None:
            self._tf = tarfile.open(self.root)
        member, target = self.samples[index]
        stream = self._tf.extractfile(member)
        if self.load_bytes:
            img = stream.read()
        else:
            from PIL import Image
            img = Image.open(stream).convert('RGB')
        if self.transform is not None:
            img = self.transform(img)
        return img, target

    def filenames(self, indices=None, basename=False):
        picked = ([self.samples[i][0].path for i in indices] if indices
                  else [m.path for m, _ in self.samples])
        return [os.path.basename(p) for p in picked] if basename else picked
