"""Real-image input pipeline for the CNN benchmark example.

Reference: examples/tf_cnn_benchmarks/preprocessing.py — training images
get a distorted bounding-box crop + random flip (+ optional color
distortions), eval images a central 87.5% crop; the input file set is
sharded across workers with parallax.shard
(preprocessing.py:505-520 ``ds = shard.shard(ds)``).

Here the storage format is an image folder (``root/<class>/*.jpg``)
decoded with PIL — there is no TFRecord dependency — and the shard
filter runs over the sorted file list with the same
``i % num_shards == shard_id`` semantics.  Output tensors are NCHW
float32 in [-1, 1] (the inception-style normalization the reference
applies at model input), labels are 1-based with 0 = background
(tf_cnn_benchmarks convention -> num_classes 1001).
"""

import os
import random

import numpy as np
import torch

from parallax_amd import shard as shard_lib

_EXTS = ('.jpg', '.jpeg', '.png', '.bmp')


def list_image_files(data_dir):
    """Sorted (path, label) list; label = 1 + class index over sorted
    class-directory names (0 is the background class)."""
    classes = sorted(d for d in os.listdir(data_dir)
                     if os.path.isdir(os.path.join(data_dir, d)))
    files = []
    for li, cls in enumerate(classes, start=1):
        cdir = os.path.join(data_dir, cls)
        for name in sorted(os.listdir(cdir)):
            if name.lower().endswith(_EXTS):
                files.append((os.path.join(cdir, name), li))
    if not files:
        raise ValueError('found no image files under %s' % data_dir)
    return files


def _random_resized_crop(img, rng):
    """Distorted-bbox crop (preprocessing.py train_image: area 5%-100%,
    aspect 3/4-4/3, 10 attempts, fall back to the full image)."""
    w, h = img.size
    area = w * h
    for _ in range(10):
        target = rng.uniform(0.05, 1.0) * area
        aspect = rng.uniform(3.0 / 4.0, 4.0 / 3.0)
        cw = int(round((target * aspect) ** 0.5))
        ch = int(round((target / aspect) ** 0.5))
        if 0 < cw <= w and 0 < ch <= h:
            x0 = rng.randint(0, w - cw)
            y0 = rng.randint(0, h - ch)
            return img.crop((x0, y0, x0 + cw, y0 + ch))
    return img


def _central_crop(img, fraction=0.875):
    w, h = img.size
    cw, ch = int(w * fraction), int(h * fraction)
    x0, y0 = (w - cw) // 2, (h - ch) // 2
    return img.crop((x0, y0, x0 + cw, y0 + ch))


class ImagePreprocessor(object):
    """Decode + crop + resize one image file to a CHW float tensor in
    [-1, 1] (RecordInputImagePreprocessor.preprocess equivalent)."""

    def __init__(self, image_size, train=True, distortions=False,
                 seed=None):
        self.image_size = image_size
        self.train = train
        self.distortions = distortions
        self.rng = random.Random(seed)

    def __call__(self, path):
        from PIL import Image
        with Image.open(path) as im:
            img = im.convert('RGB')
        if self.train:
            img = _random_resized_crop(img, self.rng)
            if self.rng.random() < 0.5:
                img = img.transpose(0)   # FLIP_LEFT_RIGHT
        else:
            img = _central_crop(img)
        img = img.resize((self.image_size, self.image_size))
        if self.train and self.distortions:
            from PIL import ImageEnhance
            img = ImageEnhance.Brightness(img).enhance(
                self.rng.uniform(0.875, 1.125))
            img = ImageEnhance.Color(img).enhance(
                self.rng.uniform(0.5, 1.5))
            img = ImageEnhance.Contrast(img).enhance(
                self.rng.uniform(0.5, 1.5))
        arr = np.asarray(img, dtype=np.float32) / 127.5 - 1.0
        return torch.from_numpy(arr).permute(2, 0, 1).contiguous()


def minibatch_iterator(data_dir, batch_size, image_size, train=True,
                       distortions=False, num_shards=None, shard_id=None,
                       seed=0, epochs=None):
    """Yield (images [B,3,S,S], labels [B]) batches from this worker's
    shard of the file list.  Shuffles per epoch when training; runs
    forever unless ``epochs`` is given (the reference pipeline loops).
    Incomplete trailing batches are dropped (fixed batch shape, as in
    the reference's RecordInput batching)."""
    files = list_image_files(data_dir)
    mine = list(shard_lib.shard(files, num_shards, shard_id))
    if len(mine) < 1:
        raise ValueError('shard holds no files')
    prep = ImagePreprocessor(image_size, train, distortions, seed)
    rng = random.Random(seed)
    epoch = 0
    while epochs is None or epoch < epochs:
        order = list(mine)
        if train:
            rng.shuffle(order)
        for lo in range(0, len(order) - batch_size + 1, batch_size):
            chunk = order[lo:lo + batch_size]
            xs = torch.stack([prep(p) for p, _ in chunk])
            ys = torch.tensor([l for _, l in chunk], dtype=torch.long)
            yield xs, ys
        epoch += 1
