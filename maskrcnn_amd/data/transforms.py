"""Mask-safe augmentation transforms.

The reference used imgaug Sequential with a mask-safe whitelist
(Fliplr/Flipud/Affine/CropAndPad..., mrcnn/model.py:~1260). imgaug is
unavailable offline; this module provides the equivalent composable,
mask-synchronized transforms in numpy. Every transform maps
(image [H,W,3] uint8, masks [H,W,N] bool) -> same-shape pair.

Usage:
    aug = Sequential([Fliplr(0.5), Rotate90(0.25), Multiply((0.8, 1.2))])
    trainer = Trainer(model, ds, augmentation=aug)
"""

from __future__ import annotations

import numpy as np


class Transform:
    def __call__(self, image, masks, rng=None):
        raise NotImplementedError


class Sequential(Transform):
    def __init__(self, transforms):
        self.transforms = list(transforms)

    def __call__(self, image, masks, rng=None):
        rng = rng or np.random
        for t in self.transforms:
            image, masks = t(image, masks, rng)
        return image, masks


class Fliplr(Transform):
    """Horizontal flip with probability p (the reference's default COCO
    augmentation, imgaug.Fliplr(0.5))."""

    def __init__(self, p=0.5):
        self.p = p

    def __call__(self, image, masks, rng=None):
        rng = rng or np.random
        if rng.rand() < self.p:
            image = np.fliplr(image).copy()
            masks = np.fliplr(masks).copy()
        return image, masks


class Flipud(Transform):
    def __init__(self, p=0.5):
        self.p = p

    def __call__(self, image, masks, rng=None):
        rng = rng or np.random
        if rng.rand() < self.p:
            image = np.flipud(image).copy()
            masks = np.flipud(masks).copy()
        return image, masks


class Rotate90(Transform):
    """k*90-degree rotation with probability p (square images only;
    non-square images are passed through)."""

    def __init__(self, p=0.5):
        self.p = p

    def __call__(self, image, masks, rng=None):
        rng = rng or np.random
        if image.shape[0] == image.shape[1] and rng.rand() < self.p:
            k = int(rng.randint(1, 4))
            image = np.rot90(image, k, axes=(0, 1)).copy()
            masks = np.rot90(masks, k, axes=(0, 1)).copy()
        return image, masks


class Multiply(Transform):
    """Brightness scale drawn from [lo, hi] — image-only (mask-safe by
    construction)."""

    def __init__(self, rng_range=(0.8, 1.2)):
        self.lo, self.hi = rng_range

    def __call__(self, image, masks, rng=None):
        rng = rng or np.random
        f = rng.uniform(self.lo, self.hi)
        image = np.clip(image.astype(np.float32) * f, 0, 255) \
            .astype(image.dtype)
        return image, masks


class AddGaussianNoise(Transform):
    def __init__(self, sigma=5.0, p=0.5):
        self.sigma = sigma
        self.p = p

    def __call__(self, image, masks, rng=None):
        rng = rng or np.random
        if rng.rand() < self.p:
            noise = rng.randn(*image.shape) * self.sigma
            image = np.clip(image.astype(np.float32) + noise, 0, 255) \
                .astype(image.dtype)
        return image, masks


class Crop(Transform):
    """Random crop of `frac` of each side, resized back — geometric, so
    masks follow."""

    def __init__(self, max_frac=0.1, p=0.5):
        self.max_frac = max_frac
        self.p = p

    def __call__(self, image, masks, rng=None):
        rng = rng or np.random
        if rng.rand() >= self.p:
            return image, masks
        from ..utils.image import resize
        H, W = image.shape[:2]
        fy = int(H * rng.uniform(0, self.max_frac))
        fx = int(W * rng.uniform(0, self.max_frac))
        y0 = int(rng.randint(0, fy + 1))
        x0 = int(rng.randint(0, fx + 1))
        ic = image[y0:H - (fy - y0), x0:W - (fx - x0)]
        mc = masks[y0:H - (fy - y0), x0:W - (fx - x0)]
        image = resize(ic, (H, W)).astype(image.dtype)
        mr = resize(mc.astype(np.float32), (H, W), order=0)
        if mr.ndim == 2:
            mr = mr[:, :, None]
        return image, mr > 0.5
