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


class Sometimes(Transform):
    """Apply `child` with probability p (imgaug.Sometimes combinator —
    part of the reference's MASK_AUGMENTERS whitelist,
    mrcnn/model.py:~1260)."""

    def __init__(self, p, child):
        self.p = p
        self.child = child

    def __call__(self, image, masks, rng=None):
        rng = rng or np.random
        if rng.rand() < self.p:
            return self.child(image, masks, rng)
        return image, masks


class OneOf(Transform):
    """Apply exactly one randomly chosen child (imgaug.OneOf)."""

    def __init__(self, children):
        self.children = list(children)

    def __call__(self, image, masks, rng=None):
        rng = rng or np.random
        t = self.children[int(rng.randint(0, len(self.children)))]
        return t(image, masks, rng)


class SomeOf(Transform):
    """Apply n randomly chosen children, in order (imgaug.SomeOf)."""

    def __init__(self, n, children):
        self.n = n
        self.children = list(children)

    def __call__(self, image, masks, rng=None):
        rng = rng or np.random
        k = min(self.n, len(self.children))
        idx = sorted(rng.choice(len(self.children), size=k,
                                replace=False).tolist())
        for i in idx:
            image, masks = self.children[i](image, masks, rng)
        return image, masks


class Affine(Transform):
    """Rotation / scale / translation / shear — the reference's main
    geometric augmenter (imgaug.Affine is on the mask-safe whitelist).
    Image is interpolated bilinearly, masks nearest — both through one
    shared inverse-affine map so they stay registered.

    rotate: degrees range (lo, hi); scale: factor range; translate_frac:
    fraction-of-dim range; shear: degrees range.
    """

    def __init__(self, rotate=(-0.0, 0.0), scale=(1.0, 1.0),
                 translate_frac=(0.0, 0.0), shear=(0.0, 0.0), p=1.0):
        self.rotate = rotate
        self.scale = scale
        self.translate_frac = translate_frac
        self.shear = shear
        self.p = p

    def __call__(self, image, masks, rng=None):
        rng = rng or np.random
        if rng.rand() >= self.p:
            return image, masks
        from scipy import ndimage
        H, W = image.shape[:2]
        ang = np.deg2rad(rng.uniform(*self.rotate))
        sc = rng.uniform(*self.scale)
        sh = np.deg2rad(rng.uniform(*self.shear))
        ty = rng.uniform(*self.translate_frac) * H
        tx = rng.uniform(*self.translate_frac) * W
        # Forward map: rotate+shear+scale about the center, then shift.
        c, s = np.cos(ang), np.sin(ang)
        A = np.array([[c, -np.sin(ang - sh)],
                      [s, np.cos(ang - sh)]]) * sc
        center = np.array([(H - 1) / 2.0, (W - 1) / 2.0])
        # ndimage.affine_transform applies the INVERSE map to outputs.
        Ainv = np.linalg.inv(A)
        offset = center - Ainv @ (center + np.array([ty, tx]))
        img = np.stack([
            ndimage.affine_transform(image[..., ch].astype(np.float32),
                                     Ainv, offset=offset, order=1,
                                     mode="constant", cval=0.0)
            for ch in range(image.shape[2])], axis=-1)
        image = np.clip(img, 0, 255).astype(image.dtype)
        if masks.size:
            m = np.stack([
                ndimage.affine_transform(
                    masks[..., i].astype(np.float32), Ainv,
                    offset=offset, order=0, mode="constant", cval=0.0)
                for i in range(masks.shape[2])], axis=-1)
            masks = m > 0.5
        return image, masks


class GaussianBlur(Transform):
    """Image-only blur, sigma drawn from [lo, hi] (imgaug.GaussianBlur)."""

    def __init__(self, sigma=(0.0, 2.0), p=0.5):
        self.sigma = sigma
        self.p = p

    def __call__(self, image, masks, rng=None):
        rng = rng or np.random
        if rng.rand() >= self.p:
            return image, masks
        from scipy import ndimage
        sig = rng.uniform(*self.sigma)
        if sig <= 0:
            return image, masks
        img = ndimage.gaussian_filter(
            image.astype(np.float32), sigma=(sig, sig, 0))
        return np.clip(img, 0, 255).astype(image.dtype), masks


class ContrastNormalization(Transform):
    """Scale contrast about the per-image mean (imgaug
    ContrastNormalization): out = mean + f * (in - mean)."""

    def __init__(self, rng_range=(0.75, 1.25), p=1.0):
        self.lo, self.hi = rng_range
        self.p = p

    def __call__(self, image, masks, rng=None):
        rng = rng or np.random
        if rng.rand() >= self.p:
            return image, masks
        f = rng.uniform(self.lo, self.hi)
        mean = image.astype(np.float32).mean()
        img = mean + (image.astype(np.float32) - mean) * f
        return np.clip(img, 0, 255).astype(image.dtype), masks


class Add(Transform):
    """Additive brightness shift drawn from [lo, hi] (imgaug.Add)."""

    def __init__(self, rng_range=(-25, 25), p=1.0):
        self.lo, self.hi = rng_range
        self.p = p

    def __call__(self, image, masks, rng=None):
        rng = rng or np.random
        if rng.rand() >= self.p:
            return image, masks
        v = rng.uniform(self.lo, self.hi)
        img = image.astype(np.float32) + v
        return np.clip(img, 0, 255).astype(image.dtype), masks


class CropAndPad(Transform):
    """Crop (negative) or pad (positive) each side by a fraction drawn
    from `percent`, then resize back to the original shape
    (imgaug.CropAndPad, mask-safe whitelist member)."""

    def __init__(self, percent=(-0.1, 0.1), p=1.0):
        self.percent = percent
        self.p = p

    def __call__(self, image, masks, rng=None):
        rng = rng or np.random
        if rng.rand() >= self.p:
            return image, masks
        from ..utils.image import resize
        H, W = image.shape[:2]
        f = rng.uniform(*self.percent)
        dy, dx = int(round(H * f)), int(round(W * f))
        if dy == 0 and dx == 0:
            return image, masks
        if f < 0:  # crop |dy| px off each side
            y0, x0 = -dy, -dx
            ic = image[y0:H - y0, x0:W - x0]
            mc = masks[y0:H - y0, x0:W - x0]
        else:      # pad dy px on each side
            ic = np.zeros((H + 2 * dy, W + 2 * dx) + image.shape[2:],
                          image.dtype)
            ic[dy:dy + H, dx:dx + W] = image
            mc = np.zeros((H + 2 * dy, W + 2 * dx) + masks.shape[2:],
                          masks.dtype)
            mc[dy:dy + H, dx:dx + W] = masks
        image = resize(ic, (H, W)).astype(image.dtype)
        mr = resize(mc.astype(np.float32), (H, W), order=0)
        if mr.ndim == 2:
            mr = mr[:, :, None]
        return image, mr > 0.5


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


class PiecewiseAffine(Transform):
    """Grid-displacement warp (imgaug.PiecewiseAffine equivalent; the
    last member of the reference's mask-safe whitelist,
    mrcnn/model.py:~1260). A coarse nb_rows x nb_cols grid of control
    points is jittered by N(0, scale * size) and bilinearly upsampled
    to a dense displacement field; image sampled order-1, masks
    order-0, through the SAME field so they stay registered."""

    def __init__(self, scale=(0.01, 0.05), nb_rows=4, nb_cols=4, p=1.0):
        self.scale = scale if isinstance(scale, tuple) else (scale, scale)
        self.nb_rows = nb_rows
        self.nb_cols = nb_cols
        self.p = p

    def __call__(self, image, masks, rng=None):
        rng = rng or np.random
        if rng.rand() >= self.p:
            return image, masks
        from scipy import ndimage
        H, W = image.shape[:2]
        s = rng.uniform(*self.scale)
        if s <= 0:
            return image, masks
        dy = rng.randn(self.nb_rows, self.nb_cols) * s * H
        dx = rng.randn(self.nb_rows, self.nb_cols) * s * W
        zoom = (H / self.nb_rows, W / self.nb_cols)
        dy = ndimage.zoom(dy, zoom, order=1)[:H, :W]
        dx = ndimage.zoom(dx, zoom, order=1)[:H, :W]
        yy, xx = np.mgrid[0:H, 0:W].astype(np.float32)
        coords = np.stack([yy + dy, xx + dx])
        img = np.stack([
            ndimage.map_coordinates(image[..., ch].astype(np.float32),
                                    coords, order=1, mode="reflect")
            for ch in range(image.shape[2])], axis=-1)
        image = np.clip(img, 0, 255).astype(image.dtype)
        if masks.size:
            m = np.stack([
                ndimage.map_coordinates(masks[..., i].astype(np.float32),
                                        coords, order=0, mode="constant",
                                        cval=0.0)
                for i in range(masks.shape[2])], axis=-1)
            masks = m > 0.5
        return image, masks
