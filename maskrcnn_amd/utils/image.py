"""Image resize / mold utilities.

Replicates reference mrcnn/utils.py resize_image (~450) — modes
square / pad64 / crop / none — and mrcnn/model.py mold_image / unmold_image
(~2640). `resize` is reimplemented with torch bilinear interpolation
(the reference delegated to skimage, unavailable here; numerics are
interpolation-equivalent, not bit-identical).
"""

from __future__ import annotations

import random

import numpy as np
import torch
import torch.nn.functional as F


def resize(image: np.ndarray, output_shape, order=1, preserve_range=True,
           anti_aliasing=False):
    """Resize an image array to output_shape (H, W[, C]).

    Reference: mrcnn/utils.py resize (skimage wrapper, ~890). Implemented
    with torch.nn.functional.interpolate (bilinear for order=1, nearest for
    order=0). Values are returned in the input range (preserve_range
    semantics; the flag is accepted for API parity).
    """
    oh, ow = int(output_shape[0]), int(output_shape[1])
    if image.shape[0] == oh and image.shape[1] == ow:
        return image.copy()
    arr = np.asarray(image)
    if not arr.flags.writeable:  # e.g. PIL-backed arrays
        arr = arr.copy()
    squeeze = False
    if arr.ndim == 2:
        arr = arr[:, :, None]
        squeeze = True
    t = torch.from_numpy(np.ascontiguousarray(arr)).float()
    t = t.permute(2, 0, 1).unsqueeze(0)  # [1, C, H, W]
    if order == 0:
        out = F.interpolate(t, size=(oh, ow), mode="nearest")
    else:
        out = F.interpolate(t, size=(oh, ow), mode="bilinear",
                            align_corners=False, antialias=anti_aliasing)
    out = out.squeeze(0).permute(1, 2, 0).numpy()
    if squeeze:
        out = out[:, :, 0]
    if np.issubdtype(image.dtype, np.bool_):
        return out > 0.5
    if np.issubdtype(image.dtype, np.integer) and order == 0:
        return out.astype(image.dtype)
    return out


def resize_image(image, min_dim=None, max_dim=None, min_scale=None,
                 mode="square"):
    """Resize an image keeping the aspect ratio unchanged.

    Reference: mrcnn/utils.py resize_image:~450. Modes:
      none:   no resizing/padding
      square: scale up to min_dim (capped so max side <= max_dim), pad with
              zeros to [max_dim, max_dim]
      pad64:  scale to min_dim (optionally), pad so both sides are multiples
              of 64 (no max_dim cap allowed with min_dim)
      crop:   scale to min_dim then take a random min_dim x min_dim crop

    Returns (image, window, scale, padding, crop):
      window: (y1, x1, y2, x2) of the real image inside the padded one
      scale: applied scale factor
      padding: [(top, bottom), (left, right), (0, 0)]
    """
    image_dtype = image.dtype
    h, w = image.shape[:2]
    window = (0, 0, h, w)
    scale = 1
    padding = [(0, 0), (0, 0), (0, 0)]
    crop = None

    if mode == "none":
        return image, window, scale, padding, crop

    if min_dim:
        scale = max(1, min_dim / min(h, w))
    if min_scale and scale < min_scale:
        scale = min_scale

    if max_dim and mode == "square":
        image_max = max(h, w)
        if round(image_max * scale) > max_dim:
            scale = max_dim / image_max

    if scale != 1:
        image = resize(image, (round(h * scale), round(w * scale)),
                       preserve_range=True)

    if mode == "square":
        h, w = image.shape[:2]
        top_pad = (max_dim - h) // 2
        bottom_pad = max_dim - h - top_pad
        left_pad = (max_dim - w) // 2
        right_pad = max_dim - w - left_pad
        padding = [(top_pad, bottom_pad), (left_pad, right_pad), (0, 0)]
        image = np.pad(image, padding, mode='constant', constant_values=0)
        window = (top_pad, left_pad, h + top_pad, w + left_pad)
    elif mode == "pad64":
        h, w = image.shape[:2]
        assert min_dim % 64 == 0, "Minimum dimension must be a multiple of 64"
        if h % 64 > 0:
            max_h = h - (h % 64) + 64
            top_pad = (max_h - h) // 2
            bottom_pad = max_h - h - top_pad
        else:
            top_pad = bottom_pad = 0
        if w % 64 > 0:
            max_w = w - (w % 64) + 64
            left_pad = (max_w - w) // 2
            right_pad = max_w - w - left_pad
        else:
            left_pad = right_pad = 0
        padding = [(top_pad, bottom_pad), (left_pad, right_pad), (0, 0)]
        image = np.pad(image, padding, mode='constant', constant_values=0)
        window = (top_pad, left_pad, h + top_pad, w + left_pad)
    elif mode == "crop":
        h, w = image.shape[:2]
        y = random.randint(0, (h - min_dim))
        x = random.randint(0, (w - min_dim))
        crop = (y, x, min_dim, min_dim)
        image = image[y:y + min_dim, x:x + min_dim]
        window = (0, 0, min_dim, min_dim)
    else:
        raise Exception(f"Mode {mode} not supported")
    return image.astype(image_dtype), window, scale, padding, crop


def mold_image(images: np.ndarray, config) -> np.ndarray:
    """Subtract the mean pixel and cast to float (reference mold_image)."""
    return images.astype(np.float32) - np.array(config.MEAN_PIXEL)


def unmold_image(normalized_images: np.ndarray, config) -> np.ndarray:
    """Reverse of mold_image."""
    return (normalized_images + np.array(config.MEAN_PIXEL)).astype(np.uint8)


def download_trained_weights(coco_model_path, verbose=1):
    """Reference utils.download_trained_weights fetched the COCO .h5 from
    the Mask_RCNN GitHub releases (utils.py:~870). This build is offline
    by design — kept for API parity with an actionable error."""
    raise RuntimeError(
        "Pretrained COCO weights cannot be downloaded in this offline "
        "build. Train with samples/coco.py or provide a local checkpoint "
        f"at {coco_model_path!r}.")
