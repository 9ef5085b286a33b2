"""Mask utilities: resize, mini-mask compression, mold/unmold.

Replicates reference mrcnn/utils.py:~560-660 (resize_mask, minimize_mask,
expand_mask, mold_mask, unmold_mask). Also provides an in-repo RLE codec
(COCO-compatible uncompressed RLE) since pycocotools is absent.
"""

from __future__ import annotations

import numpy as np
import scipy.ndimage

from .image import resize


def resize_mask(mask: np.ndarray, scale, padding, crop=None):
    """Resize a [H, W, N] mask with the scale/padding of its image.

    Reference: mrcnn/utils.py resize_mask:~560 (scipy zoom order 0).
    """
    mask = scipy.ndimage.zoom(mask, zoom=[scale, scale, 1], order=0)
    if crop is not None:
        y, x, h, w = crop
        mask = mask[y:y + h, x:x + w]
    else:
        mask = np.pad(mask, padding, mode='constant', constant_values=0)
    return mask


def minimize_mask(bbox, mask, mini_shape):
    """Crop masks to their bbox and resize to mini_shape (memory saving).

    Reference: mrcnn/utils.py minimize_mask:~580.
    """
    mini_mask = np.zeros(mini_shape + (mask.shape[-1],), dtype=bool)
    for i in range(mask.shape[-1]):
        m = mask[:, :, i].astype(bool)
        y1, x1, y2, x2 = bbox[i][:4]
        m = m[y1:y2, x1:x2]
        if m.size == 0:
            raise Exception("Invalid bounding box with area of zero")
        m = resize(m, mini_shape)
        mini_mask[:, :, i] = np.around(m).astype(bool)
    return mini_mask


def expand_mask(bbox, mini_mask, image_shape):
    """Reverse of minimize_mask (reference expand_mask:~600)."""
    mask = np.zeros(image_shape[:2] + (mini_mask.shape[-1],), dtype=bool)
    for i in range(mask.shape[-1]):
        m = mini_mask[:, :, i]
        y1, x1, y2, x2 = bbox[i][:4]
        h = y2 - y1
        w = x2 - x1
        m = resize(m, (h, w))
        mask[y1:y2, x1:x2, i] = np.around(m).astype(bool)
    return mask


def mold_mask(mask, config):
    pass  # kept for API parity with the reference (a TODO there as well)


def unmold_mask(mask: np.ndarray, bbox, image_shape):
    """Convert a 28x28 float mask to a full-size binary mask.

    Reference: mrcnn/utils.py unmold_mask:~630 — resize to bbox size,
    threshold at 0.5, paste into a zero [H, W] canvas.
    """
    threshold = 0.5
    y1, x1, y2, x2 = bbox
    mask = resize(mask, (y2 - y1, x2 - x1))
    mask = np.where(mask >= threshold, 1, 0).astype(bool)
    full_mask = np.zeros(image_shape[:2], dtype=bool)
    full_mask[y1:y2, x1:x2] = mask
    return full_mask


# --------------------------------------------------------------------------
# RLE codec (COCO-compatible; replaces pycocotools maskApi.c for eval)
# --------------------------------------------------------------------------

def _rle_ext():
    """The native C++ RLE codec (csrc/rle.cpp, pycocotools maskApi
    parity) when the extension is built; None otherwise."""
    from ..ops._ext import load_ext
    ext = load_ext()
    return ext if ext is not None and hasattr(ext, "rle_encode") else None


def rle_encode(mask: np.ndarray) -> dict:
    """Encode a binary [H, W] mask as COCO uncompressed RLE.

    Column-major (Fortran) order, counts alternate 0s/1s starting with 0s —
    the pycocotools convention (reference N12). Uses the native C++ codec
    when built.
    """
    h, w = mask.shape
    ext = _rle_ext()
    if ext is not None:
        import torch
        counts = ext.rle_encode(
            torch.from_numpy(np.ascontiguousarray(mask))).tolist()
        return {"size": [h, w], "counts": counts}
    flat = np.asfortranarray(mask).flatten(order="F").astype(np.uint8)
    diffs = np.flatnonzero(flat[1:] != flat[:-1]) + 1
    runs = np.diff(np.concatenate([[0], diffs, [flat.size]]))
    counts = runs.tolist()
    if flat.size and flat[0] == 1:
        counts = [0] + counts
    return {"size": [h, w], "counts": counts}


def rle_decode(rle: dict) -> np.ndarray:
    """Decode COCO uncompressed RLE to a binary [H, W] mask."""
    h, w = rle["size"]
    counts = rle["counts"]
    ext = _rle_ext()
    if ext is not None:
        import torch
        return ext.rle_decode(torch.tensor(counts, dtype=torch.long),
                              h, w).numpy()
    flat = np.zeros(h * w, dtype=bool)
    pos = 0
    val = False
    for c in counts:
        if val:
            flat[pos:pos + c] = True
        pos += c
        val = not val
    return flat.reshape((h, w), order="F")


def rle_iou(rle1: dict, rle2: dict, iscrowd: bool = False) -> float:
    """IoU of two RLE masks (pycocotools iou semantics; crowd => inter/area1)."""
    m1 = rle_decode(rle1)
    m2 = rle_decode(rle2)
    inter = np.logical_and(m1, m2).sum()
    if iscrowd:
        denom = m1.sum()
    else:
        denom = np.logical_or(m1, m2).sum()
    return float(inter) / max(float(denom), 1e-8)
