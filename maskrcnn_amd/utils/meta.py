"""Image-meta codec.

Replicates reference mrcnn/model.py:~2560-2660 (compose_image_meta,
parse_image_meta, parse_image_meta_graph). The meta vector layout defines
the serving signature's `input_image_meta` tensor:
  [image_id(1), original_shape(3), molded_shape(3), window(4), scale(1),
   active_class_ids(NUM_CLASSES)]
"""

from __future__ import annotations

import numpy as np
import torch


def compose_image_meta(image_id, original_image_shape, image_shape, window,
                       scale, active_class_ids) -> np.ndarray:
    """Build a length-IMAGE_META_SIZE float vector."""
    meta = np.array(
        [image_id] +
        list(original_image_shape) +
        list(image_shape) +
        list(window) +
        [scale] +
        list(active_class_ids)
    )
    return meta.astype(np.float32)


def parse_image_meta(meta: np.ndarray) -> dict:
    """Parse a [B, META] numpy array into components."""
    image_id = meta[:, 0]
    original_image_shape = meta[:, 1:4]
    image_shape = meta[:, 4:7]
    window = meta[:, 7:11]
    scale = meta[:, 11]
    active_class_ids = meta[:, 12:]
    return {
        "image_id": image_id.astype(np.int32),
        "original_image_shape": original_image_shape.astype(np.int32),
        "image_shape": image_shape.astype(np.int32),
        "window": window.astype(np.int32),
        "scale": scale,
        "active_class_ids": active_class_ids.astype(np.int32),
    }


def parse_image_meta_t(meta: torch.Tensor) -> dict:
    """Parse a [B, META] torch tensor (reference parse_image_meta_graph)."""
    return {
        "image_id": meta[:, 0],
        "original_image_shape": meta[:, 1:4],
        "image_shape": meta[:, 4:7],
        "window": meta[:, 7:11],
        "scale": meta[:, 11],
        "active_class_ids": meta[:, 12:],
    }


def log(text, array=None):
    """Print a text message + optional array stats (reference
    mrcnn/model.py log():~23 — the notebook debugging helper)."""
    if array is not None:
        import numpy as np
        a = np.asarray(array)
        text = text.ljust(25)
        text += (f"shape: {str(a.shape):20}  "
                 f"min: {a.min():10.5f}  max: {a.max():10.5f}"
                 if a.size else f"shape: {str(a.shape):20}  (empty)")
        text += f"  {a.dtype}"
    print(text)
