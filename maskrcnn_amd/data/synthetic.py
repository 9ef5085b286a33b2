"""Synthetic shapes dataset — the hermetic end-to-end fixture.

Replicates the role of reference samples/shapes: random
circles / squares / triangles on a noise background, generated on the fly
with numpy (the reference drew with OpenCV; unavailable and unneeded).
Every layer of the stack can train and detect on this with no downloads.
"""

from __future__ import annotations

import numpy as np

from ..config import Config
from ..utils.boxes import non_max_suppression
from .dataset import Dataset


class ShapesConfig(Config):
    NAME = "shapes"
    GPU_COUNT = 1
    IMAGES_PER_GPU = 2
    NUM_CLASSES = 1 + 3  # background + 3 shape classes
    IMAGE_MIN_DIM = 128
    IMAGE_MAX_DIM = 128
    RPN_ANCHOR_SCALES = (8, 16, 32, 64, 128)
    TRAIN_ROIS_PER_IMAGE = 32
    STEPS_PER_EPOCH = 100
    VALIDATION_STEPS = 5
    MAX_GT_INSTANCES = 4
    DETECTION_MAX_INSTANCES = 10
    BACKBONE = "resnet50"


class ShapesDataset(Dataset):
    """Procedural shapes; specs are generated in load_shapes, drawn on read."""

    SHAPES = ["square", "circle", "triangle"]

    def load_shapes(self, count, height, width, seed=0):
        self.add_class("shapes", 1, "square")
        self.add_class("shapes", 2, "circle")
        self.add_class("shapes", 3, "triangle")
        rng = np.random.RandomState(seed)
        for i in range(count):
            bg_color, shapes = self._random_image(rng, height, width)
            self.add_image("shapes", image_id=i, path=None,
                           width=width, height=height,
                           bg_color=bg_color, shapes=shapes)

    def _random_shape(self, rng, height, width):
        shape = self.SHAPES[rng.randint(0, 3)]
        color = tuple([rng.randint(0, 255) for _ in range(3)])
        # Reference used a fixed 20 px buffer at 128²; scale it so small
        # test images work too.
        buffer = max(4, height // 7)
        y = rng.randint(buffer, height - buffer - 1)
        x = rng.randint(buffer, width - buffer - 1)
        s = rng.randint(buffer, max(height // 4, buffer + 1))
        return shape, color, (x, y, s)

    def _random_image(self, rng, height, width):
        bg_color = np.array([rng.randint(0, 255) for _ in range(3)])
        shapes = []
        boxes = []
        N = rng.randint(1, 4)
        for _ in range(N):
            shape, color, dims = self._random_shape(rng, height, width)
            shapes.append((shape, color, dims))
            x, y, s = dims
            boxes.append([y - s, x - s, y + s, x + s])
        # Suppress heavy overlaps (reference used NMS at 0.3 here too).
        keep_ix = non_max_suppression(
            np.array(boxes), np.arange(N)[::-1].astype(np.float64), 0.3)
        shapes = [s for i, s in enumerate(shapes) if i in keep_ix]
        return bg_color, shapes

    @staticmethod
    def _draw_shape(mask_or_img, shape, dims, value):
        """Draw filled shape into a 2D or 3D array."""
        x, y, s = dims
        H, W = mask_or_img.shape[:2]
        yy, xx = np.mgrid[0:H, 0:W]
        if shape == "square":
            sel = (np.abs(yy - y) <= s) & (np.abs(xx - x) <= s)
        elif shape == "circle":
            sel = (yy - y) ** 2 + (xx - x) ** 2 <= s ** 2
        else:  # triangle: |x - x0| <= (y - (y0 - s)) / 2, y within [y0-s, y0+s]
            sel = (yy >= y - s) & (yy <= y + s) & \
                  (np.abs(xx - x) <= (yy - (y - s)) / 2)
        mask_or_img[sel] = value
        return mask_or_img

    def load_image(self, image_id):
        info = self.image_info[image_id]
        h, w = info["height"], info["width"]
        bg_color = np.array(info["bg_color"]).reshape([1, 1, 3])
        image = np.ones([h, w, 3], dtype=np.uint8) * \
            bg_color.astype(np.uint8)
        for shape, color, dims in info["shapes"]:
            image = self._draw_shape(image, shape, dims,
                                     np.array(color, dtype=np.uint8))
        # Mild noise so the backbone sees texture.
        rng = np.random.RandomState(info["id"])
        noise = rng.randint(-10, 10, size=image.shape)
        return np.clip(image.astype(np.int32) + noise, 0, 255) \
            .astype(np.uint8)

    def load_mask(self, image_id):
        info = self.image_info[image_id]
        shapes = info["shapes"]
        h, w = info["height"], info["width"]
        count = len(shapes)
        mask = np.zeros([h, w, count], dtype=np.uint8)
        for i, (shape, _, dims) in enumerate(shapes):
            mask[:, :, i:i + 1] = self._draw_shape(
                mask[:, :, i:i + 1].copy(), shape, dims, 1)
        # Handle occlusions: later shapes occlude earlier ones.
        occlusion = np.logical_not(mask[:, :, -1]).astype(np.uint8)
        for i in range(count - 2, -1, -1):
            mask[:, :, i] = mask[:, :, i] * occlusion
            occlusion = np.logical_and(
                occlusion, np.logical_not(mask[:, :, i]))
        class_ids = np.array([self.class_names.index(s[0])
                              for s in shapes], dtype=np.int32)
        return mask.astype(bool), class_ids

    def image_reference(self, image_id):
        info = self.image_info[image_id]
        if info["source"] == "shapes":
            return info["shapes"]
        return super().image_reference(image_id)
