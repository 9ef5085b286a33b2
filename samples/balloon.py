#!/usr/bin/env python3
"""Balloon sample: single-class fine-tune + color-splash inference.

Replicates reference samples/balloon/balloon.py (P20): a VIA-format
(VGG Image Annotator) polygon dataset with one class, plus the
color_splash effect demo.

  python samples/balloon.py train --dataset=/path/balloon --model=last
  python samples/balloon.py splash --model=last --image=photo.jpg
"""

from __future__ import annotations

import argparse
import json
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from maskrcnn_amd._rocm_env import ensure_env_or_reexec  # noqa: E402

if __name__ == "__main__":
    # re-exec only when run as a script (library imports must never
    # replace the host process; see maskrcnn_amd/_rocm_env.py)
    ensure_env_or_reexec()

from maskrcnn_amd.config import Config  # noqa: E402
from maskrcnn_amd.data.dataset import Dataset  # noqa: E402


class BalloonConfig(Config):
    NAME = "balloon"
    IMAGES_PER_GPU = 2
    NUM_CLASSES = 1 + 1
    STEPS_PER_EPOCH = 100
    DETECTION_MIN_CONFIDENCE = 0.9


class BalloonDataset(Dataset):
    def load_balloon(self, dataset_dir, subset):
        """Load the VIA-annotated balloon dataset (reference parity)."""
        self.add_class("balloon", 1, "balloon")
        assert subset in ("train", "val")
        dataset_dir = os.path.join(dataset_dir, subset)
        with open(os.path.join(dataset_dir,
                               "via_region_data.json")) as f:
            annotations = json.load(f)
        annotations = [a for a in annotations.values() if a["regions"]]
        for a in annotations:
            regions = a["regions"]
            if isinstance(regions, dict):
                regions = list(regions.values())
            polygons = [r["shape_attributes"] for r in regions]
            image_path = os.path.join(dataset_dir, a["filename"])
            from PIL import Image
            with Image.open(image_path) as im:
                width, height = im.size
            self.add_image("balloon", image_id=a["filename"],
                           path=image_path, width=width, height=height,
                           polygons=polygons)

    def load_mask(self, image_id):
        info = self.image_info[image_id]
        if info["source"] != "balloon":
            return super().load_mask(image_id)
        from PIL import Image, ImageDraw
        mask = np.zeros([info["height"], info["width"],
                         len(info["polygons"])], dtype=bool)
        for i, p in enumerate(info["polygons"]):
            img = Image.new("1", (info["width"], info["height"]), 0)
            ImageDraw.Draw(img).polygon(
                list(zip(p["all_points_x"], p["all_points_y"])),
                outline=1, fill=1)
            mask[:, :, i] = np.asarray(img, dtype=bool)
        return mask, np.ones([mask.shape[-1]], dtype=np.int32)

    def image_reference(self, image_id):
        info = self.image_info[image_id]
        if info["source"] == "balloon":
            return info["path"]
        return super().image_reference(image_id)


def color_splash(image, mask):
    """Gray image with color kept where masks are (reference parity)."""
    gray = np.sum(image * np.array([0.299, 0.587, 0.114]),
                  -1, keepdims=True).astype(np.uint8)
    gray = np.repeat(gray, 3, axis=2)
    if mask.shape[-1] > 0:
        mask = (np.sum(mask, -1, keepdims=True) >= 1)
        splash = np.where(mask, image, gray).astype(np.uint8)
    else:
        splash = gray.astype(np.uint8)
    return splash


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("command", choices=["train", "splash"])
    ap.add_argument("--dataset")
    ap.add_argument("--model", default=None)
    ap.add_argument("--logs", default="logs")
    ap.add_argument("--image")
    ap.add_argument("--out", default="splash.png")
    args = ap.parse_args()

    from maskrcnn_amd.model import MaskRCNN

    if args.command == "train":
        config = BalloonConfig()
    else:
        class InferenceConfig(BalloonConfig):
            GPU_COUNT = 1
            IMAGES_PER_GPU = 1
        config = InferenceConfig()

    mode = "training" if args.command == "train" else "inference"
    model = MaskRCNN(mode, config, model_dir=args.logs)
    if args.model:
        path = model.find_last() if args.model == "last" else args.model
        model.load_weights(path, by_name=True)

    if args.command == "train":
        ds_train = BalloonDataset()
        ds_train.load_balloon(args.dataset, "train")
        ds_train.prepare()
        ds_val = BalloonDataset()
        ds_val.load_balloon(args.dataset, "val")
        ds_val.prepare()
        model.train(ds_train, ds_val, config.LEARNING_RATE, 30, "heads",
                    augmentation="fliplr")
    else:
        from PIL import Image
        image = np.asarray(Image.open(args.image).convert("RGB"))
        r = model.detect([image])[0]
        splash = color_splash(image, r["masks"])
        Image.fromarray(splash).save(args.out)
        print("Saved to", args.out)


if __name__ == "__main__":
    main()
