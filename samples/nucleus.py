#!/usr/bin/env python3
"""Nucleus segmentation sample (Kaggle Data Science Bowl 2018 layout).

Replicates reference samples/nucleus/nucleus.py (P20): pad64 resize mode,
crop-based training on large images, high instance counts, small RPN
anchors.

  python samples/nucleus.py train --dataset=/path/stage1_train
  python samples/nucleus.py detect --dataset=/path/stage1_test --model last

Dataset layout (DSB2018): <root>/<image_id>/images/<image_id>.png and
<root>/<image_id>/masks/*.png (one binary mask per instance).
"""

from __future__ import annotations

import argparse
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


class NucleusConfig(Config):
    """Reference NucleusConfig parity (samples/nucleus/nucleus.py:~40)."""
    NAME = "nucleus"
    IMAGES_PER_GPU = 6
    NUM_CLASSES = 1 + 1
    STEPS_PER_EPOCH = 200
    DETECTION_MIN_CONFIDENCE = 0
    IMAGE_RESIZE_MODE = "crop"
    IMAGE_MIN_DIM = 512
    IMAGE_MAX_DIM = 512
    IMAGE_MIN_SCALE = 2.0
    RPN_ANCHOR_SCALES = (8, 16, 32, 64, 128)
    POST_NMS_ROIS_TRAINING = 1000
    POST_NMS_ROIS_INFERENCE = 2000
    RPN_NMS_THRESHOLD = 0.9
    RPN_TRAIN_ANCHORS_PER_IMAGE = 64
    PRE_NMS_LIMIT = 6000
    USE_MINI_MASK = True
    MINI_MASK_SHAPE = (56, 56)
    TRAIN_ROIS_PER_IMAGE = 128
    MAX_GT_INSTANCES = 200
    DETECTION_MAX_INSTANCES = 400


class NucleusInferenceConfig(NucleusConfig):
    GPU_COUNT = 1
    IMAGES_PER_GPU = 1
    IMAGE_RESIZE_MODE = "pad64"
    RPN_NMS_THRESHOLD = 0.7


class NucleusDataset(Dataset):
    def load_nucleus(self, dataset_dir, subset="train"):
        self.add_class("nucleus", 1, "nucleus")
        image_ids = [d for d in os.listdir(dataset_dir)
                     if os.path.isdir(os.path.join(dataset_dir, d))]
        for image_id in sorted(image_ids):
            self.add_image(
                "nucleus", image_id=image_id,
                path=os.path.join(dataset_dir, image_id, "images",
                                  f"{image_id}.png"))

    def load_mask(self, image_id):
        from PIL import Image
        info = self.image_info[image_id]
        mask_dir = os.path.join(
            os.path.dirname(os.path.dirname(info["path"])), "masks")
        masks = []
        for f in sorted(os.listdir(mask_dir)):
            if f.endswith(".png"):
                m = np.asarray(Image.open(
                    os.path.join(mask_dir, f)).convert("1"))
                masks.append(m)
        mask = np.stack(masks, axis=-1) if masks else \
            np.empty((0, 0, 0), dtype=bool)
        return mask, np.ones([mask.shape[-1]], dtype=np.int32)

    def image_reference(self, image_id):
        info = self.image_info[image_id]
        if info["source"] == "nucleus":
            return info["id"]
        return super().image_reference(image_id)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("command", choices=["train", "detect"])
    ap.add_argument("--dataset", required=True)
    ap.add_argument("--model", default=None)
    ap.add_argument("--logs", default="logs")
    ap.add_argument("--out", default="nucleus_out")
    args = ap.parse_args()

    from maskrcnn_amd.model import MaskRCNN

    if args.command == "train":
        config = NucleusConfig()
    else:
        config = NucleusInferenceConfig()

    mode = "training" if args.command == "train" else "inference"
    model = MaskRCNN(mode, config, model_dir=args.logs)
    if args.model:
        path = model.find_last() if args.model == "last" else args.model
        model.load_weights(path, by_name=True)

    ds = NucleusDataset()
    ds.load_nucleus(args.dataset)
    ds.prepare()

    if args.command == "train":
        model.train(ds, ds, config.LEARNING_RATE, 20, "heads",
                    augmentation="fliplr")
        model.train(ds, ds, config.LEARNING_RATE, 40, "all",
                    augmentation="fliplr")
    else:
        os.makedirs(args.out, exist_ok=True)
        from maskrcnn_amd import visualize
        for image_id in ds.image_ids:
            image = ds.load_image(image_id)
            r = model.detect([image])[0]
            visualize.display_instances(
                image, r["rois"], r["masks"], r["class_ids"],
                ds.class_names, r["scores"],
                save_path=os.path.join(
                    args.out, f"{ds.image_info[image_id]['id']}.png"))
            print(ds.image_info[image_id]["id"],
                  len(r["class_ids"]), "instances")


if __name__ == "__main__":
    main()
