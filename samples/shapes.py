#!/usr/bin/env python3
"""Shapes sample — the hermetic end-to-end train/detect demo.

Replicates the role of reference samples/shapes + train_shapes.ipynb
(SURVEY.md P20/§4): train on procedurally-generated shapes, then detect
and report mAP, with no downloads.

  python samples/shapes.py train --epochs 2 --steps 50
  python samples/shapes.py evaluate --model last
"""

from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from maskrcnn_amd._rocm_env import ensure_env_or_reexec  # noqa: E402

if __name__ == "__main__":
    # re-exec only when run as a script (library imports must never
    # replace the host process; see maskrcnn_amd/_rocm_env.py)
    ensure_env_or_reexec()


def main():
    ap = argparse.ArgumentParser(description="Mask R-CNN on shapes")
    ap.add_argument("command", choices=["train", "evaluate", "detect"])
    ap.add_argument("--model", default=None,
                    help="'last' or a checkpoint path")
    ap.add_argument("--logs", default="logs")
    ap.add_argument("--epochs", type=int, default=2)
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument("--layers", default="all")
    ap.add_argument("--limit", type=int, default=20)
    ap.add_argument("--out", default="/tmp/shapes_detect.png")
    args = ap.parse_args()

    from maskrcnn_amd.data.synthetic import ShapesConfig, ShapesDataset
    from maskrcnn_amd.model import MaskRCNN

    config = ShapesConfig()
    config.STEPS_PER_EPOCH = args.steps
    config.display()

    ds_train = ShapesDataset()
    ds_train.load_shapes(200, 128, 128, seed=1)
    ds_train.prepare()
    ds_val = ShapesDataset()
    ds_val.load_shapes(40, 128, 128, seed=2)
    ds_val.prepare()

    if args.command == "train":
        model = MaskRCNN("training", config, model_dir=args.logs)
        if args.model:
            path = model.find_last() if args.model == "last" else args.model
            model.load_weights(path)
        model.train(ds_train, ds_val, config.LEARNING_RATE, args.epochs,
                    args.layers, augmentation="fliplr")
    else:
        class InferenceConfig(ShapesConfig):
            IMAGES_PER_GPU = 1
        model = MaskRCNN("inference", InferenceConfig(),
                         model_dir=args.logs)
        if args.model:
            path = model.find_last() if args.model == "last" else args.model
            model.load_weights(path)
        if args.command == "evaluate":
            from maskrcnn_amd.engine.evaluator import evaluate_dataset
            evaluate_dataset(model, ds_val, limit=args.limit)
        else:
            import numpy as np
            from maskrcnn_amd import visualize
            image_id = int(np.random.choice(ds_val.image_ids))
            image = ds_val.load_image(image_id)
            r = model.detect([image])[0]
            visualize.display_instances(
                image, r["rois"], r["masks"], r["class_ids"],
                ds_val.class_names, r["scores"], save_path=args.out)
            print(f"saved {args.out} with {len(r['class_ids'])} detections")


if __name__ == "__main__":
    main()
