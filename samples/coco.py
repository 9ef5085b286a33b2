#!/usr/bin/env python3
"""COCO training / evaluation CLI.

Replicates reference samples/coco/coco.py main() (P19):

  python samples/coco.py train --dataset=/path/to/coco --model=last
  python samples/coco.py evaluate --dataset=/path/to/coco --model=last

Training schedule parity: heads (40 epochs @ lr) -> 4+ (120 @ lr) ->
all (160 @ lr/10), horizontal-flip augmentation. auto_download is not
available in this offline build.
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

from maskrcnn_amd.config import Config  # noqa: E402

DEFAULT_LOGS_DIR = "logs"


class CocoConfig(Config):
    """Reference CocoConfig parity (samples/coco/coco.py:~60)."""
    NAME = "coco"
    IMAGES_PER_GPU = 2
    NUM_CLASSES = 1 + 80


def main():
    ap = argparse.ArgumentParser(description="Train Mask R-CNN on COCO")
    ap.add_argument("command", choices=["train", "evaluate"])
    ap.add_argument("--dataset", required=True,
                    help="COCO dataset root directory")
    ap.add_argument("--year", default="2017", choices=["2014", "2017"])
    ap.add_argument("--model", default=None,
                    help="'last', 'imagenet', or a checkpoint path")
    ap.add_argument("--logs", default=DEFAULT_LOGS_DIR)
    ap.add_argument("--limit", type=int, default=500,
                    help="images to use for evaluation")
    ap.add_argument("--download", default=False,
                    help="accepted for CLI parity; offline build cannot "
                         "download")
    args = ap.parse_args()

    from maskrcnn_amd.data.coco import CocoDataset
    from maskrcnn_amd.model import MaskRCNN

    if args.command == "train":
        config = CocoConfig()
    else:
        class InferenceConfig(CocoConfig):
            GPU_COUNT = 1
            IMAGES_PER_GPU = 1
            DETECTION_MIN_CONFIDENCE = 0
        config = InferenceConfig()
    config.display()

    mode = "training" if args.command == "train" else "inference"
    model = MaskRCNN(mode, config, model_dir=args.logs)

    if args.model:
        if args.model == "last":
            model_path = model.find_last()
        elif args.model == "imagenet":
            raise SystemExit("imagenet bootstrap weights are not "
                             "available in this offline build")
        else:
            model_path = args.model
        print("Loading weights", model_path)
        model.load_weights(model_path, by_name=True)

    if args.command == "train":
        dataset_train = CocoDataset()
        dataset_train.load_coco(args.dataset, "train", year=args.year)
        if args.year == "2014":
            dataset_train.load_coco(args.dataset, "valminusminival",
                                    year=args.year)
        dataset_train.prepare()
        dataset_val = CocoDataset()
        val_type = "val" if args.year == "2017" else "minival"
        dataset_val.load_coco(args.dataset, val_type, year=args.year)
        dataset_val.prepare()

        # Reference schedule (coco.py main, :~470).
        print("Training network heads")
        model.train(dataset_train, dataset_val, config.LEARNING_RATE, 40,
                    "heads", augmentation="fliplr")
        print("Fine tune Resnet stage 4 and up")
        model.train(dataset_train, dataset_val, config.LEARNING_RATE, 120,
                    "4+", augmentation="fliplr")
        print("Fine tune all layers")
        model.train(dataset_train, dataset_val, config.LEARNING_RATE / 10,
                    160, "all", augmentation="fliplr")
    else:
        dataset_val = CocoDataset()
        val_type = "val" if args.year == "2017" else "minival"
        dataset_val.load_coco(args.dataset, val_type, year=args.year)
        dataset_val.prepare()
        print(f"Running COCO evaluation on {args.limit} images.")
        from maskrcnn_amd.engine.evaluator import evaluate_dataset
        evaluate_dataset(model, dataset_val, limit=args.limit)


if __name__ == "__main__":
    main()
