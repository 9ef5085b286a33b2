#!/usr/bin/env python3
"""Export a training checkpoint as a SavedModel-layout serving directory.

Replicates reference S1 (export_saved_model.py-style script):

  python tools/export_saved_model.py --checkpoint ckpt.pth \
      --export-dir /models/mask_rcnn --version 1 --num-classes 81
"""

from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--checkpoint", required=True)
    ap.add_argument("--export-dir", required=True)
    ap.add_argument("--version", type=int, default=1)
    ap.add_argument("--num-classes", type=int, default=81)
    ap.add_argument("--backbone", default="resnet101")
    ap.add_argument("--image-size", type=int, default=1024)
    ap.add_argument("--weight-dtype", default="fp32",
                    choices=["fp32", "fp8"])
    ap.add_argument("--warmup-batches", type=int, nargs="*",
                    default=None,
                    help="write assets.extra/tf_serving_warmup_requests"
                         " for these batch sizes (e.g. --warmup-batches"
                         " 32); the server replays them at load")
    args = ap.parse_args()

    from maskrcnn_amd.config import Config
    from maskrcnn_amd.model import MaskRCNN
    from maskrcnn_amd.serving.savedmodel import export_saved_model

    class ExportConfig(Config):
        NAME = "export"
        BACKBONE = args.backbone
        NUM_CLASSES = args.num_classes
        IMAGES_PER_GPU = 1
        IMAGE_MIN_DIM = args.image_size
        IMAGE_MAX_DIM = args.image_size

    cfg = ExportConfig()
    model = MaskRCNN("inference", cfg, model_dir="/tmp")
    model.load_weights(args.checkpoint, by_name=True)
    out = export_saved_model(model, cfg, args.export_dir,
                             version=args.version,
                             extra_meta={"checkpoint": args.checkpoint},
                             weight_dtype=args.weight_dtype,
                             warmup_batches=args.warmup_batches)
    print("exported", out)


if __name__ == "__main__":
    main()
