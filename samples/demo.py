#!/usr/bin/env python3
"""Detection demo (reference samples/demo.ipynb as a script): load a
checkpoint (or random weights), detect on an image (or a generated shapes
image), save a rendered visualization.

  python samples/demo.py --image photo.jpg --weights ckpt.pth
  python samples/demo.py            # synthetic shapes image, random init
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


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--image", default=None)
    ap.add_argument("--weights", default=None)
    ap.add_argument("--out", default="demo_out.png")
    args = ap.parse_args()

    from maskrcnn_amd import visualize
    from maskrcnn_amd.data.synthetic import ShapesConfig, ShapesDataset
    from maskrcnn_amd.model import MaskRCNN

    class DemoConfig(ShapesConfig):
        IMAGES_PER_GPU = 1
        DETECTION_MIN_CONFIDENCE = 0.0

    config = DemoConfig()
    model = MaskRCNN("inference", config, model_dir="logs")
    if args.weights:
        model.load_weights(args.weights, by_name=True)

    if args.image:
        from PIL import Image
        image = np.asarray(Image.open(args.image).convert("RGB"))
        class_names = [f"class{i}" for i in range(config.NUM_CLASSES)]
    else:
        ds = ShapesDataset()
        ds.load_shapes(1, 128, 128, seed=42)
        ds.prepare()
        image = ds.load_image(0)
        class_names = ds.class_names

    r = model.detect([image])[0]
    visualize.display_instances(image, r["rois"], r["masks"],
                                r["class_ids"], class_names, r["scores"],
                                save_path=args.out)
    print(f"{len(r['class_ids'])} detections -> {args.out}")


if __name__ == "__main__":
    main()
