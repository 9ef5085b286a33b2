#!/usr/bin/env python3
"""Model inspection CLI — the reference's inspect_model / inspect_weights
notebooks as scripts.

  python tools/inspect_model.py graph --names rois mrcnn_detection fpn
  python tools/inspect_model.py weights [--checkpoint ckpt.pth]
  python tools/inspect_model.py data   # anchors / RPN-target stats
"""

from __future__ import annotations

import argparse
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("command", choices=["graph", "weights", "data"])
    ap.add_argument("--checkpoint")
    ap.add_argument("--names", nargs="+",
                    default=["rois", "mrcnn_detection", "rpn_class"])
    args = ap.parse_args()

    from maskrcnn_amd.data.synthetic import ShapesConfig, ShapesDataset

    class Cfg(ShapesConfig):
        IMAGES_PER_GPU = 1
        DETECTION_MIN_CONFIDENCE = 0.0

    cfg = Cfg()
    ds = ShapesDataset()
    ds.load_shapes(4, 128, 128, seed=3)
    ds.prepare()

    if args.command == "data":
        # inspect_data parity: anchors + RPN target matching stats.
        from maskrcnn_amd.data.targets import build_rpn_targets, load_image_gt
        from maskrcnn_amd.utils.anchors import AnchorCache
        anchors = AnchorCache(cfg).get_pixel(cfg.IMAGE_SHAPE)
        print(f"anchors: {anchors.shape[0]} "
              f"(scales {cfg.RPN_ANCHOR_SCALES}, "
              f"ratios {cfg.RPN_ANCHOR_RATIOS})")
        image, meta, cls, bbox, mask = load_image_gt(ds, cfg, 0)
        rpn_match, rpn_bbox = build_rpn_targets(
            image.shape, anchors, cls, bbox, cfg)
        print(f"image 0: {len(cls)} instances, "
              f"rpn +{int((rpn_match == 1).sum())} "
              f"-{int((rpn_match == -1).sum())} anchors, "
              f"mini-masks {mask.shape}")
        return

    from maskrcnn_amd.model import MaskRCNN
    model = MaskRCNN("inference", cfg, model_dir="/tmp/inspect_logs")
    if args.checkpoint:
        model.load_weights(args.checkpoint, by_name=True)

    if args.command == "weights":
        from maskrcnn_amd.visualize import display_weight_stats
        display_weight_stats(model.model)
        return

    image = ds.load_image(0)
    out = model.run_graph([image], args.names)
    for name, v in out.items():
        if isinstance(v, list):
            print(f"{name}: list of {len(v)}: "
                  + ", ".join(str(np.asarray(x).shape) for x in v))
        else:
            v = np.asarray(v)
            print(f"{name}: shape {v.shape}  min {v.min():.4f}  "
                  f"max {v.max():.4f}  mean {v.mean():.4f}")


if __name__ == "__main__":
    main()
