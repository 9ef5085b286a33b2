#!/usr/bin/env python3
"""Checkpoint format converter.

Any-to-any between the formats the stack touches:
  .pth           torch checkpoint ({"model": state_dict, ...} or raw)
  .h5            matterport Keras weights (read-only; needs h5py)
  SavedModel dir TF-Serving layout (<base>/<version>/saved_model.pb)

  python tools/convert_checkpoint.py --src ckpt.h5 --dst ckpt.pth \
      --num-classes 81 [--backbone resnet101]
  python tools/convert_checkpoint.py --src ckpt.pth --dst /models/m \
      --num-classes 81 --version 1          # -> SavedModel
  python tools/convert_checkpoint.py --src /models/m --dst out.pth
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))


def build_model(args, config=None):
    from maskrcnn_amd.config import Config
    from maskrcnn_amd.model.mask_rcnn import MaskRCNNModel

    if config is None:
        class ConvConfig(Config):
            NAME = "convert"
            NUM_CLASSES = args.num_classes
            BACKBONE = args.backbone
            IMAGES_PER_GPU = 1
        config = ConvConfig()
    return MaskRCNNModel(config), config


def load_any(args):
    """-> (model, config) with weights loaded from --src."""
    import torch
    src = args.src
    if os.path.isdir(src):
        from maskrcnn_amd.serving.engine import InferenceEngine
        from maskrcnn_amd.serving.savedmodel import find_latest_version
        version = args.version or find_latest_version(src)
        eng = InferenceEngine.from_saved_model(
            os.path.join(src, str(version)), device="cpu")
        return eng.model, eng.config
    model, config = build_model(args)
    if src.endswith(".h5"):
        from maskrcnn_amd.utils.keras_import import load_keras_h5
        loaded, missing, unexpected = load_keras_h5(model, src)
        print(f"h5: {len(loaded)} tensors loaded, {len(missing)} "
              f"unmatched in file, {len(unexpected)} unknown")
    else:
        state = torch.load(src, map_location="cpu", weights_only=False)
        sd = state["model"] if isinstance(state, dict) and "model" in state \
            else state
        model.load_state_dict(sd)
    return model, config


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--src", required=True)
    ap.add_argument("--dst", required=True)
    ap.add_argument("--num-classes", type=int, default=81)
    ap.add_argument("--backbone", default="resnet101",
                    choices=["resnet50", "resnet101"])
    ap.add_argument("--version", type=int, default=None,
                    help="SavedModel version (src: which to read; "
                         "dst: which to write, default 1)")
    ap.add_argument("--weight-dtype", default="fp32",
                    choices=["fp32", "fp8"],
                    help="SavedModel export weight storage")
    args = ap.parse_args()

    import torch
    model, config = load_any(args)
    if args.dst.endswith(".pth"):
        torch.save({"model": model.state_dict(),
                    "config": config.to_dict()}, args.dst)
        print(f"wrote {args.dst}")
    else:
        from maskrcnn_amd.serving.savedmodel import export_saved_model
        path = export_saved_model(model, config, args.dst,
                                  version=args.version or 1,
                                  weight_dtype=args.weight_dtype)
        print(f"wrote {path}")


if __name__ == "__main__":
    main()
