#!/usr/bin/env python3
"""Distributed training CLI (torchrun-able).

  # single GPU
  python tools/train.py --dataset shapes --epochs 2

  # 8-GPU DP over RCCL/xGMI
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 tools/train.py --dataset shapes --epochs 2

Resume: --model last. COCO: --dataset coco --coco-root /path/to/coco.
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
    ap = argparse.ArgumentParser()
    ap.add_argument("--dataset", default="shapes",
                    choices=["shapes", "coco"])
    ap.add_argument("--coco-root")
    ap.add_argument("--year", default="2017")
    ap.add_argument("--model", default=None)
    ap.add_argument("--logs", default="logs")
    ap.add_argument("--epochs", type=int, default=2)
    ap.add_argument("--steps", type=int, default=None)
    ap.add_argument("--layers", default="all")
    ap.add_argument("--lr", type=float, default=None)
    ap.add_argument("--num-workers", type=int, default=4)
    args = ap.parse_args()

    import torch

    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if world > 1:
        import torch.distributed as dist
        backend = "nccl" if torch.cuda.is_available() else "gloo"
        if torch.cuda.is_available():
            torch.cuda.set_device(local_rank)
        dist.init_process_group(backend=backend)

    if args.dataset == "shapes":
        from maskrcnn_amd.data.synthetic import ShapesConfig, ShapesDataset
        config = ShapesConfig()
        ds_train = ShapesDataset()
        ds_train.load_shapes(400, 128, 128, seed=1)
        ds_train.prepare()
        ds_val = ShapesDataset()
        ds_val.load_shapes(50, 128, 128, seed=2)
        ds_val.prepare()
    else:
        from maskrcnn_amd.data.coco import CocoDataset
        from samples.coco import CocoConfig
        config = CocoConfig()
        ds_train = CocoDataset()
        ds_train.load_coco(args.coco_root, "train", year=args.year)
        ds_train.prepare()
        ds_val = CocoDataset()
        ds_val.load_coco(args.coco_root, "val", year=args.year)
        ds_val.prepare()

    if args.steps:
        config.STEPS_PER_EPOCH = args.steps

    from maskrcnn_amd.engine.trainer import Trainer
    from maskrcnn_amd.model import MaskRCNN

    model = MaskRCNN("training", config, model_dir=args.logs)
    if args.model:
        path = model.find_last() if args.model == "last" else args.model
        model.load_weights(path, by_name=True)

    trainer = Trainer(model, ds_train, ds_val, augmentation="fliplr",
                      num_workers=args.num_workers)
    trainer.fit(args.lr or config.LEARNING_RATE, args.epochs, args.layers)

    if world > 1:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
