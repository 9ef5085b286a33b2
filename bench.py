#!/usr/bin/env python3
"""Flagship benchmark: R101-FPN Mask R-CNN bf16 training step throughput.

Measures the BASELINE.json headline metric — images/sec of ResNet-101-FPN
Mask R-CNN training (per-GPU batch 2, 1024x1024, synthetic COCO-shaped
data, random-init weights) on 1..8 MI355X GPUs (weak scaling, DP over
RCCL/xGMI).

Contract (driver): `python bench.py --gpus N --steps K --warmup W`; for
N>1 launched under torch.distributed.run with one rank per GPU. Rank 0
prints ONE JSON line with the whole-job aggregate.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
# MIOPEN_FIND_MODE must be in the environment libMIOpen starts with
# (profiles/r01_naive_conv.md); re-exec once if the shell didn't set it.
from maskrcnn_amd._rocm_env import ensure_env_or_reexec  # noqa: E402

if __name__ == "__main__":
    # re-exec only when run as a script (library imports must never
    # replace the host process; see maskrcnn_amd/_rocm_env.py)
    ensure_env_or_reexec()

import numpy as np  # noqa: E402
import torch  # noqa: E402


def build_config(args):
    from maskrcnn_amd.config import Config

    class BenchConfig(Config):
        NAME = "coco_bench"
        BACKBONE = args.backbone
        NUM_CLASSES = 81
        IMAGES_PER_GPU = args.batch
        GPU_COUNT = 1  # one process per GPU; DP handled by DDP wrapper
        IMAGE_MIN_DIM = args.image_size
        IMAGE_MAX_DIM = args.image_size
        STEPS_PER_EPOCH = args.steps
        COMPUTE_DTYPE = "bf16" if args.dtype == "bf16" else "fp32"

    return BenchConfig()


def make_synthetic_batch(config, device, seed=0):
    """One COCO-shaped synthetic training batch, fully on device.

    Random molded images + random GT; RPN targets computed once (untimed)
    with the real anchor-matching code path.
    """
    from maskrcnn_amd.data.targets import build_rpn_targets
    from maskrcnn_amd.utils.anchors import AnchorCache
    from maskrcnn_amd.utils.meta import compose_image_meta

    rng = np.random.RandomState(seed)
    B = config.BATCH_SIZE
    H, W, _ = config.IMAGE_SHAPE
    cache = AnchorCache(config)
    anchors_px = cache.get_pixel(config.IMAGE_SHAPE)

    images = rng.randn(B, 3, H, W).astype(np.float32) * 40.0
    metas, rpn_matches, rpn_bboxes = [], [], []
    gt_cls = np.zeros((B, config.MAX_GT_INSTANCES), dtype=np.int64)
    gt_boxes = np.zeros((B, config.MAX_GT_INSTANCES, 4), dtype=np.float32)
    mh, mw = config.MINI_MASK_SHAPE
    gt_masks = np.zeros((B, mh, mw, config.MAX_GT_INSTANCES),
                        dtype=np.float32)

    for b in range(B):
        # Typical COCO instance count range, capped by config.
        n = min(int(rng.randint(8, 30)), config.MAX_GT_INSTANCES)
        ys = rng.uniform(0, H - 64, n)
        xs = rng.uniform(0, W - 64, n)
        hs = rng.uniform(32, H / 3, n)
        ws = rng.uniform(32, W / 3, n)
        boxes_px = np.stack([ys, xs, np.minimum(ys + hs, H - 1),
                             np.minimum(xs + ws, W - 1)], axis=1)
        cls = rng.randint(1, config.NUM_CLASSES, n)
        rpn_match, rpn_bbox = build_rpn_targets(
            (H, W, 3), anchors_px, cls, boxes_px, config, rng=rng)
        rpn_matches.append(rpn_match[:, None].astype(np.int64))
        rpn_bboxes.append(rpn_bbox.astype(np.float32))
        gt_cls[b, :n] = cls
        gt_boxes[b, :n] = boxes_px / np.array([H - 1, W - 1, H - 1, W - 1],
                                              dtype=np.float32)
        gt_masks[b, :, :, :n] = (rng.rand(mh, mw, n) > 0.5)
        metas.append(compose_image_meta(
            b, (H, W, 3), (H, W, 3), (0, 0, H, W), 1.0,
            np.ones(config.NUM_CLASSES, np.int32)))

    t = lambda x: torch.from_numpy(np.ascontiguousarray(x)).to(device)
    return {
        "image": t(images),
        "image_meta": t(np.stack(metas).astype(np.float32)),
        "rpn_match": t(np.stack(rpn_matches)),
        "rpn_bbox": t(np.stack(rpn_bboxes)),
        "gt_class_ids": t(gt_cls),
        "gt_boxes": t(gt_boxes),
        "gt_masks": t(gt_masks),
    }


def _phase_timing(model, config, batch, anchors, meta, optimizer,
                  use_bf16, n_iters=5):
    """Per-phase wall breakdown of a training step (GPU-synced per phase;
    adds sync overhead, so the per-phase SUM exceeds the free-running
    step time — use it for ranking, not absolutes)."""
    import time as _t

    from maskrcnn_amd.model.losses import compute_losses

    def sync():
        if torch.cuda.is_available():
            torch.cuda.synchronize()

    acc = {}

    def mark(name, t0):
        sync()
        t1 = _t.perf_counter()
        acc[name] = acc.get(name, 0.0) + (t1 - t0)
        return t1

    for _ in range(n_iters):
        sync()
        t = _t.perf_counter()
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16,
                            enabled=use_bf16):
            feats = model.extract_features(batch["image"])
            t = mark("backbone+fpn", t)
            rpn_logits, rpn_class, rpn_bbox = model.rpn(feats)
            t = mark("rpn", t)
            rois_p = model.proposal_layer(rpn_class, rpn_bbox, anchors,
                                          config.POST_NMS_ROIS_TRAINING)
            t = mark("proposal(topk+nms)", t)
            rois, tci, tdel, tmask = model.detection_target_layer(
                rois_p, batch["gt_class_ids"], batch["gt_boxes"],
                batch["gt_masks"])
            t = mark("detection_targets", t)
            cl_logits, cl_probs, cl_bbox = model.classifier_head(
                rois, feats[:4])
            t = mark("classifier_head", t)
            mask_logits = model.mask_head(rois, feats[:4])
            t = mark("mask_head", t)
            losses = compute_losses(
                config,
                {"rpn_match": batch["rpn_match"],
                 "rpn_bbox": batch["rpn_bbox"],
                 "active_class_ids": meta["active_class_ids"]},
                {"rpn_class_logits": rpn_logits, "rpn_bbox": rpn_bbox,
                 "target_class_ids": tci, "target_deltas": tdel,
                 "target_masks": tmask, "mrcnn_class_logits": cl_logits,
                 "mrcnn_bbox": cl_bbox, "mrcnn_mask": mask_logits})
            t = mark("losses", t)
        optimizer.zero_grad(set_to_none=True)
        losses["total_loss"].backward()
        t = mark("backward", t)
        torch.nn.utils.clip_grad_norm_(
            (p for p in model.parameters() if p.requires_grad),
            config.GRADIENT_CLIP_NORM)
        optimizer.step()
        t = mark("clip+opt_step", t)
    total = sum(acc.values())
    print("=== phase timing (avg ms over "
          f"{n_iters} iters; per-phase synced) ===", file=sys.stderr)
    for k, v in sorted(acc.items(), key=lambda kv: -kv[1]):
        print(f"  {k:22s} {1000 * v / n_iters:8.2f} ms "
              f"({100 * v / total:5.1f}%)", file=sys.stderr)
    print(f"  {'TOTAL':22s} {1000 * total / n_iters:8.2f} ms",
          file=sys.stderr)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--batch", type=int, default=2)
    ap.add_argument("--image-size", type=int, default=1024)
    ap.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    ap.add_argument("--backbone", default="resnet101",
                    choices=["resnet101", "resnet50"],
                    help="resnet50 only for the CPU/CI smoke path; the "
                         "headline config is resnet101 (BASELINE.json)")
    ap.add_argument("--backend", default=None,
                    help="override dist backend (nccl|gloo)")
    ap.add_argument("--phase-timing", action="store_true",
                    help="print a per-phase GPU/wall breakdown to stderr")
    ap.add_argument("--torch-profile", metavar="DIR", default=None,
                    help="capture a kineto (torch.profiler) chrome trace "
                         "of 3 steady steps into DIR")
    ap.add_argument("--data", default="synthetic",
                    choices=["synthetic", "loader"],
                    help="synthetic = one cached on-device batch "
                         "(BASELINE contract: pure step throughput); "
                         "loader = real DataLoader pipeline with "
                         "per-step HtoD inside the timed region")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    distributed = world > 1

    use_cuda = torch.cuda.is_available()
    device = torch.device(f"cuda:{local_rank}" if use_cuda else "cpu")
    if use_cuda:
        torch.cuda.set_device(device)
        from maskrcnn_amd._rocm_env import setup_rocm_perf
        setup_rocm_perf(local_rank)

    if distributed:
        import torch.distributed as dist
        backend = args.backend or ("nccl" if use_cuda else "gloo")
        dist.init_process_group(backend=backend)

    config = build_config(args)

    from maskrcnn_amd.model.mask_rcnn import MaskRCNNModel
    from maskrcnn_amd.model.losses import compute_losses
    from maskrcnn_amd.engine.trainer import build_optimizer
    from maskrcnn_amd.utils.anchors import AnchorCache
    from maskrcnn_amd.utils.meta import parse_image_meta_t

    torch.manual_seed(1234 + rank)
    model = MaskRCNNModel(config).to(device)
    if config.CHANNELS_LAST and use_cuda:
        model = model.to(memory_format=torch.channels_last)
    model.train()

    if distributed:
        from maskrcnn_amd.distributed import DistributedTrainer
        ddp = DistributedTrainer(model, config)
    else:
        ddp = None

    optimizer = build_optimizer(model, config, config.LEARNING_RATE)
    anchors = AnchorCache(config).get(config.IMAGE_SHAPE, device=device)
    batch = make_synthetic_batch(config, device, seed=100 + rank)
    if config.CHANNELS_LAST and use_cuda:
        batch["image"] = batch["image"].to(
            memory_format=torch.channels_last)
    meta = parse_image_meta_t(batch["image_meta"])

    use_bf16 = (config.COMPUTE_DTYPE == "bf16" and use_cuda)

    loader_iter = None
    if args.data == "loader":
        # End-to-end variant: shapes drawn by worker processes, collated
        # batches HtoD-copied inside the timed step.
        from maskrcnn_amd.data.loader import make_data_loader
        from maskrcnn_amd.data.synthetic import ShapesDataset
        ds = ShapesDataset()
        ds.load_shapes(64, args.image_size, args.image_size,
                       seed=7 + rank)
        # Pad the class table to the benched NUM_CLASSES (COCO-shaped
        # meta: active_class_ids length must equal config.NUM_CLASSES).
        for ci in range(4, config.NUM_CLASSES):
            ds.add_class("shapes", ci, f"pad{ci}")
        ds.prepare()
        loader = make_data_loader(ds, config,
                                  batch_size=config.BATCH_SIZE,
                                  num_workers=2, seed=rank)

        def _cycle():
            while True:
                for b in loader:
                    yield b
        loader_iter = _cycle()

    def one_step():
        nonlocal batch, meta
        if loader_iter is not None:
            host = next(loader_iter)
            batch = {k: v.to(device, non_blocking=True)
                     for k, v in host.items()}
            if config.CHANNELS_LAST and use_cuda:
                batch["image"] = batch["image"].to(
                    memory_format=torch.channels_last)
            meta = parse_image_meta_t(batch["image_meta"])
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16,
                            enabled=use_bf16):
            outputs = model(batch["image"], batch["image_meta"], anchors,
                            gt_class_ids=batch["gt_class_ids"],
                            gt_boxes=batch["gt_boxes"],
                            gt_masks=batch["gt_masks"], mode="training")
            losses = compute_losses(
                config,
                {"rpn_match": batch["rpn_match"],
                 "rpn_bbox": batch["rpn_bbox"],
                 "active_class_ids": meta["active_class_ids"]},
                outputs)
        optimizer.zero_grad(set_to_none=True)
        losses["total_loss"].backward()
        if ddp is not None:
            ddp.finalize_backward()
        torch.nn.utils.clip_grad_norm_(
            (p for p in model.parameters() if p.requires_grad),
            config.GRADIENT_CLIP_NORM)
        optimizer.step()
        return losses

    for _ in range(args.warmup):
        one_step()

    if args.phase_timing and rank == 0:
        _phase_timing(model, config, batch, anchors, meta, optimizer,
                      use_bf16, n_iters=5)

    if args.torch_profile and rank == 0:
        from torch.profiler import ProfilerActivity, profile
        activities = [ProfilerActivity.CPU]
        if use_cuda:
            activities.append(ProfilerActivity.CUDA)
        with profile(activities=activities, record_shapes=False,
                     with_stack=False) as prof:
            for _ in range(3):
                one_step()
            if use_cuda:
                torch.cuda.synchronize()
        os.makedirs(args.torch_profile, exist_ok=True)
        trace = os.path.join(args.torch_profile, "bench_trace.json")
        prof.export_chrome_trace(trace)
        print(f"torch.profiler trace -> {trace}", file=sys.stderr)
        print(prof.key_averages().table(
            sort_by="cuda_time_total" if use_cuda else "cpu_time_total",
            row_limit=25), file=sys.stderr)

    if distributed:
        import torch.distributed as dist
        dist.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step()
    if use_cuda:
        torch.cuda.synchronize()
    if distributed:
        import torch.distributed as dist
        dist.barrier()
    elapsed = time.perf_counter() - t0

    # Max over ranks = whole-job time.
    if distributed:
        import torch.distributed as dist
        t = torch.tensor([elapsed], device=device if use_cuda else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    ms_per_step = elapsed / args.steps * 1000.0
    n_gpus = world if distributed else 1
    global_batch = config.BATCH_SIZE * n_gpus
    images_per_sec = global_batch * args.steps / elapsed

    if rank == 0:
        print(json.dumps({
            "metric": "train_images_per_sec",
            "value": round(images_per_sec, 3),
            "unit": "images/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype if use_cuda else "fp32",
            "data": ("synthetic" if args.data == "synthetic"
                     else "synthetic-loader-e2e"),
            "config": {
                "model": f"{config.BACKBONE}-FPN maskrcnn",
                "global_batch": global_batch,
                "image_size": args.image_size,
                "num_classes": config.NUM_CLASSES,
                "parallelism": f"dp{n_gpus}",
            },
        }))

    if distributed:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
