#!/usr/bin/env python3
"""Standalone DDP worker for tests/test_distributed.py (run as a
subprocess; prints one JSON result line)."""

import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))


def run_linear(rank, world):
    import numpy as np
    import torch
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(100 + rank)
    model = torch.nn.Sequential(
        torch.nn.Linear(16, 64), torch.nn.ReLU(),
        torch.nn.Linear(64, 64), torch.nn.ReLU(),
        torch.nn.Linear(64, 4))
    from maskrcnn_amd.distributed import DistributedTrainer
    ddp = DistributedTrainer(model, bucket_mb=0.01)
    w0 = model[0].weight.detach().clone()
    torch.manual_seed(500 + rank)
    x = torch.randn(8, 16)
    y = torch.randn(8, 4)
    loss = torch.nn.functional.mse_loss(model(x), y)
    loss.backward()
    ddp.finalize_backward()
    grads = torch.cat([p.grad.flatten() for p in model.parameters()])
    out = {"rank": rank,
           "w0_sum": float(np.abs(w0.numpy()).sum()),
           "grad_sum": float(grads.abs().sum()),
           "grad_head": grads[:8].tolist()}
    dist.destroy_process_group()
    return out


def run_maskrcnn(rank, world):
    import torch
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    from maskrcnn_amd.data.synthetic import ShapesConfig
    from maskrcnn_amd.distributed import DistributedTrainer
    from maskrcnn_amd.model.mask_rcnn import MaskRCNNModel
    from maskrcnn_amd.model.losses import compute_losses
    from maskrcnn_amd.utils.anchors import AnchorCache
    from maskrcnn_amd.utils.meta import parse_image_meta_t

    class C(ShapesConfig):
        IMAGES_PER_GPU = 1
        PRE_NMS_LIMIT = 200
        POST_NMS_ROIS_TRAINING = 50
        TRAIN_ROIS_PER_IMAGE = 16
        BACKBONE = "resnet50"

    cfg = C()
    torch.manual_seed(rank)
    model = MaskRCNNModel(cfg)
    ddp = DistributedTrainer(model, cfg)

    import bench
    batch = bench.make_synthetic_batch(cfg, torch.device("cpu"),
                                       seed=rank)
    anchors = AnchorCache(cfg).get(cfg.IMAGE_SHAPE)
    meta = parse_image_meta_t(batch["image_meta"])
    out = model(batch["image"], batch["image_meta"], anchors,
                gt_class_ids=batch["gt_class_ids"],
                gt_boxes=batch["gt_boxes"],
                gt_masks=batch["gt_masks"], mode="training")
    losses = compute_losses(
        cfg, {"rpn_match": batch["rpn_match"],
              "rpn_bbox": batch["rpn_bbox"],
              "active_class_ids": meta["active_class_ids"]}, out)
    losses["total_loss"].backward()
    ddp.finalize_backward()
    g = torch.cat([p.grad.flatten() for p in model.parameters()
                   if p.requires_grad])
    result = {"rank": rank, "grad_sum": float(g.abs().sum()),
              "loss": float(losses["total_loss"])}
    dist.destroy_process_group()
    return result


def run_staged(rank, world):
    """Matterport's canonical staged schedule under DDP: fit('heads')
    then fit('all') in one process — exercises _rebuild_ddp()
    (remove_hooks + re-bucket + re-broadcast). Double-registered hooks
    would double-count bucket.pending and hang/raise in
    finalize_backward."""
    import tempfile

    import torch
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    from maskrcnn_amd.data.synthetic import ShapesConfig, ShapesDataset
    from maskrcnn_amd.engine.trainer import Trainer
    from maskrcnn_amd.model import MaskRCNN

    class C(ShapesConfig):
        IMAGES_PER_GPU = 1
        PRE_NMS_LIMIT = 200
        POST_NMS_ROIS_TRAINING = 50
        TRAIN_ROIS_PER_IMAGE = 16
        BACKBONE = "resnet50"

    cfg = C()
    ds = ShapesDataset()
    ds.load_shapes(8, 128, 128, seed=7)
    ds.prepare()
    logs = tempfile.mkdtemp(prefix=f"staged_r{rank}_")
    m = MaskRCNN("training", cfg, model_dir=logs)
    tr = Trainer(m, ds)
    tr.fit(cfg.LEARNING_RATE, epochs=1, layers="heads",
           steps_per_epoch=2)
    tr.fit(cfg.LEARNING_RATE / 10, epochs=2, layers="all",
           steps_per_epoch=2)
    # Weights must be identical across ranks after synced training.
    w = torch.cat([p.detach().flatten()
                   for p in m.model.parameters()])[:8]
    result = {"rank": rank, "epochs": m.epoch,
              "w_head": [round(float(v), 6) for v in w],
              "losses_finite": all(
                  all(v == v for v in h.values()) for h in tr.history)}
    dist.destroy_process_group()
    return result


def run_accum(rank, world):
    """no_sync() gradient accumulation: 2 local microbatches per rank,
    one all-reduce. Verifies against an in-process replay of EVERY
    rank's microbatches (seeds are deterministic) through a copy of the
    broadcast model."""
    import copy

    import torch
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(100 + rank)
    model = torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.ReLU(),
        torch.nn.Linear(32, 4))
    from maskrcnn_amd.distributed import DistributedTrainer
    ddp = DistributedTrainer(model, bucket_mb=0.01)
    ref = copy.deepcopy(model)  # post-broadcast weights

    def batch(r, step):
        g = torch.Generator().manual_seed(1000 + 10 * r + step)
        return (torch.randn(4, 16, generator=g),
                torch.randn(4, 4, generator=g))

    # Microbatch 0 under no_sync (accumulate), microbatch 1 synced.
    with ddp.no_sync():
        x, y = batch(rank, 0)
        torch.nn.functional.mse_loss(model(x), y).backward()
    x, y = batch(rank, 1)
    torch.nn.functional.mse_loss(model(x), y).backward()
    ddp.finalize_backward()

    # Expected: mean over ranks of (sum of that rank's microbatch grads).
    for r in range(world):
        for s in range(2):
            x, y = batch(r, s)
            torch.nn.functional.mse_loss(ref(x), y).backward()
    ok = True
    for p_m, p_r in zip(model.parameters(), ref.parameters()):
        if not torch.allclose(p_m.grad, p_r.grad / world, atol=1e-5):
            ok = False
    result = {"rank": rank, "match": ok,
              "grad_sum": float(sum(p.grad.abs().sum()
                                    for p in model.parameters()))}
    dist.destroy_process_group()
    return result


if __name__ == "__main__":
    mode = sys.argv[1]
    rank = int(sys.argv[2])
    world = int(sys.argv[3])
    fn = {"linear": run_linear, "maskrcnn": run_maskrcnn,
          "staged": run_staged,
          "accum": run_accum}[mode]
    print(json.dumps(fn(rank, world)), flush=True)
