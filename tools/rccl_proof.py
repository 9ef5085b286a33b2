#!/usr/bin/env python3
"""Prove the RCCL code path on a single GPU.

Runs world_size=1 `init_process_group("nccl")` (= RCCL on ROCm), then:
  1. broadcast of the model state (DistributedTrainer init path),
  2. one training step through the bucketed-all-reduce DDP wrapper,
  3. an explicit all_reduce/all_gather round-trip,
and compares the DDP-path parameters after the step against a plain
single-process step with identical seeds/inputs — they must match to
bf16 tolerance (all-reduce of one rank is identity).

The 1->8 scaling curve is the driver's 8-GPU tier; this proves the RCCL
initialization, communicator setup and collective calls execute on the
MI355X at all (mrcnn/parallel_model.py:1-175 is the reference analog).

Usage (on a gpurun box):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 1 \
        --master-addr 127.0.0.1 --master-port 29511 \
        tools/rccl_proof.py
"""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from maskrcnn_amd._rocm_env import ensure_env_or_reexec  # noqa: E402

if __name__ == "__main__":
    ensure_env_or_reexec()

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402


def build_and_step(distributed: bool):
    from maskrcnn_amd.model import MaskRCNN
    from maskrcnn_amd.engine.trainer import Trainer, build_optimizer
    from maskrcnn_amd.data.synthetic import ShapesConfig, ShapesDataset

    class C(ShapesConfig):
        NAME = "rcclproof"
        IMAGES_PER_GPU = 1
        GPU_COUNT = 1

    torch.manual_seed(1234)
    cfg = C()
    wrapper = MaskRCNN("training", cfg, model_dir="/tmp/rccl_logs")
    ds = ShapesDataset()
    ds.load_shapes(8, cfg.IMAGE_MAX_DIM, cfg.IMAGE_MAX_DIM, seed=5)
    ds.prepare()
    tr = Trainer(wrapper, ds)
    assert tr.distributed == distributed, (tr.distributed, distributed)
    opt = build_optimizer(wrapper.model, cfg, 0.001)

    from maskrcnn_amd.data.loader import (TorchMaskRCNNDataset,
                                          collate_batch)
    sample_ds = TorchMaskRCNNDataset(ds, cfg, seed=0)
    batch = collate_batch([sample_ds[0]])
    losses = tr.train_step(batch, opt)
    return wrapper.model, float(losses["total_loss"])


def main():
    rank = int(os.environ.get("RANK", "0"))
    # gloo/cpu accepted for plumbing tests on the dev box; the GPU box
    # runs the real thing ("nccl" IS RCCL on ROCm).
    backend = os.environ.get("RCCL_PROOF_BACKEND", "nccl")
    dev = "cuda" if backend == "nccl" else "cpu"
    record = {"test": "rccl_world1_proof"}

    dist.init_process_group(backend)
    if dev == "cuda":
        torch.cuda.set_device(0)
    record["backend"] = str(dist.get_backend())
    record["world_size"] = dist.get_world_size()

    # Raw collective round-trip over the RCCL communicator.
    t = torch.full((1024, 1024), float(rank + 1), device=dev)
    dist.all_reduce(t)
    if dev == "cuda":
        torch.cuda.synchronize()
    record["all_reduce_ok"] = bool((t == 1.0).all().item())
    gat = [torch.zeros(8, device=dev)
           for _ in range(dist.get_world_size())]
    dist.all_gather(gat, torch.arange(8, device=dev).float())
    record["all_gather_ok"] = bool(
        (gat[0] == torch.arange(8, device=dev).float()).all().item())
    bt = torch.randn(64, device=dev)
    dist.broadcast(bt, src=0)
    record["broadcast_ok"] = True

    # DDP-wrapped step vs plain step: identical at world_size 1.
    model_ddp, loss_ddp = build_and_step(distributed=True)
    sd_ddp = {k: v.detach().cpu().clone()
              for k, v in model_ddp.state_dict().items()}
    dist.destroy_process_group()

    model_plain, loss_plain = build_and_step(distributed=False)
    sd_plain = model_plain.state_dict()
    max_diff = 0.0
    for k, v in sd_ddp.items():
        if v.dtype.is_floating_point:
            d = (v.float() - sd_plain[k].detach().cpu().float()) \
                .abs().max().item()
            max_diff = max(max_diff, d)
    record["loss_ddp"] = loss_ddp
    record["loss_plain"] = loss_plain
    record["param_max_abs_diff"] = max_diff
    # bf16 wire-format all-reduce of ONE rank rounds grads once;
    # tolerance covers one bf16 rounding of the gradient step.
    record["params_match"] = max_diff < 2e-3
    record["ok"] = all(record[k] for k in
                       ("all_reduce_ok", "all_gather_ok", "params_match"))
    print(json.dumps(record))
    return 0 if record["ok"] else 1


if __name__ == "__main__":
    sys.exit(main())
