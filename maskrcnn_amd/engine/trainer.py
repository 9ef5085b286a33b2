"""Training engine.

Replicates the reference training behavior (MaskRCNN.train/compile,
mrcnn/model.py:~2120-2370): SGD(momentum=0.9, clipnorm=5.0), the five
weighted losses + L2 weight decay over non-BN weights, layer-freeze
schedules ('heads'/'3+'/'4+'/'5+'/'all'), per-epoch checkpoints
(mask_rcnn_{name}_{epoch:04d}.pth) and resume via find_last().

MI355X-native: bf16 autocast on GPU, channels-last conv layout, DDP via
maskrcnn_amd.distributed (RCCL over xGMI) when launched under torchrun.
"""

from __future__ import annotations

import logging
import os
import time

import torch

from ..data.loader import make_data_loader

log = logging.getLogger(__name__)


def build_optimizer(model, config, learning_rate):
    """SGD + momentum; weight decay applied only to conv/linear weights
    (reference compile() adds L2 over non-BN weight matrices, not biases)."""
    # Classify no-decay params by module TYPE (norm layers), not by name
    # substring: BN inside nn.Sequential heads is named convs.1 etc. and
    # a name check would wrongly decay its gamma/beta.
    norm_types = (torch.nn.modules.batchnorm._BatchNorm,
                  torch.nn.GroupNorm, torch.nn.LayerNorm)
    try:
        from ..model.resnet import FrozenBatchNorm2d
        norm_types = norm_types + (FrozenBatchNorm2d,)
    except ImportError:
        pass
    norm_param_ids = set()
    for mod in model.modules():
        if isinstance(mod, norm_types):
            norm_param_ids.update(id(p) for p in mod.parameters(recurse=False))
    decay, no_decay = [], []
    for name, p in model.named_parameters():
        if not p.requires_grad:
            continue
        if name.endswith(".bias") or id(p) in norm_param_ids:
            no_decay.append(p)
        else:
            decay.append(p)
    return torch.optim.SGD(
        [{"params": decay, "weight_decay": config.WEIGHT_DECAY},
         {"params": no_decay, "weight_decay": 0.0}],
        lr=learning_rate, momentum=config.LEARNING_MOMENTUM)


class Trainer:
    def __init__(self, wrapper, train_dataset, val_dataset=None,
                 augmentation=None, num_workers=0):
        self.wrapper = wrapper              # MaskRCNN facade
        self.config = wrapper.config
        self.model = wrapper.model
        self.device = wrapper.device
        self.train_dataset = train_dataset
        self.val_dataset = val_dataset
        self.augmentation = augmentation
        self.num_workers = num_workers
        self.history = []
        from .._rocm_env import setup_rocm_perf
        setup_rocm_perf()
        # DDP over RCCL/xGMI (or gloo on CPU) when launched under
        # torchrun — replaces the reference's ParallelModel.
        import torch.distributed as dist
        self.distributed = dist.is_available() and dist.is_initialized()
        self.rank = dist.get_rank() if self.distributed else 0
        self.ddp = None
        if self.distributed:
            self._rebuild_ddp()

    def _rebuild_ddp(self):
        """(Re)build the bucketed all-reduce wrapper over the CURRENT
        trainable set — must re-run after set_trainable() changes which
        params get grads (buckets of frozen params would never fill)."""
        from ..distributed import DistributedTrainer
        if self.ddp is not None:
            self.ddp.remove_hooks()
        self.ddp = DistributedTrainer(self.model, self.config)

    def _autocast(self):
        use_bf16 = (self.config.COMPUTE_DTYPE == "bf16" and
                    self.device.type == "cuda")
        return torch.autocast(device_type="cuda", dtype=torch.bfloat16,
                              enabled=use_bf16)

    def train_step(self, batch, optimizer):
        cfg = self.config
        self.model.train()
        batch = {k: v.to(self.device, non_blocking=True)
                 for k, v in batch.items()}
        images = batch["image"]
        if cfg.CHANNELS_LAST and self.device.type == "cuda":
            images = images.to(memory_format=torch.channels_last)
        anchors = self.wrapper.anchor_cache.get(cfg.IMAGE_SHAPE,
                                                device=self.device)
        from ..utils.meta import parse_image_meta_t
        meta = parse_image_meta_t(batch["image_meta"])
        with self._autocast():
            outputs = self.model(
                images, batch["image_meta"], anchors,
                gt_class_ids=batch["gt_class_ids"],
                gt_boxes=batch["gt_boxes"],
                gt_masks=batch["gt_masks"],
                mode="training")
            losses = self.wrapper.compute_losses(
                {"rpn_match": batch["rpn_match"],
                 "rpn_bbox": batch["rpn_bbox"],
                 "active_class_ids": meta["active_class_ids"]},
                outputs)
        optimizer.zero_grad(set_to_none=True)
        losses["total_loss"].backward()
        if self.ddp is not None:
            self.ddp.finalize_backward()
        torch.nn.utils.clip_grad_norm_(
            (p for p in self.model.parameters() if p.requires_grad),
            self.config.GRADIENT_CLIP_NORM)
        optimizer.step()
        return {k: float(v.detach()) for k, v in losses.items()}

    @torch.no_grad()
    def _validate(self, steps=None):
        """Mean losses over VALIDATION_STEPS val batches (the reference's
        fit_generator validation_data pass; keras val_* scalars)."""
        if self.val_dataset is None:
            return {}
        cfg = self.config
        steps = steps or cfg.VALIDATION_STEPS
        loader = make_data_loader(self.val_dataset, cfg, shuffle=False,
                                  distributed=self.distributed)
        from ..utils.meta import parse_image_meta_t
        totals = {}
        n = 0
        it = iter(loader)
        anchors = self.wrapper.anchor_cache.get(cfg.IMAGE_SHAPE,
                                                device=self.device)
        was_training = self.model.training
        self.model.train()  # training graph (targets + 5 losses)
        for _ in range(steps):
            try:
                batch = next(it)
            except StopIteration:
                break
            batch = {k: v.to(self.device, non_blocking=True)
                     for k, v in batch.items()}
            images = batch["image"]
            if cfg.CHANNELS_LAST and self.device.type == "cuda":
                images = images.to(memory_format=torch.channels_last)
            meta = parse_image_meta_t(batch["image_meta"])
            with self._autocast():
                outputs = self.model(
                    images, batch["image_meta"], anchors,
                    gt_class_ids=batch["gt_class_ids"],
                    gt_boxes=batch["gt_boxes"],
                    gt_masks=batch["gt_masks"], mode="training")
                losses = self.wrapper.compute_losses(
                    {"rpn_match": batch["rpn_match"],
                     "rpn_bbox": batch["rpn_bbox"],
                     "active_class_ids": meta["active_class_ids"]},
                    outputs)
            for k, v in losses.items():
                totals[k] = totals.get(k, 0.0) + float(v)
            n += 1
        if not was_training:
            self.model.eval()
        return {f"val_{k}": v / n for k, v in totals.items()} if n else {}

    def _log_scalars(self, epoch, means, seconds):
        """Per-epoch scalars: CSV always; TensorBoard when importable
        (reference used a TensorBoard callback, model.py:~2340)."""
        if self.rank != 0:
            return
        import csv
        path = os.path.join(self.wrapper.log_dir, "metrics.csv")
        new = not os.path.exists(path)
        with open(path, "a", newline="") as f:
            wtr = csv.writer(f)
            if new:
                wtr.writerow(["epoch", "seconds"] + sorted(means))
            wtr.writerow([epoch, round(seconds, 2)] +
                         [round(means[k], 6) for k in sorted(means)])
        try:
            from torch.utils.tensorboard import SummaryWriter
            if not hasattr(self, "_tb"):
                self._tb = SummaryWriter(self.wrapper.log_dir)
            for k, v in means.items():
                self._tb.add_scalar(k, v, epoch)
        except ImportError:
            pass

    def fit(self, learning_rate, epochs, layers="all",
            steps_per_epoch=None, custom_callbacks=None):
        """custom_callbacks: list of callables(epoch, scalars_dict) run
        per epoch on rank 0 (the reference's Keras custom_callbacks
        parameter, model.py:~2310)."""
        cfg = self.config
        self.wrapper.set_trainable(layers)
        if self.distributed:
            self._rebuild_ddp()
        optimizer = build_optimizer(self.model, cfg, learning_rate)
        self._resumed_opt = self._maybe_resume_optimizer(optimizer)
        loader = make_data_loader(self.train_dataset, cfg,
                                  augmentation=self.augmentation,
                                  num_workers=self.num_workers,
                                  distributed=self.distributed)
        steps_per_epoch = steps_per_epoch or cfg.STEPS_PER_EPOCH
        if self.rank == 0:
            os.makedirs(self.wrapper.log_dir, exist_ok=True)

        while self.wrapper.epoch < epochs:
            epoch = self.wrapper.epoch
            if self.distributed and loader.sampler is not None and \
                    hasattr(loader.sampler, "set_epoch"):
                loader.sampler.set_epoch(epoch)
            if hasattr(loader.dataset, "set_epoch"):
                loader.dataset.set_epoch(epoch)
            t0 = time.time()
            step = 0
            epoch_losses = {}
            it = iter(loader)
            while step < steps_per_epoch:
                try:
                    batch = next(it)
                except StopIteration:
                    it = iter(loader)
                    batch = next(it)
                losses = self.train_step(batch, optimizer)
                for k, v in losses.items():
                    epoch_losses[k] = epoch_losses.get(k, 0.0) + v
                step += 1
            means = {k: v / step for k, v in epoch_losses.items()}
            means.update(self._validate())
            self.history.append(means)
            dt = time.time() - t0
            log.info("epoch %d: %.1fs %s", epoch, dt,
                     {k: round(v, 4) for k, v in means.items()})
            self._log_scalars(epoch, means, dt)
            if custom_callbacks and self.rank == 0:
                for cb in custom_callbacks:
                    cb(epoch, dict(means))
            if self.rank == 0:
                ckpt = self.wrapper.checkpoint_path.format(epoch=epoch)
                self.wrapper.save_weights(ckpt)
                # Full train state beside the weights (the reference
                # saved weights only; SGD momentum + epoch make resume
                # exact). Latest-only: overwritten each epoch.
                torch.save({"optimizer": optimizer.state_dict(),
                            "epoch": epoch},
                           os.path.join(self.wrapper.log_dir,
                                        "train_state.pth"))
            self.wrapper.epoch += 1
        return self.history

    def _maybe_resume_optimizer(self, optimizer) -> bool:
        """Restore SGD momentum when resuming (find_last/set_log_dir set
        wrapper.epoch > 0 and the matching train_state.pth is present).
        Param-group mismatch (different frozen set) is non-fatal: log
        and train with fresh momentum."""
        if self.wrapper.epoch == 0:
            return False
        path = os.path.join(self.wrapper.log_dir, "train_state.pth")
        if not os.path.exists(path):
            return False
        try:
            state = torch.load(path, map_location="cpu",
                               weights_only=True)
            if state.get("epoch") != self.wrapper.epoch - 1:
                log.warning("train_state.pth is for epoch %s, resuming "
                            "at %d: skipping optimizer restore",
                            state.get("epoch"), self.wrapper.epoch)
                return False
            optimizer.load_state_dict(state["optimizer"])
            log.info("restored optimizer state from %s", path)
            return True
        except (ValueError, KeyError, RuntimeError) as e:
            log.warning("could not restore optimizer state: %s", e)
            return False
