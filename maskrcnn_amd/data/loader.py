"""Torch Dataset / DataLoader glue replacing the reference's python
generator + fit_generator multiprocessing (mrcnn/model.py data_generator
:~1560). Error-tolerant like the reference: bad images are logged and a
different index is drawn.
"""

from __future__ import annotations

import logging

import numpy as np
import torch
from torch.utils.data import DataLoader
from torch.utils.data import Dataset as _TorchDataset

from ..utils.anchors import AnchorCache
from ..utils.boxes import norm_boxes
from ..utils.image import mold_image
from .targets import build_rpn_targets, load_image_gt

log = logging.getLogger(__name__)


class TorchMaskRCNNDataset(_TorchDataset):
    """Yields the 7-tensor training sample of the reference generator."""

    def __init__(self, dataset, config, augmentation=None, seed=0):
        self.dataset = dataset
        self.config = config
        self.augmentation = augmentation
        self.anchor_cache = AnchorCache(config)
        self.anchors_px = self.anchor_cache.get_pixel(config.IMAGE_SHAPE)
        self.seed = seed
        self.epoch = 0

    def set_epoch(self, epoch: int):
        """Vary augmentation/sampling randomness across epochs (the
        reference's generator kept RNG state across passes; a pure
        (seed, idx) stream would repeat the same flips every epoch)."""
        self.epoch = int(epoch)

    def __len__(self):
        return len(self.dataset.image_ids)

    def __getitem__(self, idx):
        rng = np.random.RandomState(
            (self.seed * 1_000_003 + self.epoch * 7_919 + idx)
            % (2 ** 31))
        for attempt in range(5):
            image_id = self.dataset.image_ids[idx]
            try:
                sample = self._build(image_id, rng)
            except Exception:
                log.exception("Error processing image %s",
                              self.dataset.image_reference(image_id))
                sample = None
            if sample is not None:
                return sample
            idx = int(rng.randint(0, len(self.dataset.image_ids)))
        raise RuntimeError("Too many bad images in a row")

    def _build(self, image_id, rng):
        cfg = self.config
        image, image_meta, gt_class_ids, gt_boxes, gt_masks = load_image_gt(
            self.dataset, cfg, image_id, augmentation=self.augmentation,
            rng=rng)
        # Skip images without any mapped instances (reference behavior).
        if not np.any(gt_class_ids > 0):
            return None

        rpn_match, rpn_bbox = build_rpn_targets(
            image.shape, self.anchors_px, gt_class_ids, gt_boxes, cfg,
            rng=rng)

        # Cap GT instances.
        if gt_boxes.shape[0] > cfg.MAX_GT_INSTANCES:
            ids = rng.choice(np.arange(gt_boxes.shape[0]),
                             cfg.MAX_GT_INSTANCES, replace=False)
            gt_class_ids = gt_class_ids[ids]
            gt_boxes = gt_boxes[ids]
            gt_masks = gt_masks[:, :, ids]

        # Pad GT to MAX_GT_INSTANCES.
        G = cfg.MAX_GT_INSTANCES
        n = gt_boxes.shape[0]
        class_ids_p = np.zeros(G, dtype=np.int64)
        class_ids_p[:n] = gt_class_ids
        boxes_p = np.zeros((G, 4), dtype=np.float32)
        boxes_p[:n] = norm_boxes(gt_boxes.astype(np.float32),
                                 image.shape[:2])
        mh, mw = gt_masks.shape[:2]
        masks_p = np.zeros((mh, mw, G), dtype=np.float32)
        masks_p[:, :, :n] = gt_masks.astype(np.float32)

        molded = mold_image(image, cfg)
        return {
            "image": torch.from_numpy(
                np.ascontiguousarray(molded.transpose(2, 0, 1))).float(),
            "image_meta": torch.from_numpy(image_meta).float(),
            "rpn_match": torch.from_numpy(
                rpn_match[:, None].astype(np.int64)),
            "rpn_bbox": torch.from_numpy(rpn_bbox.astype(np.float32)),
            "gt_class_ids": torch.from_numpy(class_ids_p),
            "gt_boxes": torch.from_numpy(boxes_p),
            "gt_masks": torch.from_numpy(masks_p),
        }


def collate_batch(samples):
    return {k: torch.stack([s[k] for s in samples]) for k in samples[0]}


def make_data_loader(dataset, config, batch_size=None, num_workers=0,
                     augmentation=None, shuffle=True, seed=0,
                     distributed=False):
    tds = TorchMaskRCNNDataset(dataset, config, augmentation=augmentation,
                               seed=seed)
    sampler = None
    if distributed:
        from torch.utils.data.distributed import DistributedSampler
        sampler = DistributedSampler(tds, shuffle=shuffle, seed=seed)
        shuffle = False
    return DataLoader(tds,
                      batch_size=batch_size or config.BATCH_SIZE,
                      shuffle=shuffle, sampler=sampler,
                      num_workers=num_workers, collate_fn=collate_batch,
                      drop_last=True,
                      persistent_workers=num_workers > 0,
                      pin_memory=torch.cuda.is_available())
