"""MaskRCNN: the model facade (build / detect / load / save).

Replicates the capability surface of reference class MaskRCNN
(mrcnn/model.py:~1730-2550): build() in training/inference mode,
mold_inputs/get_anchors/unmold_detections, detect(), detect_molded(),
load_weights(by_name, exclude) with head-surgery support, find_last(),
set_trainable(layer-regex). Training itself lives in engine/trainer.py.

Torch-native NCHW; images are molded client-side exactly as the reference
does (resize 'square' + mean subtraction), so the serving signature
(input_image, input_image_meta, input_anchors) is preserved.
"""

from __future__ import annotations

import datetime
import os
import re

import numpy as np
import torch
import torch.nn as nn

from ..config import Config
from ..utils import (
    compose_image_meta, denorm_boxes, mold_image, resize_image, unmold_mask,
)
from ..utils.anchors import AnchorCache
from .detection import DetectionLayer
from .fpn import FPN
from .heads import FPNClassifierHead, FPNMaskHead
from .losses import compute_losses
from .proposal import ProposalLayer
from .resnet import ResNet
from .rpn import RPN
from .targets import DetectionTargetLayer


class MaskRCNNModel(nn.Module):
    """The network graph (reference MaskRCNN.build, model.py:~1750-2030)."""

    #: when True, forward() logs a GPU-synced wall time per stage
    #: (diagnostic only; adds syncs).
    phase_log = False

    def _mark(self, name, t_prev):
        if not self.phase_log:
            return t_prev
        import logging
        import time
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        t = time.perf_counter()
        logging.getLogger(__name__).info(
            "phase %-18s %8.3f s", name, t - t_prev)
        return t

    def __init__(self, config: Config):
        super().__init__()
        self.config = config
        assert config.IMAGE_SHAPE[0] % 64 == 0 and \
            config.IMAGE_SHAPE[1] % 64 == 0, \
            "Image size must be dividable by 64 (FPN downsampling)."

        self.backbone = ResNet(config.BACKBONE, stage5=True,
                               train_bn=bool(config.TRAIN_BN))
        self.fpn = FPN(out_channels=config.TOP_DOWN_PYRAMID_SIZE)
        self.rpn = RPN(in_channels=config.TOP_DOWN_PYRAMID_SIZE,
                       anchors_per_location=len(config.RPN_ANCHOR_RATIOS),
                       anchor_stride=config.RPN_ANCHOR_STRIDE)
        self.proposal_layer = ProposalLayer(config)
        self.detection_target_layer = DetectionTargetLayer(config)
        self.classifier_head = FPNClassifierHead(
            config, in_channels=config.TOP_DOWN_PYRAMID_SIZE)
        self.mask_head = FPNMaskHead(
            config, in_channels=config.TOP_DOWN_PYRAMID_SIZE)
        self.detection_layer = DetectionLayer(config)

    def extract_features(self, images):
        """Backbone + FPN. images: [B, 3, H, W] molded."""
        _, c2, c3, c4, c5 = self.backbone(images)
        return self.fpn(c2, c3, c4, c5)

    def forward(self, images, image_meta, anchors, gt_class_ids=None,
                gt_boxes=None, gt_masks=None, mode="inference",
                input_rois=None):
        """
        images: [B, 3, H, W] molded; anchors: [A, 4] normalized;
        gt_*: training mode only (gt_boxes normalized, gt_masks
        [B, mh, mw, MAX_GT]).
        Returns a dict of named outputs matching the serving signature.
        """
        cfg = self.config
        import time as _time
        t = _time.perf_counter() if self.phase_log else 0
        p2, p3, p4, p5, p6 = self.extract_features(images)
        t = self._mark("backbone+fpn", t)
        rpn_feature_maps = [p2, p3, p4, p5, p6]
        mrcnn_feature_maps = [p2, p3, p4, p5]

        rpn_class_logits, rpn_class, rpn_bbox = self.rpn(rpn_feature_maps)
        t = self._mark("rpn", t)

        proposal_count = (cfg.POST_NMS_ROIS_TRAINING if mode == "training"
                          else cfg.POST_NMS_ROIS_INFERENCE)
        rpn_rois = self.proposal_layer(rpn_class, rpn_bbox, anchors,
                                       proposal_count)
        t = self._mark("proposal", t)

        if mode == "training":
            # USE_RPN_ROIS=False: train heads on externally provided rois
            # (reference input_rois path, model.py:~1915).
            target_rois = rpn_rois if cfg.USE_RPN_ROIS else input_rois
            assert target_rois is not None, \
                "USE_RPN_ROIS=False requires input_rois"
            rois, target_class_ids, target_deltas, target_masks = \
                self.detection_target_layer(target_rois, gt_class_ids,
                                            gt_boxes, gt_masks)
            mrcnn_class_logits, mrcnn_class, mrcnn_bbox = \
                self.classifier_head(rois, mrcnn_feature_maps)
            mrcnn_mask = self.mask_head(rois, mrcnn_feature_maps)
            return {
                "rpn_class_logits": rpn_class_logits,
                "rpn_class": rpn_class,
                "rpn_bbox": rpn_bbox,
                "rpn_rois": rpn_rois,
                "rois": rois,
                "target_class_ids": target_class_ids,
                "target_deltas": target_deltas,
                "target_masks": target_masks,
                "mrcnn_class_logits": mrcnn_class_logits,
                "mrcnn_class": mrcnn_class,
                "mrcnn_bbox": mrcnn_bbox,
                "mrcnn_mask": mrcnn_mask,
            }
        else:
            mrcnn_class_logits, mrcnn_class, mrcnn_bbox = \
                self.classifier_head(rpn_rois, mrcnn_feature_maps)
            t = self._mark("classifier_head", t)
            detections = self.detection_layer(rpn_rois, mrcnn_class,
                                              mrcnn_bbox, image_meta)
            t = self._mark("detection_layer", t)
            detection_boxes = detections[..., :4]
            # Mask head runs on the FINAL detections (reference :~2000).
            mrcnn_mask = torch.sigmoid(
                self.mask_head(detection_boxes, mrcnn_feature_maps))
            t = self._mark("mask_head", t)
            return {
                "mrcnn_detection": detections,
                "mrcnn_class": mrcnn_class,
                "mrcnn_bbox": mrcnn_bbox,
                "mrcnn_mask": mrcnn_mask,
                "rois": rpn_rois,
                "rpn_class": rpn_class,
                "rpn_bbox": rpn_bbox,
            }


class MaskRCNN:
    """User-facing facade (reference MaskRCNN class API)."""

    def __init__(self, mode: str, config: Config, model_dir: str = "logs"):
        assert mode in ("training", "inference")
        self.mode = mode
        self.config = config
        self.model_dir = model_dir
        self.epoch = 0
        self.anchor_cache = AnchorCache(config)
        self.keras_model = None  # parity alias; set to self.model below
        self.model = MaskRCNNModel(config)
        self.keras_model = self.model
        self.device = torch.device(
            "cuda" if torch.cuda.is_available() else "cpu")
        self.model.to(self.device)
        if mode == "inference":
            self.model.eval()
        self.set_log_dir()

    # ------------------------------------------------------------------ io
    def set_log_dir(self, model_path: str = None):
        """Set log dir + next checkpoint path (reference set_log_dir:~2230)."""
        self.epoch = 0
        now = datetime.datetime.now()
        if model_path:
            regex = r".*[/\\][\w\-]+(\d{4})(\d{2})(\d{2})T(\d{2})(\d{2})[/\\]mask\_rcnn\_[\w\-]+(\d{4})\.pth"
            m = re.match(regex, model_path.replace(".pt", ".pth"))
            if m:
                now = datetime.datetime(int(m.group(1)), int(m.group(2)),
                                        int(m.group(3)), int(m.group(4)),
                                        int(m.group(5)))
                self.epoch = int(m.group(6)) + 1
                print(f"Re-starting from epoch {self.epoch}")
        name = (self.config.NAME or "maskrcnn").lower()
        self.log_dir = os.path.join(
            self.model_dir, f"{name}{now:%Y%m%dT%H%M}")
        self.checkpoint_path = os.path.join(
            self.log_dir, f"mask_rcnn_{name}_{{epoch:04d}}.pth")

    def find_last(self) -> str:
        """Newest checkpoint in the newest log dir (reference find_last)."""
        name = (self.config.NAME or "maskrcnn").lower()
        dir_names = sorted(
            d for d in (os.listdir(self.model_dir)
                        if os.path.isdir(self.model_dir) else [])
            if d.startswith(name))
        if not dir_names:
            raise FileNotFoundError(
                f"Could not find model directory under {self.model_dir}")
        for d in reversed(dir_names):
            ckpts = sorted(f for f in os.listdir(
                os.path.join(self.model_dir, d))
                if f.startswith("mask_rcnn") and f.endswith(".pth"))
            if ckpts:
                return os.path.join(self.model_dir, d, ckpts[-1])
        raise FileNotFoundError(
            f"Could not find weight files in {self.model_dir}")

    def save_weights(self, path: str):
        os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
        torch.save({"model": self.model.state_dict(),
                    "config": self.config.to_dict()}, path)

    def load_weights(self, path: str, by_name: bool = False, exclude=None):
        """Load weights; `exclude` drops head layers for class-count surgery
        (reference load_weights:~2080). by_name accepted for parity.

        `.h5` paths load matterport Keras checkpoints directly via the
        name/transpose map in utils/keras_import.py (requires h5py);
        exclude then takes the reference's KERAS layer names."""
        if path.endswith(".h5"):
            from ..utils.keras_import import load_keras_h5
            load_keras_h5(self.model, path, exclude=exclude)
            self.set_log_dir(path)
            return []
        # weights_only=True: checkpoints store only tensors + a plain
        # config dict (save_weights above), so the pickle code-exec
        # surface is closed even for paths found via find_last().
        state = torch.load(path, map_location="cpu", weights_only=True)
        sd = state["model"] if isinstance(state, dict) and "model" in state \
            else state
        if exclude:
            pats = [re.compile(p) for p in exclude]
            sd = {k: v for k, v in sd.items()
                  if not any(p.search(k) for p in pats)}
        missing, unexpected = self.model.load_state_dict(sd, strict=False)
        if not exclude and (missing or unexpected):
            # Only head-surgery loads may be partial.
            allowed_missing = [m for m in missing]
            if unexpected:
                raise RuntimeError(f"Unexpected keys: {unexpected[:5]}...")
            if allowed_missing and not by_name:
                raise RuntimeError(f"Missing keys: {allowed_missing[:5]}...")
        self.set_log_dir(path)
        return missing if exclude else []

    def get_imagenet_weights(self):
        """Reference downloads ResNet50-notop .h5 from the Keras release
        bucket (model.py:~2100). This build is offline by design."""
        raise RuntimeError(
            "ImageNet bootstrap weights cannot be downloaded in this "
            "offline build; pass a local checkpoint to load_weights() "
            "instead.")

    def get_trainable_layers(self):
        """Modules that own at least one trainable parameter
        (reference get_trainable_layers)."""
        out = []
        for name, module in self.model.named_modules():
            if name and any(p.requires_grad
                            for p in module.parameters(recurse=False)):
                out.append((name, module))
        return out

    def find_trainable_layer(self, name: str):
        """Innermost trainable module containing `name`
        (reference find_trainable_layer/ancestor analog)."""
        parts = name.split(".")
        modules = dict(self.model.named_modules())
        for i in range(len(parts), 0, -1):
            cand = ".".join(parts[:i])
            m = modules.get(cand)
            if m is not None and any(p.requires_grad
                                     for p in m.parameters(recurse=False)):
                return cand, m
        return None

    # ------------------------------------------------------- trainability
    LAYER_REGEX = {
        # Heads: everything but the backbone.
        "heads": r"^(fpn|rpn|classifier_head|mask_head)\.",
        "3+": r"^(backbone\.layer[345])|^(fpn|rpn|classifier_head|mask_head)\.",
        "4+": r"^(backbone\.layer[45])|^(fpn|rpn|classifier_head|mask_head)\.",
        "5+": r"^(backbone\.layer5)|^(fpn|rpn|classifier_head|mask_head)\.",
        "all": r".*",
    }

    def set_trainable(self, layer_regex: str, verbose: int = 0):
        """Freeze/unfreeze parameters by regex (reference set_trainable)."""
        pattern = self.LAYER_REGEX.get(layer_regex, layer_regex)
        rx = re.compile(pattern)
        n_train = 0
        for name, p in self.model.named_parameters():
            trainable = bool(rx.match(name))
            p.requires_grad_(trainable)
            n_train += trainable
            if verbose and trainable:
                print("train:", name)
        return n_train

    # ------------------------------------------------------------ molding
    def mold_inputs(self, images):
        """List of raw RGB images -> (molded [B,H,W,3], metas, windows).

        Reference mold_inputs:~2380.
        """
        molded_images, image_metas, windows = [], [], []
        for image in images:
            molded, window, scale, padding, crop = resize_image(
                image,
                min_dim=self.config.IMAGE_MIN_DIM,
                min_scale=self.config.IMAGE_MIN_SCALE,
                max_dim=self.config.IMAGE_MAX_DIM,
                mode=self.config.IMAGE_RESIZE_MODE)
            molded = mold_image(molded, self.config)
            meta = compose_image_meta(
                0, image.shape, molded.shape, window, scale,
                np.zeros([self.config.NUM_CLASSES], dtype=np.int32))
            molded_images.append(molded)
            image_metas.append(meta)
            windows.append(window)
        return (np.stack(molded_images), np.stack(image_metas),
                np.stack(windows))

    def get_anchors(self, image_shape):
        """Normalized anchors (cached per shape); numpy for serving parity."""
        return self.anchor_cache.get(image_shape).numpy()

    @staticmethod
    def unmold_detections(detections, mrcnn_mask, original_image_shape,
                          image_shape, window):
        """Reference unmold_detections:~2500 (static: pure numpy).

        detections: [D, 6]; mrcnn_mask: [D, C, mh, mw] (torch order);
        Returns (boxes px [N,4], class_ids [N], scores [N], masks [H,W,N]).
        """
        zero_ix = np.where(detections[:, 4] == 0)[0]
        N = zero_ix[0] if zero_ix.shape[0] > 0 else detections.shape[0]

        boxes = detections[:N, :4]
        class_ids = detections[:N, 4].astype(np.int32)
        scores = detections[:N, 5]
        masks = mrcnn_mask[np.arange(N), class_ids]  # [N, mh, mw]

        # Translate normalized coords in the padded image to pixel coords in
        # the original.
        from ..utils.boxes import norm_boxes as _norm
        wy1, wx1, wy2, wx2 = _norm(np.array(window, dtype=np.float32),
                                   image_shape[:2])
        shift = np.array([wy1, wx1, wy1, wx1])
        wh = wy2 - wy1
        ww = wx2 - wx1
        scale = np.array([wh, ww, wh, ww])
        boxes = np.divide(boxes - shift, scale)
        boxes = denorm_boxes(boxes, original_image_shape[:2])

        # Filter zero-area boxes.
        exclude_ix = np.where(
            (boxes[:, 2] - boxes[:, 0]) * (boxes[:, 3] - boxes[:, 1]) <= 0)[0]
        if exclude_ix.shape[0] > 0:
            boxes = np.delete(boxes, exclude_ix, axis=0)
            class_ids = np.delete(class_ids, exclude_ix, axis=0)
            scores = np.delete(scores, exclude_ix, axis=0)
            masks = np.delete(masks, exclude_ix, axis=0)
            N = class_ids.shape[0]

        # Paste each box-local resize straight into the [H, W, N]
        # canvas: building per-detection full frames and np.stack-ing
        # them was 1.11 s of a 1.18 s unmold at 100 dets @1024²
        # (strided copy of N full images); this is ~60 ms.
        from ..utils.image import resize as _resize
        Ho, Wo = original_image_shape[:2]
        # Native path: one fused C++ pass (bilinear resize + threshold
        # + paste, at::parallel_for over detections) — ~10x the
        # threaded python loop below at 100 dets.
        from ..ops._ext import load_ext as _load_ext
        _ext = _load_ext()
        if N > 0 and _ext is not None and hasattr(_ext, "unmold_masks"):
            fm = _ext.unmold_masks(
                torch.from_numpy(np.ascontiguousarray(masks)),
                torch.from_numpy(np.ascontiguousarray(boxes)).long(),
                int(Ho), int(Wo), 0.5).numpy()
            return boxes, class_ids, scores, np.transpose(fm, (1, 2, 0))
        # [N, H, W] internally (contiguous pastes and contiguous
        # masks[:, :, i] slices for callers); returned transposed as
        # the reference's [H, W, N] convention — a view, no copy.
        full_masks = np.zeros((N, Ho, Wo), dtype=bool)

        def _paste(i):
            y1, x1, y2, x2 = boxes[i]
            m = _resize(masks[i], (y2 - y1, x2 - x1)) >= 0.5
            full_masks[i, y1:y2, x1:x2] = m

        # The bilinear resizes dominate (~0.5 ms each at 100 dets) and
        # torch ops release the GIL — thread the paste loop (disjoint
        # output rows, no races). ~3x at 100 dets on 4 cores.
        if N >= 8:
            import os as _os
            from concurrent.futures import ThreadPoolExecutor
            nw = min(4, _os.cpu_count() or 4)  # 8 workers contended
            with ThreadPoolExecutor(max_workers=nw) as pool:
                list(pool.map(_paste, range(N)))
        else:
            for i in range(N):
                _paste(i)
        return boxes, class_ids, scores, \
            np.transpose(full_masks, (1, 2, 0))

    # ---------------------------------------------------------- inference
    @torch.no_grad()
    def detect(self, images, verbose: int = 0):
        """Run detection on a list of raw images (reference detect:~2450)."""
        assert self.mode == "inference", "Create model in inference mode."
        assert len(images) == self.config.BATCH_SIZE, \
            "len(images) must equal BATCH_SIZE"
        if verbose:
            from ..utils.meta import log as _log
            _log(f"Processing {len(images)} images")
            for image in images:
                _log("image", image)
        molded_images, image_metas, windows = self.mold_inputs(images)
        if verbose:
            from ..utils.meta import log as _log
            _log("molded_images", molded_images)
            _log("image_metas", image_metas)
        anchors = self.anchor_cache.get(molded_images[0].shape,
                                        device=self.device)
        outputs = self.predict_molded(molded_images, image_metas, anchors)
        detections = outputs["mrcnn_detection"].cpu().numpy()
        mrcnn_mask = outputs["mrcnn_mask"].cpu().numpy()

        results = []
        for i, image in enumerate(images):
            boxes, class_ids, scores, masks = self.unmold_detections(
                detections[i], mrcnn_mask[i], image.shape,
                molded_images[i].shape, windows[i])
            results.append({"rois": boxes, "class_ids": class_ids,
                            "scores": scores, "masks": masks})
        return results

    @torch.no_grad()
    def detect_molded(self, molded_images, image_metas, verbose: int = 0):
        """Detection on already-molded inputs -> list of result dicts
        (reference detect_molded:~2480 returns UNMOLDED results, same
        contract as detect()). Use predict_molded for raw tensors."""
        from ..utils.meta import parse_image_meta
        outputs = self.predict_molded(
            molded_images, image_metas,
            self.anchor_cache.get(molded_images[0].shape,
                                  device=self.device))
        detections = outputs["mrcnn_detection"].cpu().numpy()
        mrcnn_mask = outputs["mrcnn_mask"].cpu().numpy()
        results = []
        for i in range(molded_images.shape[0]):
            meta = parse_image_meta(image_metas[i:i + 1])
            boxes, class_ids, scores, masks = self.unmold_detections(
                detections[i], mrcnn_mask[i],
                tuple(int(v) for v in meta["original_image_shape"][0]),
                tuple(int(v) for v in meta["image_shape"][0]),
                tuple(int(v) for v in meta["window"][0]))
            results.append({"rois": boxes, "class_ids": class_ids,
                            "scores": scores, "masks": masks})
        return results

    @torch.no_grad()
    def predict_molded(self, molded_images, image_metas, anchors):
        """numpy molded [B,H,W,3] + metas -> dict of torch outputs."""
        images_t = torch.from_numpy(
            np.ascontiguousarray(molded_images)).float() \
            .permute(0, 3, 1, 2).to(self.device)
        metas_t = torch.from_numpy(image_metas).float().to(self.device)
        if isinstance(anchors, np.ndarray):
            anchors = torch.from_numpy(anchors).to(self.device)
        self.model.eval()
        return self.model(images_t, metas_t, anchors, mode="inference")

    @torch.no_grad()
    def run_graph(self, images, names, mode="inference"):
        """Evaluate named intermediates (reference run_graph, model.py
        :~2530 — the inspect_model.ipynb probe mechanism).

        names: list of either forward-output keys (e.g. "rois",
        "mrcnn_detection") or dotted module names (e.g. "backbone.layer2",
        "fpn", "rpn") whose outputs are captured via hooks.
        Returns {name: numpy array / list of arrays}.
        """
        captured = {}
        hooks = []
        module_map = dict(self.model.named_modules())
        for name in names:
            if name in module_map:
                def make_hook(n):
                    def hook(_m, _inp, out):
                        captured[n] = out
                    return hook
                hooks.append(
                    module_map[name].register_forward_hook(
                        make_hook(name)))
        try:
            molded_images, image_metas, _ = self.mold_inputs(images)
            anchors = self.anchor_cache.get(molded_images[0].shape,
                                            device=self.device)
            outputs = self.predict_molded(molded_images, image_metas,
                                          anchors)
        finally:
            for h in hooks:
                h.remove()
        for name in names:
            if name in outputs:
                captured[name] = outputs[name]

        def to_np(v):
            if isinstance(v, torch.Tensor):
                return v.detach().float().cpu().numpy()
            if isinstance(v, (list, tuple)):
                return [to_np(x) for x in v]
            return v
        missing = [n for n in names if n not in captured]
        if missing:
            raise KeyError(f"run_graph: unknown names {missing}; "
                           f"outputs={list(outputs)}")
        return {n: to_np(captured[n]) for n in names}

    # ------------------------------------------------------------ training
    def compute_losses(self, inputs, outputs):
        return compute_losses(self.config, inputs, outputs)

    def train(self, train_dataset, val_dataset, learning_rate, epochs,
              layers, augmentation=None, custom_callbacks=None,
              no_augmentation_sources=None):
        """Delegates to engine.trainer.Trainer (reference train:~2310)."""
        from ..engine.trainer import Trainer
        trainer = Trainer(self, train_dataset, val_dataset,
                          augmentation=augmentation)
        trainer.fit(learning_rate, epochs, layers,
                    custom_callbacks=custom_callbacks)
