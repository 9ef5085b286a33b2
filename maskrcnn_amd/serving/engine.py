"""Per-GPU inference engine.

The C++ ModelServer's SavedModelBundle+session maps to:
one engine per GPU running the MaskRCNNModel in inference mode (bf16,
channels-last) fed by the dynamic batcher. Anchors are computed
server-side per image shape (fixing the reference's ~4 MB-per-request
anchor payload wart) but client-provided `input_anchors`
are honored for signature compatibility.
"""

from __future__ import annotations

import threading
from typing import Dict, Optional

import numpy as np

from .._rocm_env import setup_rocm_perf
from ..config import Config
from ..utils.anchors import AnchorCache

try:  # per-stage serve-loop timers (SURVEY §5 tracing row)
    from prometheus_client import Histogram
    _STAGE = Histogram(
        "predict_stage_seconds", "engine predict stage wall time",
        ["stage"],  # upload=HtoD+mold, forward=launch side,
                    # dtoh=drain+transfer (forward is async: kernel
                    # time lands in dtoh unless hipGraph replay syncs)
        buckets=(.001, .0025, .005, .01, .025, .05, .1, .25, .5, 1, 2.5))
    _PROM = True
except ImportError:  # pragma: no cover
    _PROM = False


class InferenceEngine:
    def __init__(self, config: Config, state_dict=None, device=None,
                 use_bf16: Optional[bool] = None):
        import torch
        from ..model.mask_rcnn import MaskRCNNModel

        setup_rocm_perf()
        self.config = config
        self.device = torch.device(
            device or ("cuda" if torch.cuda.is_available() else "cpu"))
        self.model = MaskRCNNModel(config)
        if state_dict is not None:
            self.model.load_state_dict(state_dict)
        self.model.to(self.device).eval()
        if config.CHANNELS_LAST and self.device.type == "cuda":
            self.model = self.model.to(memory_format=torch.channels_last)
        self.use_bf16 = (use_bf16 if use_bf16 is not None
                         else (config.COMPUTE_DTYPE == "bf16" and
                               self.device.type == "cuda"))
        self.anchor_cache = AnchorCache(config)
        self._lock = threading.Lock()  # one inference stream per engine
        self._pinned: Dict[tuple, "object"] = {}  # staging buffers
        # hipGraph replay of the steady-state forward (N14). The
        # inference path is sync-free by construction (padded NMS,
        # masked scatter — no .item()/.nonzero() host syncs), which is
        # what makes whole-forward capture legal. Off by default until
        # GPU-validated (round-2 block); enable per-engine or via
        # MASKRCNN_SERVING_HIPGRAPH=1.
        import os
        self.use_hip_graph = (
            self.device.type == "cuda" and
            os.environ.get("MASKRCNN_SERVING_HIPGRAPH", "0") == "1")
        self._graphs: Dict[tuple, dict] = {}  # (B,H,W) -> capture state

    def _upload(self, arr: np.ndarray):
        """Host->device through a reused pinned buffer (a 32x1024² fp32
        batch is 384 MB; pageable uploads stall the copy engine)."""
        import torch
        t = torch.from_numpy(np.ascontiguousarray(arr))
        if self.device.type != "cuda":
            return t
        key = (t.shape, t.dtype)
        buf = self._pinned.get(key)
        if buf is None:
            if len(self._pinned) >= 8:
                # Bounded cache: shape-bucketed batching keeps this at
                # 2-3 entries in practice; evict oldest under churn
                # (pinned pages are a scarce host resource).
                self._pinned.pop(next(iter(self._pinned)))
            buf = torch.empty_like(t, pin_memory=True)
            self._pinned[key] = buf
        buf.copy_(t)
        return buf.to(self.device, non_blocking=True)

    def _upload_batch(self, parts, total: int):
        """Collate a list of [bi, H, W, C] arrays (same dtype/shape
        tail) into ONE pinned [total, H, W, C] buffer — each request's
        bytes are copied exactly once, and rows past the real batch
        (pad-to-max) are zero-filled."""
        import torch
        if len(parts) == 1 and parts[0].shape[0] == total:
            return self._upload(parts[0])
        tail = parts[0].shape[1:]
        np_dtype = parts[0].dtype
        if self.device.type != "cuda":
            outn = np.zeros((total,) + tail, np_dtype)
            off = 0
            for p in parts:
                outn[off:off + p.shape[0]] = p
                off += p.shape[0]
            return torch.from_numpy(outn)
        ref = torch.from_numpy(
            np.empty((0,) + tail, np_dtype))  # dtype carrier
        key = ((total,) + tail, ref.dtype)
        buf = self._pinned.get(key)
        if buf is None:
            if len(self._pinned) >= 8:
                self._pinned.pop(next(iter(self._pinned)))
            buf = torch.empty((total,) + tail, dtype=ref.dtype,
                              pin_memory=True)
            self._pinned[key] = buf
        off = 0
        for p in parts:
            n = p.shape[0]
            buf[off:off + n].copy_(
                torch.from_numpy(np.ascontiguousarray(p)))
            off += n
        if off < total:
            buf[off:].zero_()  # pinned buffers are reused: clear pad
        return buf.to(self.device, non_blocking=True)

    #: the 7 signature outputs (S1 order)
    OUTPUT_NAMES = ("mrcnn_detection", "mrcnn_class", "mrcnn_bbox",
                    "mrcnn_mask", "rois", "rpn_class", "rpn_bbox")

    def _autocast(self):
        import torch
        return torch.autocast(device_type="cuda", dtype=torch.bfloat16,
                              enabled=self.use_bf16)

    def _forward(self, images, metas, anchors):
        """Eager forward, or hipGraph replay when enabled.

        Replay path: one captured graph per (batch, image-shape, anchor
        count); inputs are copied into the capture's static buffers and
        the whole forward replays as a single graph launch (the serving
        forward is ~300 small launches; replay removes the per-launch
        CPU dispatch that dominates at batch 1)."""
        import torch
        if not (self.use_hip_graph and images.is_cuda):
            with self._autocast():
                return self.model(images, metas, anchors,
                                  mode="inference")
        key = (tuple(images.shape), tuple(anchors.shape))
        with torch.cuda.device(self.device):  # replicas: cuda:N != current
            st = self._graphs.get(key)
            if st is None:
                st = self._capture(images, metas, anchors)
            st["img"].copy_(images)
            st["meta"].copy_(metas)
            st["anc"].copy_(anchors)
            st["graph"].replay()
        return st["out"]

    def _capture(self, images, metas, anchors):
        """Warm up (3 eager runs on a side stream — also drives MIOpen
        Find for the shape) then capture one forward into a hipGraph
        (torch.cuda.CUDAGraph IS hipGraph on ROCm)."""
        import torch
        static_img = images.clone()
        static_meta = metas.clone()
        static_anc = anchors.clone()
        # cache_enabled=False: autocast's weight-cast cache must not be
        # captured (stale casts would be baked into the graph — the
        # torch.cuda.graph + autocast documented requirement).
        ac = lambda: torch.autocast(  # noqa: E731
            device_type="cuda", dtype=torch.bfloat16,
            enabled=self.use_bf16, cache_enabled=False)
        s = torch.cuda.Stream(device=self.device)
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(3):
                with ac():
                    self.model(static_img, static_meta, static_anc,
                               mode="inference")
        torch.cuda.current_stream().wait_stream(s)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            with ac():
                out = self.model(static_img, static_meta, static_anc,
                                 mode="inference")
        st = {"graph": g, "img": static_img, "meta": static_meta,
              "anc": static_anc, "out": out}
        self._graphs[(tuple(images.shape), tuple(anchors.shape))] = st
        return st

    def predict(self, input_image,
                input_image_meta: np.ndarray,
                input_anchors: Optional[np.ndarray] = None,
                trim_to: Optional[int] = None,
                outputs: Optional[list] = None,
                pad_to: Optional[int] = None
                ) -> Dict[str, np.ndarray]:
        """Run the Predict signature.

        input_image: [B, H, W, 3] molded float32 (NHWC, TF convention)
        — or a LIST of such arrays (the batcher's request list): parts
        are collated directly into the pinned staging buffer, skipping
        the host-side np.concatenate (~100 MB / ~20 ms per uint8
        batch-32 under the GIL). pad_to zero-fills trailing rows up to
        the target batch (pad-to-max batching).
        Returns the 7 named outputs as numpy (TF output layout:
        mrcnn_mask as [B, D, 28, 28, C]).
        """
        import torch

        import time as _time

        parts = (list(input_image)
                 if isinstance(input_image, (list, tuple))
                 else [input_image])
        _, H, W, _c = parts[0].shape
        B = pad_to if pad_to is not None \
            else sum(p.shape[0] for p in parts)
        with self._lock, torch.no_grad():
            _t0 = _time.monotonic()
            # NHWC numpy IS the channels_last layout of an NCHW tensor:
            # upload as-is (pinned, async), then view-permute — no host
            # transpose and no extra device copy when CHANNELS_LAST.
            #
            # uint8 extension: an un-molded uint8 image uploads at 1/4
            # the bytes (wire AND PCIe); mold (float - MEAN_PIXEL) runs
            # on the GPU. fp32 input stays the stock molded contract.
            if parts[0].dtype == np.uint8:
                up = self._upload_batch(parts, B)
                mean = getattr(self, "_mean_pixel_t", None)
                if mean is None or mean.device != self.device:
                    mean = torch.tensor(
                        self.config.MEAN_PIXEL,
                        dtype=torch.float32, device=self.device)
                    self._mean_pixel_t = mean
                up = up.float() - mean.view(1, 1, 1, 3)
            else:
                up = self._upload_batch(
                    [p.astype(np.float32, copy=False) for p in parts],
                    B)
            images = up.permute(0, 3, 1, 2)
            if self.config.CHANNELS_LAST and self.device.type == "cuda":
                images = images.contiguous(
                    memory_format=torch.channels_last)
            else:
                images = images.contiguous()
            metas = self._upload(
                input_image_meta.astype(np.float32, copy=False)).float()
            if input_anchors is not None:
                anchors = torch.from_numpy(
                    np.ascontiguousarray(
                        input_anchors[0] if input_anchors.ndim == 3
                        else input_anchors)).float().to(self.device)
            else:
                anchors = self.anchor_cache.get((H, W, 3),
                                                device=self.device)
            _t1 = _time.monotonic()
            out = self._forward(images, metas, anchors)
            _t2 = _time.monotonic()
            # Trim the padded batch rows GPU-side BEFORE DtoH (a padded
            # batch-32 output set is ~1 GB) and transfer only the
            # requested outputs (a full set adds ~12 MB/image of rpn/roi
            # tensors most clients never read).
            k = trim_to if trim_to is not None else B
            wanted = [n for n in self.OUTPUT_NAMES
                      if outputs is None or n in outputs]
            result = {}
            # mrcnn_mask_top extension resolved GPU-SIDE: gather the
            # detected class's mask BEFORE DtoH — [B,D,mh,mw] instead
            # of [B,D,mh,mw,C] is 81x fewer transfer bytes at COCO (a
            # padded batch-32 full mask is ~800 MB), and the CPU-side
            # strided gather it replaces cost ~17 ms/batch under the
            # GIL (server.py previously did this from the full numpy
            # mask).
            if outputs is not None and "mrcnn_mask_top" in outputs:
                import torch as _torch
                det = out["mrcnn_detection"][:k]           # [k, D, 6]
                cls = det[:, :, 4].long()                  # [k, D]
                m = out["mrcnn_mask"][:k]                  # [k,D,C,mh,mw]
                bb = _torch.arange(m.shape[0],
                                   device=m.device)[:, None]
                dd = _torch.arange(m.shape[1],
                                   device=m.device)[None, :]
                top = m[bb, dd, cls].float()               # [k,D,mh,mw]
                result["mrcnn_mask_top"] = top.cpu().numpy()
            for name in wanted:
                t = out[name][:k].float()
                if name == "mrcnn_mask":
                    # Torch [B,D,C,mh,mw] -> TF layout [B,D,mh,mw,C].
                    t = t.permute(0, 1, 3, 4, 2).contiguous()
                result[name] = t.cpu().numpy()
            _t3 = _time.monotonic()
            # upload = HtoD + GPU mold; forward = launch side (async);
            # dtoh = execution drain + transfer. Kept as attributes for
            # tests/tools and exported to Prometheus when available.
            self.last_stage_ms = {
                "upload": (_t1 - _t0) * 1e3,
                "forward": (_t2 - _t1) * 1e3,
                "dtoh": (_t3 - _t2) * 1e3,
            }
            if _PROM:
                _STAGE.labels("upload").observe(_t1 - _t0)
                _STAGE.labels("forward").observe(_t2 - _t1)
                _STAGE.labels("dtoh").observe(_t3 - _t2)
            return result

    @classmethod
    def from_saved_model(cls, export_dir: str, device=None,
                         config: Optional[Config] = None):
        """Build an engine from a SavedModel-layout directory."""
        import torch
        from .savedmodel import load_saved_model

        state, config_dict, sigs = load_saved_model(export_dir)
        if config is None:
            class LoadedConfig(Config):
                NAME = str(config_dict.get("NAME", "serving"))
            config = LoadedConfig()
            for k, v in config_dict.items():
                if hasattr(config, k) and not isinstance(
                        getattr(Config, k, None), property):
                    try:
                        setattr(config, k, v)
                    except Exception:
                        pass
            config.__init__()
        state = {k: (v if isinstance(v, torch.Tensor) else torch.tensor(v))
                 for k, v in state.items()}
        return cls(config, state_dict=state, device=device)
