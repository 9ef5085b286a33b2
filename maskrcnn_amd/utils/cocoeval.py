"""COCO-style dataset-wide evaluation (pycocotools COCOeval stand-in).

Computes AP@[.5:.95], AP50, AP75, and AR for bbox or segm over a whole
result set, per-class averaged, with greedy score-ordered matching per
image — the COCOeval semantics the reference's evaluate_coco relies on
(samples/coco/coco.py:~300). Area-range breakdowns use
the COCO small/medium/large splits.

Masks are COCO RLE dicts (uncompressed counts list, utils.masks codec) or
[H, W] arrays.
"""

from __future__ import annotations

from collections import defaultdict

import numpy as np

from .masks import rle_decode

AREA_RANGES = {
    "all": (0, 1e10),
    "small": (0, 32 ** 2),
    "medium": (32 ** 2, 96 ** 2),
    "large": (96 ** 2, 1e10),
}


def _box_iou_matrix(dt, gt, iscrowd):
    """IoU [D, G] of [x, y, w, h] boxes; crowd GT uses inter/area_dt."""
    d = np.asarray(dt, dtype=np.float64)
    g = np.asarray(gt, dtype=np.float64)
    ious = np.zeros((len(d), len(g)))
    for j, gb in enumerate(g):
        gx1, gy1, gw, gh = gb
        gx2, gy2 = gx1 + gw, gy1 + gh
        garea = gw * gh
        for i, db in enumerate(d):
            dx1, dy1, dw, dh = db
            dx2, dy2 = dx1 + dw, dy1 + dh
            iw = min(dx2, gx2) - max(dx1, gx1)
            ih = min(dy2, gy2) - max(dy1, gy1)
            if iw <= 0 or ih <= 0:
                continue
            inter = iw * ih
            darea = dw * dh
            union = darea if iscrowd[j] else darea + garea - inter
            ious[i, j] = inter / max(union, 1e-10)
    return ious


def _mask_iou_matrix(dt, gt, iscrowd, decode_cache=None):
    # Fast path: all-RLE inputs go through the native run-walk IoU
    # (csrc/rle.cpp) without ever decoding to pixels.
    if all(isinstance(m, dict) for m in dt) and \
            all(isinstance(m, dict) for m in gt):
        from ..ops._ext import load_ext
        ext = load_ext()
        if ext is not None and hasattr(ext, "rle_iou_matrix"):
            import torch
            dts = [torch.tensor(m["counts"], dtype=torch.long)
                   for m in dt]
            gts = [torch.tensor(m["counts"], dtype=torch.long)
                   for m in gt]
            return ext.rle_iou_matrix(
                dts, gts,
                torch.tensor(iscrowd, dtype=torch.bool)).numpy()

    def dec(m):
        if isinstance(m, np.ndarray):
            return m
        if decode_cache is None:
            return rle_decode(m)
        key = id(m)
        cached = decode_cache.get(key)
        if cached is None:
            cached = rle_decode(m)
            decode_cache[key] = cached
        return cached

    dms = [dec(m) for m in dt]
    gms = [dec(m) for m in gt]
    dareas = [int(m.sum()) for m in dms]
    gareas = [int(m.sum()) for m in gms]
    ious = np.zeros((len(dms), len(gms)))
    for j, gm in enumerate(gms):
        for i, dm in enumerate(dms):
            # Quick reject via area bound before touching pixels.
            if dareas[i] == 0 or gareas[j] == 0:
                continue
            inter = np.logical_and(dm, gm).sum()
            if inter == 0:
                continue
            union = dareas[i] if iscrowd[j] \
                else dareas[i] + gareas[j] - inter
            ious[i, j] = inter / max(union, 1e-10)
    return ious


class COCOEvalLite:
    """Evaluate detection results against COCO-format ground truth.

    gt_anns: list of dicts {image_id, category_id, bbox [x,y,w,h],
             area, iscrowd, segmentation (RLE dict or mask, for segm)}
    dt_anns: list of dicts {image_id, category_id, bbox, score,
             segmentation (for segm)}
    """

    def __init__(self, gt_anns, dt_anns, iou_type="bbox",
                 iou_thrs=None, max_dets=100):
        self.iou_type = iou_type
        self.iou_thrs = iou_thrs if iou_thrs is not None else \
            np.round(np.arange(0.5, 1.0, 0.05), 2)
        self.max_dets = max_dets
        self.gt_by_img_cat = defaultdict(list)
        self.dt_by_img_cat = defaultdict(list)
        self.cats = set()
        self.imgs = set()
        for a in gt_anns:
            self.gt_by_img_cat[(a["image_id"], a["category_id"])].append(a)
            self.cats.add(a["category_id"])
            self.imgs.add(a["image_id"])
        for a in dt_anns:
            self.dt_by_img_cat[(a["image_id"], a["category_id"])].append(a)
            self.imgs.add(a["image_id"])
        # RLE decode cache: an annotation's mask is needed once per
        # (category, IoU-pass); decoding dominates segm eval without it.
        self._decode_cache: dict = {}

    def _match_image(self, img, cat, area_rng):
        """Returns (dt_scores, dt_matched [T, D], dt_ignored [T, D],
        n_gt) for one (image, class)."""
        gts = self.gt_by_img_cat.get((img, cat), [])
        dts = sorted(self.dt_by_img_cat.get((img, cat), []),
                     key=lambda d: -d["score"])[:self.max_dets]
        if not gts and not dts:
            return None
        lo, hi = area_rng
        gt_ignore = np.array(
            [bool(g.get("iscrowd", 0)) or not
             (lo <= g.get("area", g["bbox"][2] * g["bbox"][3]) < hi)
             for g in gts], dtype=bool)
        iscrowd = [int(g.get("iscrowd", 0)) for g in gts]
        if self.iou_type == "bbox":
            ious = _box_iou_matrix([d["bbox"] for d in dts],
                                   [g["bbox"] for g in gts], iscrowd)
        else:
            ious = _mask_iou_matrix([d["segmentation"] for d in dts],
                                    [g["segmentation"] for g in gts],
                                    iscrowd, self._decode_cache)
        T = len(self.iou_thrs)
        D = len(dts)
        G = len(gts)
        dt_matched = np.zeros((T, D), dtype=bool)
        dt_ignored = np.zeros((T, D), dtype=bool)
        gt_matched = np.zeros((T, G), dtype=bool)
        for t, thr in enumerate(self.iou_thrs):
            for di in range(D):
                best_iou = min(thr, 1 - 1e-10)
                best = -1
                for gi in range(G):
                    if gt_matched[t, gi] and not iscrowd[gi]:
                        continue
                    # Prefer non-ignored matches: once we have one, only
                    # better non-ignored GTs can replace it.
                    if best > -1 and not gt_ignore[best] \
                            and gt_ignore[gi]:
                        break
                    if ious[di, gi] < best_iou:
                        continue
                    best_iou = ious[di, gi]
                    best = gi
                if best == -1:
                    # Unmatched: ignore the det if outside area range.
                    d = dts[di]
                    darea = d.get("area", d["bbox"][2] * d["bbox"][3])
                    dt_ignored[t, di] = not (lo <= darea < hi)
                    continue
                gt_matched[t, best] = True
                dt_matched[t, di] = True
                dt_ignored[t, di] = gt_ignore[best]
        n_gt = int((~gt_ignore).sum())
        return (np.array([d["score"] for d in dts]), dt_matched,
                dt_ignored, n_gt)

    def _pr_for_cat(self, cat, area_rng):
        scores, matched, ignored = [], [], []
        n_gt = 0
        for img in self.imgs:
            r = self._match_image(img, cat, area_rng)
            if r is None:
                continue
            s, m, ig, g = r
            scores.append(s)
            matched.append(m)
            ignored.append(ig)
            n_gt += g
        if n_gt == 0:
            return None
        if scores:
            scores = np.concatenate(scores)
            matched = np.concatenate(matched, axis=1)
            ignored = np.concatenate(ignored, axis=1)
        else:
            scores = np.zeros(0)
            matched = np.zeros((len(self.iou_thrs), 0), dtype=bool)
            ignored = matched.copy()
        order = np.argsort(-scores, kind="mergesort")
        matched = matched[:, order]
        ignored = ignored[:, order]
        return matched, ignored, n_gt

    def _ap_at(self, matched, ignored, n_gt):
        """101-point interpolated AP per IoU threshold row."""
        T = matched.shape[0]
        aps = np.zeros(T)
        recalls = np.zeros(T)
        rec_thrs = np.linspace(0, 1, 101)
        for t in range(T):
            keep = ~ignored[t]
            tp = np.cumsum(matched[t][keep])
            fp = np.cumsum(~matched[t][keep])
            if len(tp) == 0:
                continue
            rc = tp / n_gt
            pr = tp / np.maximum(tp + fp, 1e-10)
            recalls[t] = rc[-1]
            # Monotone-decreasing precision envelope.
            for i in range(len(pr) - 1, 0, -1):
                pr[i - 1] = max(pr[i - 1], pr[i])
            inds = np.searchsorted(rc, rec_thrs, side="left")
            q = np.zeros(len(rec_thrs))
            valid = inds < len(pr)
            q[valid] = pr[inds[valid]]
            aps[t] = q.mean()
        return aps, recalls

    def evaluate(self, area="all"):
        """Returns a stats dict: AP, AP50, AP75, APs/m/l (when area=all),
        AR, per_class AP."""
        area_rng = AREA_RANGES[area]
        per_class = {}
        ap_rows = []
        ar_rows = []
        for cat in sorted(self.cats):
            r = self._pr_for_cat(cat, area_rng)
            if r is None:
                continue
            aps, recalls = self._ap_at(*r)
            per_class[cat] = float(aps.mean())
            ap_rows.append(aps)
            ar_rows.append(recalls)
        if not ap_rows:
            return {"AP": 0.0, "AP50": 0.0, "AP75": 0.0, "AR": 0.0,
                    "per_class": {}}
        ap_rows = np.stack(ap_rows)
        t50 = int(np.argmin(np.abs(self.iou_thrs - 0.5)))
        t75 = int(np.argmin(np.abs(self.iou_thrs - 0.75)))
        return {
            "AP": float(ap_rows.mean()),
            "AP50": float(ap_rows[:, t50].mean()),
            "AP75": float(ap_rows[:, t75].mean()),
            "AR": float(np.stack(ar_rows).mean()),
            "per_class": per_class,
        }

    def summarize(self, prefix=""):
        stats = self.evaluate()
        for k in ("AP", "AP50", "AP75", "AR"):
            print(f" {prefix}{self.iou_type} {k:5s} = {stats[k]:.3f}")
        return stats
