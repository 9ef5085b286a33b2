"""COCO-format dataset loader — pycocotools-free.

Replicates reference samples/coco/coco.py CocoDataset (P19): loads COCO
instance annotation JSON, maps class ids across sources, renders masks
from polygons / RLE, marks crowds with negative class ids. The RLE codecs
(including COCO's compressed LEB128-style string encoding) are in-repo.
"""

from __future__ import annotations

import json
import os
from collections import defaultdict

import numpy as np

from .dataset import Dataset


# ----------------------------- COCO compressed RLE (maskApi.c parity) ----

def rle_from_string(counts_str: str, size) -> dict:
    """Decode COCO's compressed RLE string into a counts list."""
    s = counts_str.encode("ascii") if isinstance(counts_str, str) \
        else counts_str
    counts = []
    p = 0
    while p < len(s):
        x = 0
        k = 0
        more = True
        while more:
            c = s[p] - 48
            x |= (c & 0x1F) << (5 * k)
            more = bool(c & 0x20)
            p += 1
            k += 1
            if not more and (c & 0x10):
                x |= -1 << (5 * k)  # sign extension
        if len(counts) > 2:
            x += counts[-2]
        counts.append(x)
    return {"size": list(size), "counts": counts}


def rle_to_string(rle: dict) -> str:
    """Encode a counts list as COCO's compressed RLE string."""
    counts = rle["counts"]
    out = bytearray()
    for i, x in enumerate(counts):
        if i > 2:
            x -= counts[i - 2]
        more = True
        while more:
            c = x & 0x1F
            x >>= 5
            more = (x != -1) if (c & 0x10) else (x != 0)
            if more:
                c |= 0x20
            out.append(c + 48)
    return out.decode("ascii")


def ann_to_mask(ann: dict, height: int, width: int) -> np.ndarray:
    """Render one annotation's segmentation to a [H, W] bool mask.

    Handles polygons (list of float lists), uncompressed RLE (counts
    list) and compressed RLE (counts string) — pycocotools annToMask
    parity.
    """
    from ..utils.masks import rle_decode

    seg = ann["segmentation"]
    if isinstance(seg, list):  # polygons
        from PIL import Image, ImageDraw
        img = Image.new("1", (width, height), 0)
        draw = ImageDraw.Draw(img)
        for poly in seg:
            pts = [(poly[i], poly[i + 1]) for i in range(0, len(poly), 2)]
            if len(pts) >= 3:
                draw.polygon(pts, outline=1, fill=1)
        return np.asarray(img, dtype=bool)
    if isinstance(seg, dict):
        counts = seg["counts"]
        if isinstance(counts, (str, bytes)):
            rle = rle_from_string(counts, seg["size"])
        else:
            rle = seg
        return rle_decode(rle)
    raise ValueError(f"unsupported segmentation type {type(seg)}")


class COCOIndex:
    """Minimal pycocotools.COCO equivalent over an instances JSON."""

    def __init__(self, annotation_file: str = None, dataset: dict = None):
        if dataset is None:
            with open(annotation_file) as f:
                dataset = json.load(f)
        self.dataset = dataset
        self.cats = {c["id"]: c for c in dataset.get("categories", [])}
        self.imgs = {i["id"]: i for i in dataset.get("images", [])}
        self.img_to_anns = defaultdict(list)
        self.anns = {}
        for a in dataset.get("annotations", []):
            self.img_to_anns[a["image_id"]].append(a)
            self.anns[a["id"]] = a

    def get_cat_ids(self, cat_names=None):
        if not cat_names:
            return sorted(self.cats)
        return [cid for cid, c in sorted(self.cats.items())
                if c["name"] in cat_names]

    def get_img_ids(self, cat_ids=None):
        if not cat_ids:
            return sorted(self.imgs)
        ids = set()
        for img_id, anns in self.img_to_anns.items():
            if any(a["category_id"] in cat_ids for a in anns):
                ids.add(img_id)
        return sorted(ids)

    def load_anns(self, img_id):
        return self.img_to_anns.get(img_id, [])


class CocoDataset(Dataset):
    def load_coco(self, dataset_dir, subset, year="2017", class_ids=None,
                  return_coco=False, annotation_file=None):
        """Load a COCO-format dataset.

        dataset_dir/annotations/instances_{subset}{year}.json +
        dataset_dir/{subset}{year}/ images, or an explicit
        annotation_file. (auto_download is not available offline —
        reference parity documented, not implemented.)
        """
        if annotation_file is None:
            annotation_file = os.path.join(
                dataset_dir, "annotations",
                f"instances_{subset}{year}.json")
        coco = COCOIndex(annotation_file)
        image_dir = os.path.join(dataset_dir, f"{subset}{year}")

        if not class_ids:
            class_ids = coco.get_cat_ids()
        if class_ids:
            image_ids = []
            for cid in class_ids:
                image_ids.extend(coco.get_img_ids(cat_ids=[cid]))
            image_ids = list(set(image_ids))
        else:
            image_ids = coco.get_img_ids()

        for i in class_ids:
            self.add_class("coco", i, coco.cats[i]["name"])
        for i in image_ids:
            info = coco.imgs[i]
            self.add_image(
                "coco", image_id=i,
                path=os.path.join(image_dir, info["file_name"]),
                width=info["width"], height=info["height"],
                annotations=[a for a in coco.load_anns(i)
                             if a["category_id"] in class_ids])
        if return_coco:
            return coco

    def load_mask(self, image_id):
        """[H, W, N] bool masks + class ids; crowds get negative ids and
        tiny masks are skipped (reference CocoDataset.load_mask)."""
        info = self.image_info[image_id]
        if info["source"] != "coco":
            return super().load_mask(image_id)
        instance_masks = []
        class_ids = []
        for ann in info["annotations"]:
            class_id = self.map_source_class_id(
                f"coco.{ann['category_id']}")
            if not class_id:
                continue
            m = ann_to_mask(ann, info["height"], info["width"])
            if m.max() < 1:
                continue
            if ann.get("iscrowd", 0):
                class_id *= -1
                # Crowd masks can be smaller than the image declared size.
                if m.shape != (info["height"], info["width"]):
                    full = np.zeros((info["height"], info["width"]),
                                    dtype=bool)
                    full[:m.shape[0], :m.shape[1]] = \
                        m[:info["height"], :info["width"]]
                    m = full
            instance_masks.append(m)
            class_ids.append(class_id)
        if class_ids:
            mask = np.stack(instance_masks, axis=2)
            return mask, np.array(class_ids, dtype=np.int32)
        return super().load_mask(image_id)

    def image_reference(self, image_id):
        info = self.image_info[image_id]
        if info["source"] == "coco":
            return f"coco id {info['id']}"
        return super().image_reference(image_id)


def build_coco_results(dataset, image_ids, rois, class_ids, scores, masks):
    """Detections -> COCO result dicts (reference build_coco_results)."""
    from ..utils.masks import rle_encode

    results = []
    for image_id in image_ids:
        for i in range(rois.shape[0]):
            class_id = class_ids[i]
            score = scores[i]
            bbox = np.around(rois[i], 1)
            mask = masks[:, :, i]
            rle = rle_encode(mask.astype(bool))
            results.append({
                "image_id": image_id,
                "category_id": dataset.get_source_class_id(class_id,
                                                           "coco"),
                "bbox": [bbox[1], bbox[0], bbox[3] - bbox[1],
                         bbox[2] - bbox[0]],
                "score": score,
                "segmentation": rle,
            })
    return results
