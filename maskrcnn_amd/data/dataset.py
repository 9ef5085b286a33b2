"""Dataset ABC.

Replicates reference utils.Dataset (mrcnn/utils.py:~240-450): add_class /
add_image / prepare() / class mapping across sources / load_image /
load_mask / image_reference. Images are RGB numpy arrays.
"""

from __future__ import annotations

import numpy as np


class Dataset:
    def __init__(self, class_map=None):
        self._image_ids = []
        self.image_info = []
        # Background is always the first class.
        self.class_info = [{"source": "", "id": 0, "name": "BG"}]
        self.source_class_ids = {}

    def add_class(self, source, class_id, class_name):
        assert "." not in source, "Source name cannot contain a dot"
        for info in self.class_info:
            if info["source"] == source and info["id"] == class_id:
                return
        self.class_info.append({"source": source, "id": class_id,
                                "name": class_name})

    def add_image(self, source, image_id, path, **kwargs):
        image_info = {"id": image_id, "source": source, "path": path}
        image_info.update(kwargs)
        self.image_info.append(image_info)

    def image_reference(self, image_id):
        return ""

    def prepare(self, class_map=None):
        """Build class/image lookup tables (reference prepare)."""
        def clean_name(name):
            return ",".join(name.split(",")[:1])

        self.num_classes = len(self.class_info)
        self.class_ids = np.arange(self.num_classes)
        self.class_names = [clean_name(c["name"]) for c in self.class_info]
        self.num_images = len(self.image_info)
        self._image_ids = np.arange(self.num_images)

        self.class_from_source_map = {
            f"{info['source']}.{info['id']}": idx
            for info, idx in zip(self.class_info, self.class_ids)}
        self.image_from_source_map = {
            f"{info['source']}.{info['id']}": idx
            for info, idx in zip(self.image_info, self._image_ids)}

        self.sources = list(set(i["source"] for i in self.class_info))
        self.source_class_ids = {}
        for source in self.sources:
            self.source_class_ids[source] = []
            for i, info in enumerate(self.class_info):
                if i == 0 or source == info["source"]:
                    self.source_class_ids[source].append(i)

    def map_source_class_id(self, source_class_id):
        return self.class_from_source_map[source_class_id]

    def get_source_class_id(self, class_id, source):
        info = self.class_info[class_id]
        assert info["source"] == source
        return info["id"]

    @property
    def image_ids(self):
        return self._image_ids

    def source_image_link(self, image_id):
        return self.image_info[image_id]["path"]

    def load_image(self, image_id):
        """Load an image as an RGB [H, W, 3] uint8 array."""
        from PIL import Image
        image = np.asarray(Image.open(self.image_info[image_id]["path"]))
        if image.ndim != 3:
            image = np.stack([image] * 3, axis=-1)
        if image.shape[-1] == 4:
            image = image[..., :3]
        return image

    def load_mask(self, image_id):
        """Override: return (masks [H, W, N] bool, class_ids [N] int32)."""
        return (np.empty([0, 0, 0], dtype=bool),
                np.empty([0], np.int32))
