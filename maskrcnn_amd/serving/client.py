"""Predict clients (gRPC + REST) with client-side mold/unmold.

Replicates reference S3/S4: the pre/post-processing lives
in the client — mold_inputs/get_anchors before the call,
unmold_detections after — so the server sees exactly the 3-input
signature. The gRPC path speaks the standard
tensorflow.serving.PredictionService/Predict method using the vendored
wire protos (no TF needed).
"""

from __future__ import annotations

from typing import Dict, List

import numpy as np

from ..config import Config
from ..utils import compose_image_meta, mold_image, resize_image
from ..utils.anchors import AnchorCache
from .wireproto import (ModelSpec, PredictRequest, PredictResponse,
                        numpy_to_tensor, tensor_to_numpy)

PREDICT_METHOD = "/tensorflow.serving.PredictionService/Predict"


class ServingClientBase:
    def __init__(self, config: Config, model_name: str = "mask_rcnn",
                 send_anchors: bool = True):
        self.config = config
        self.model_name = model_name
        self.send_anchors = send_anchors
        self.anchor_cache = AnchorCache(config)

    # --- mold/unmold (duplicated client-side like the reference clients)
    def mold_inputs(self, images):
        molded_images, image_metas, windows = [], [], []
        for image in images:
            molded, window, scale, padding, crop = resize_image(
                image, min_dim=self.config.IMAGE_MIN_DIM,
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
        return (np.stack(molded_images).astype(np.float32),
                np.stack(image_metas).astype(np.float32),
                np.stack(windows))

    def unmold(self, outputs: Dict[str, np.ndarray], images,
               molded_shape, windows):
        from ..model.mask_rcnn import MaskRCNN
        results = []
        detections = outputs["mrcnn_detection"]
        masks_tf = outputs["mrcnn_mask"]  # [B, D, mh, mw, C]
        masks = np.transpose(masks_tf, (0, 1, 4, 2, 3))  # torch layout
        for i, image in enumerate(images):
            boxes, class_ids, scores, full_masks = \
                MaskRCNN.unmold_detections(
                    detections[i], masks[i], image.shape,
                    molded_shape, windows[i])
            results.append({"rois": boxes, "class_ids": class_ids,
                            "scores": scores, "masks": full_masks})
        return results

    def predict_raw(self, images_np, metas_np, anchors_np):
        raise NotImplementedError

    def detect(self, images: List[np.ndarray]):
        """Full round trip: mold -> Predict -> unmold."""
        molded, metas, windows = self.mold_inputs(images)
        anchors = None
        if self.send_anchors:
            anchors = self.anchor_cache.get(molded[0].shape).numpy() \
                .astype(np.float32)[None]
        outputs = self.predict_raw(molded, metas, anchors)
        return self.unmold(outputs, images, molded[0].shape, windows)


class GrpcServingClient(ServingClientBase):
    def __init__(self, target: str = "localhost:8500",
                 config: Config = None, model_name: str = "mask_rcnn",
                 send_anchors: bool = True, timeout: float = 60.0):
        super().__init__(config, model_name, send_anchors)
        import grpc
        self.timeout = timeout
        self.channel = grpc.insecure_channel(
            target,
            options=[("grpc.max_receive_message_length",
                      256 * 1024 * 1024),
                     ("grpc.max_send_message_length", 256 * 1024 * 1024)])
        self._predict = self.channel.unary_unary(
            PREDICT_METHOD,
            request_serializer=lambda r: r.encode(),
            response_deserializer=PredictResponse.decode)

    def predict_raw(self, images_np, metas_np, anchors_np):
        req = PredictRequest(
            model_spec=ModelSpec(name=self.model_name,
                                 signature_name="serving_default"))
        req.inputs["input_image"] = numpy_to_tensor(images_np)
        req.inputs["input_image_meta"] = numpy_to_tensor(metas_np)
        if anchors_np is not None:
            req.inputs["input_anchors"] = numpy_to_tensor(anchors_np)
        resp = self._predict(req, timeout=self.timeout)
        return {k: tensor_to_numpy(t) for k, t in resp.outputs.items()}


class RestServingClient(ServingClientBase):
    def __init__(self, base_url: str = "http://localhost:8501",
                 config: Config = None, model_name: str = "mask_rcnn",
                 send_anchors: bool = True, timeout: float = 60.0):
        super().__init__(config, model_name, send_anchors)
        self.base_url = base_url.rstrip("/")
        self.timeout = timeout

    def predict_raw(self, images_np, metas_np, anchors_np):
        import requests
        inputs = {
            "input_image": images_np.tolist(),
            "input_image_meta": metas_np.tolist(),
        }
        if anchors_np is not None:
            inputs["input_anchors"] = anchors_np.tolist()
        url = (f"{self.base_url}/v1/models/"
               f"{self.model_name}:predict")
        r = requests.post(url, json={
            "signature_name": "serving_default", "inputs": inputs},
            timeout=self.timeout)
        r.raise_for_status()
        out = r.json()["outputs"]
        return {k: np.asarray(v, dtype=np.float32)
                for k, v in out.items()}


class ModelServiceClient:
    """tensorflow.serving.ModelService client: model status + runtime
    config reload (the tensorflow_serving.apis model_service_pb2_grpc
    surface, spoken over the hand-vendored wire protos)."""

    def __init__(self, target: str = "localhost:8500",
                 timeout: float = 30.0):
        import grpc

        from .wireproto import (GetModelStatusResponse,
                                ReloadConfigResponse)
        self.timeout = timeout
        self.channel = grpc.insecure_channel(target)
        self._status = self.channel.unary_unary(
            "/tensorflow.serving.ModelService/GetModelStatus",
            request_serializer=lambda r: r.encode(),
            response_deserializer=GetModelStatusResponse.decode)
        self._reload = self.channel.unary_unary(
            "/tensorflow.serving.ModelService/HandleReloadConfigRequest",
            request_serializer=lambda r: r.encode(),
            response_deserializer=ReloadConfigResponse.decode)

    def get_model_status(self, model_name: str):
        """Returns a list of {'version', 'state'} dicts."""
        from .wireproto import GetModelStatusRequest, ModelSpec
        resp = self._status(GetModelStatusRequest(
            model_spec=ModelSpec(name=model_name)), timeout=self.timeout)
        return [{"version": s.version, "state": s.state}
                for s in resp.model_version_status]

    def reload_config(self, model_configs):
        """model_configs: [{'name', 'base_path'}, ...]. Raises
        RuntimeError on a rejected reload."""
        from .wireproto import ModelConfigProto, ReloadConfigRequest
        resp = self._reload(ReloadConfigRequest(model_configs=[
            ModelConfigProto(name=c["name"], base_path=c["base_path"])
            for c in model_configs]), timeout=self.timeout)
        if resp.status.error_code != 0:
            raise RuntimeError(
                f"reload rejected ({resp.status.error_code}): "
                f"{resp.status.error_message}")

    def close(self):
        self.channel.close()
