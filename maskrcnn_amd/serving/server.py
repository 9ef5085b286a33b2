"""ModelServer: gRPC (:8500) + REST (:8501) Predict with version hot-reload.

The MI355X-native replacement for TF-Serving's C++ ModelServer
(SURVEY.md N11): PredictionService/Predict served via grpc generic
handlers over the hand-vendored wire protos (wireproto.py), TF-Serving
REST JSON schema via FastAPI/uvicorn, a FileSystemStoragePathSource-style
poller that hot-loads the highest numeric version directory, per-GPU
engines behind the dynamic batcher, and Prometheus metrics.
"""

from __future__ import annotations

import logging
import os
import threading
import time
from typing import Dict, Optional

import numpy as np

from .batcher import DynamicBatcher
from .engine import InferenceEngine
from .savedmodel import find_latest_version
from .savedmodel import build_signature
from .wireproto import (GetModelMetadataRequest, GetModelMetadataResponse,
                        ModelSpec, PredictRequest, PredictResponse,
                        numpy_to_tensor, tensor_to_numpy)

log = logging.getLogger(__name__)

try:
    from prometheus_client import (Counter, Histogram, Gauge,
                                   start_http_server)
    _PROM = True
    REQS = Counter("predict_requests_total", "Predict requests",
                   ["proto"])
    ERRS = Counter("predict_errors_total", "Predict errors", ["proto"])
    REJECTS = Counter("predict_rejected_total",
                      "Predicts shed by the full batching queue "
                      "(RESOURCE_EXHAUSTED/429)", ["proto"])
    LAT = Histogram("predict_latency_seconds", "Predict latency",
                    buckets=(.005, .01, .025, .05, .1, .25, .5, 1, 2.5, 5))
    BATCH = Histogram("predict_batch_size", "executed batch size",
                      buckets=(1, 2, 4, 8, 16, 32, 64))
    VERSION = Gauge("model_version", "loaded model version")
    HBM = Gauge("hbm_bytes_allocated",
                "torch.cuda.memory_allocated across replicas")
except ImportError:  # pragma: no cover
    _PROM = False


class ModelState:
    """One served model: engine(s) + batcher(s) + version, hot-swappable.

    devices=["cuda:0", ..., "cuda:7"] serves one REPLICA per GPU behind
    one front door (BASELINE config #4's 8-replica mode; serving DP needs
    no collectives — SURVEY §2.4). Requests round-robin to the replica
    with the shortest queue.
    """

    def __init__(self, name: str, base_path: str, device=None,
                 devices=None, max_batch_size: int = 32,
                 batch_timeout_micros: int = 2000, config=None,
                 version_policy: str = "latest", version_labels=None,
                 enable_warmup: bool = True):
        """version_policy: 'latest' (TF-Serving default — new version
        replaces old) or 'all' (every discovered version stays
        servable; requests pin via model_spec.version).
        version_labels: {'stable': 1, 'canary': 2} — clients address
        ModelSpec.version_label (the TF-Serving canary mechanism;
        implies the labeled versions must be loaded, so usually
        combined with version_policy='all')."""
        assert version_policy in ("latest", "all"), version_policy
        self.name = name
        self.version_labels = dict(version_labels or {})
        self.enable_warmup = enable_warmup
        self.base_path = base_path
        self.devices = list(devices) if devices else [device]
        self.max_batch_size = max_batch_size
        self.batch_timeout_micros = batch_timeout_micros
        self.config_override = config
        self.version_policy = version_policy
        self.version: Optional[int] = None
        #: version -> {"engines": [...], "batchers": [...]}
        self.versions: Dict[int, dict] = {}
        self._rr = 0
        self._lock = threading.Lock()

    # Latest-version accessors (single-version compat).
    @property
    def engines(self):
        v = self.versions.get(self.version)
        return v["engines"] if v else []

    @property
    def batchers(self):
        v = self.versions.get(self.version)
        return v["batchers"] if v else []

    # Single-replica compat accessors.
    @property
    def engine(self):
        return self.engines[0] if self.engines else None

    @property
    def batcher(self):
        return self.batchers[0] if self.batchers else None

    def _load_version(self, version: int):
        export_dir = os.path.join(self.base_path, str(version))
        log.info("loading %s version %d from %s on %s", self.name,
                 version, export_dir, self.devices)
        engines = [InferenceEngine.from_saved_model(
            export_dir, device=dev, config=self.config_override)
            for dev in self.devices]
        batchers = [DynamicBatcher(
            eng, max_batch_size=self.max_batch_size,
            batch_timeout_micros=self.batch_timeout_micros)
            for eng in engines]
        # TF-Serving model_warmup: replay assets.extra/
        # tf_serving_warmup_requests (TFRecord of PredictionLogs)
        # through every replica BEFORE the version goes AVAILABLE, so
        # the first live request pays no MIOpen-Find/first-launch cost.
        try:
            from .warmup import read_warmup_requests
            from .wireproto import tensor_to_numpy
            reqs = read_warmup_requests(export_dir) \
                if self.enable_warmup else []
            for i, wreq in enumerate(reqs):
                images = tensor_to_numpy(wreq.inputs["input_image"])
                if images.dtype != np.uint8:
                    images = images.astype(np.float32)
                metas = tensor_to_numpy(
                    wreq.inputs["input_image_meta"]).astype(np.float32)
                anchors = None
                if "input_anchors" in wreq.inputs:
                    anchors = tensor_to_numpy(
                        wreq.inputs["input_anchors"]).astype(np.float32)
                for eng in engines:
                    eng.predict(images, metas, anchors)
            if reqs:
                log.info("replayed %d warmup request(s) for %s v%d",
                         len(reqs), self.name, version)
        except Exception:  # noqa: BLE001 — warmup must never block load
            log.exception("model warmup replay failed (%s v%d)",
                          self.name, version)
        return {"engines": engines, "batchers": batchers}

    def maybe_reload(self) -> bool:
        from .savedmodel import list_versions
        # 'all': every on-disk version is servable (TF-Serving
        # model_version_policy {all{}} — including at startup);
        # 'latest': only the newest.
        on_disk = list_versions(self.base_path)
        if not on_disk:
            return False
        want = on_disk if self.version_policy == "all" else [on_disk[-1]]
        new = {v: self._load_version(v) for v in want
               if v not in self.versions}
        latest = on_disk[-1]
        if not new and latest == self.version:
            return False
        with self._lock:
            self.versions.update(new)
            old = None
            if self.version_policy == "latest" and \
                    self.version is not None and self.version != latest:
                old = self.versions.pop(self.version, None)
            self.version = latest
        if old:
            for b in old["batchers"]:
                b.close()
        if _PROM:
            VERSION.set(latest)
        return True

    def predict(self, images, metas, anchors=None, outputs=None,
                version=None):
        # Retry on the submit-vs-hot-swap race: maybe_reload() may close
        # the batcher we just picked; re-pick from the fresh set (the
        # CURRENT latest for unpinned requests).
        explicit = version is not None
        for _attempt in range(3):
            with self._lock:
                want = version if explicit else self.version
                slot = self.versions.get(want)
                if slot is None:
                    if explicit:
                        raise KeyError(
                            f"model {self.name!r} version {version} not "
                            f"loaded (serving {sorted(self.versions)})")
                    raise RuntimeError(
                        f"model {self.name} has no loaded version")
                bl = slot["batchers"]
                # Shortest-queue replica (round-robin tiebreak).
                qsizes = [b.queue.qsize() for b in bl]
                best = min(range(len(qsizes)),
                           key=lambda i: (qsizes[i], (i - self._rr) %
                                          len(qsizes)))
                self._rr = (best + 1) % len(bl)
                batcher = bl[best]
                version = want
            try:
                fut = batcher.submit(images, metas, anchors,
                                     outputs=outputs)
            except RuntimeError as e:
                from .batcher import BatcherQueueFull
                if isinstance(e, BatcherQueueFull):
                    raise  # load shedding, not a hot-swap race
                continue
            try:
                return fut.result(timeout=300.0), version
            except RuntimeError as e:
                if "batcher closed" in str(e):
                    continue  # drained away by a reload: resubmit
                raise
        raise RuntimeError("predict failed: batcher kept closing")

    def warmup(self, image_shape=None):
        """Run one dummy max-batch predict per replica so MIOpen Find /
        algo caches are hot before traffic (TF-Serving model_warmup
        equivalent)."""
        for eng in self.engines:
            H, W, _ = image_shape or eng.config.IMAGE_SHAPE
            B = self.max_batch_size
            images = np.zeros((B, H, W, 3), np.float32)
            metas = np.zeros((B, eng.config.IMAGE_META_SIZE), np.float32)
            metas[:, 4:7] = (H, W, 3)
            metas[:, 7:11] = (0, 0, H, W)
            eng.predict(images, metas, None)


class ModelServer:
    def __init__(self, model_name: str = None, model_base_path: str = None,
                 grpc_port: int = 8500, rest_port: int = 8501,
                 metrics_port: Optional[int] = None, device=None,
                 devices=None, max_batch_size: int = 32,
                 batch_timeout_micros: int = 2000,
                 poll_interval_s: float = 5.0, config=None,
                 model_configs=None, version_policy: str = "latest",
                 grpc_ingress: str = "python",
                 enable_model_warmup: bool = True):
        """Single model (model_name + model_base_path, TF-Serving's
        --model_name/--model_base_path mode) or several (model_configs =
        parse_model_config(models.config), the --model_config_file
        mode). All models share the batching parameters and devices."""
        if model_configs is None:
            if model_name is None or model_base_path is None:
                raise ValueError("model_name+model_base_path or "
                                 "model_configs required")
            model_configs = [{"name": model_name,
                              "base_path": model_base_path}]
        self._model_kwargs = dict(
            device=device, devices=devices,
            max_batch_size=max_batch_size,
            batch_timeout_micros=batch_timeout_micros, config=config,
            version_policy=version_policy,
            enable_warmup=enable_model_warmup)
        self.models: Dict[str, ModelState] = {
            mc["name"]: ModelState(
                mc["name"], mc["base_path"],
                **{**self._model_kwargs,
                   **{k: mc[k] for k in ("version_policy",
                                         "version_labels") if k in mc}})
            for mc in model_configs}
        self.grpc_port = grpc_port
        self.rest_port = rest_port
        self.metrics_port = metrics_port
        self.poll_interval_s = poll_interval_s
        # --model_config_file_poll_wait_seconds support (set by main()
        # or callers; 0 = only reload via the ModelService RPC).
        self.model_config_path: Optional[str] = None
        self.config_poll_interval_s: float = 0.0
        self._last_config_poll = 0.0
        if grpc_ingress not in ("python", "cpp"):
            raise ValueError(f"grpc_ingress must be python|cpp, "
                             f"got {grpc_ingress!r}")
        self.grpc_ingress = grpc_ingress
        self._stop = threading.Event()
        self._grpc_server = None
        self._rest_thread = None
        self._uvicorn_server = None

    @property
    def state(self) -> ModelState:
        """The first (default) model — single-model back-compat."""
        return next(iter(self.models.values()))

    def _get_state(self, name: Optional[str]) -> ModelState:
        if not name:
            return self.state
        st = self.models.get(name)
        if st is None:
            raise KeyError(f"unknown model {name!r}")
        return st

    def reload_config(self, model_configs) -> None:
        """Apply a new served-model set (the ModelService
        HandleReloadConfigRequest semantics): load additions, keep
        matching entries, unload (close) removed models."""
        new: Dict[str, ModelState] = {}
        for mc in model_configs:
            cur = self.models.get(mc["name"])
            if cur is not None and cur.base_path == mc["base_path"]:
                new[mc["name"]] = cur
            else:
                st = ModelState(mc["name"], mc["base_path"],
                                **{**self._model_kwargs,
                                   **{k: mc[k]
                                      for k in ("version_policy",
                                                "version_labels")
                                      if k in mc}})
                st.maybe_reload()
                if st.version is None:
                    raise FileNotFoundError(
                        f"no servable version under {mc['base_path']}")
                new[mc["name"]] = st
        kept = set(map(id, new.values()))
        removed = [s for s in self.models.values() if id(s) not in kept]
        self.models = new
        for st in removed:
            for slot in st.versions.values():  # every loaded version
                for b in slot["batchers"]:
                    b.close()

    # ----------------------------------------------------------- predict
    def _handle_predict(self, request: PredictRequest) -> PredictResponse:
        t0 = time.monotonic()
        if _PROM:
            REQS.labels("grpc").inc()
        state = self._get_state(request.model_spec.name)
        name = state.name
        sig = request.model_spec.signature_name or "serving_default"
        if sig != "serving_default":
            raise KeyError(f"unknown signature {sig!r}")
        want_version = request.model_spec.version
        label = getattr(request.model_spec, "version_label", "")
        if want_version is None and label:
            # TF-Serving version_labels routing ('stable'/'canary').
            if label not in state.version_labels:
                raise KeyError(
                    f"model {name!r} has no version label {label!r} "
                    f"(labels: {sorted(state.version_labels)})")
            want_version = state.version_labels[label]
        if want_version is not None and want_version not in state.versions:
            raise KeyError(
                f"model {name!r} version {want_version} not loaded "
                f"(serving {sorted(state.versions) or state.version})")
        try:
            images = tensor_to_numpy(request.inputs["input_image"])
            if images.dtype != np.uint8:
                # fp32 = the stock molded contract; uint8 = wire/PCIe
                # extension, molded on the GPU (engine.predict).
                images = images.astype(np.float32)
            metas = tensor_to_numpy(request.inputs["input_image_meta"]) \
                .astype(np.float32)
        except KeyError as e:
            raise KeyError(f"missing input {e}")
        anchors = None
        if "input_anchors" in request.inputs:
            anchors = tensor_to_numpy(
                request.inputs["input_anchors"]).astype(np.float32)
        # The filter is pushed into the engine: unrequested outputs skip
        # the DtoH transfer entirely.
        #
        # Extension (wire-compatible): output_filter entry
        # "mrcnn_mask_top" returns the DETECTED-CLASS mask only,
        # [B, D, 28, 28] — what unmold_detections actually consumes —
        # instead of the full [B, D, 28, 28, C] tensor (81x fewer bytes
        # at 81 COCO classes; the full mrcnn_mask stays available for
        # stock TF-Serving clients).
        # mrcnn_mask_top is resolved INSIDE the engine (GPU-side class
        # gather before DtoH — 81x fewer transfer bytes); the engine
        # needs mrcnn_detection listed so clients that filtered it out
        # still work, and it ignores unknown names.
        want = list(request.output_filter) or None
        if want and "mrcnn_mask_top" in want and \
                "mrcnn_detection" not in want:
            want = want + ["mrcnn_detection"]
        outputs, version = state.predict(
            images, metas, anchors, outputs=want,
            version=want_version)
        if request.output_filter:
            outputs = {k: v for k, v in outputs.items()
                       if k in request.output_filter}
        resp = PredictResponse(
            outputs={k: numpy_to_tensor(v) for k, v in outputs.items()},
            model_spec=ModelSpec(name=name, version=version,
                                 signature_name="serving_default"))
        if _PROM:
            LAT.observe(time.monotonic() - t0)
            BATCH.observe(images.shape[0])
            import torch
            if torch.cuda.is_available():
                HBM.set(sum(torch.cuda.memory_allocated(d)
                            for d in range(torch.cuda.device_count())))
        return resp

    # ------------------------------------------- context-free handlers
    # Shared by the grpc-python and native-C++ ingresses. Raise
    # KeyError for NOT_FOUND; anything else maps to INTERNAL.
    def _handle_metadata(self, request: GetModelMetadataRequest):
        state = self._get_state(request.model_spec.name)
        eng = state.engine
        sigs = {"serving_default": build_signature(eng.config)} \
            if eng is not None else {}
        return GetModelMetadataResponse(
            model_spec=ModelSpec(name=state.name, version=state.version),
            signature_defs=sigs)

    def _handle_status(self, request):
        from .wireproto import (GetModelStatusResponse, ModelVersionStatus,
                                STATE_AVAILABLE)
        state = self._get_state(request.model_spec.name)
        versions = sorted(state.versions)
        want = request.model_spec.version
        if want is not None:
            # TF-Serving: a pinned model_spec.version reports only that
            # version (unknown -> NOT_FOUND).
            if want not in state.versions:
                raise KeyError(
                    f"model {state.name!r} version {want} not found "
                    f"(serving {versions})")
            versions = [want]
        statuses = [ModelVersionStatus(version=v, state=STATE_AVAILABLE)
                    for v in versions] or \
            [ModelVersionStatus(version=0, state=0)]
        return GetModelStatusResponse(model_version_status=statuses)

    def _handle_reload(self, request):
        from .wireproto import ReloadConfigResponse, StatusProto
        try:
            self.reload_config(
                [{"name": c.name, "base_path": c.base_path}
                 for c in request.model_configs])
            return ReloadConfigResponse(status=StatusProto(0, ""))
        except Exception as e:  # noqa: BLE001
            log.exception("reload config failed")
            return ReloadConfigResponse(
                status=StatusProto(13, str(e)))  # 13 = INTERNAL

    # -------------------------------------------------------------- grpc
    def _build_grpc(self):
        import grpc
        from concurrent import futures as cf

        outer = self

        from .batcher import BatcherQueueFull

        def _guard(fn):
            def call(request, context):
                # Don't start model work for an already-expired RPC
                # (same shedding the cpp ingress does for queued
                # requests past their grpc-timeout). OUTSIDE the try:
                # context.abort raises to unwind, and the broad handler
                # below must not catch-and-abort a second time.
                tr = context.time_remaining()
                if tr is not None and tr <= 0:
                    if _PROM:
                        REJECTS.labels("grpc").inc()
                    context.abort(grpc.StatusCode.DEADLINE_EXCEEDED,
                                  "deadline expired before dispatch")
                try:
                    return fn(request)
                except KeyError as e:
                    if _PROM:
                        ERRS.labels("grpc").inc()
                    context.abort(grpc.StatusCode.NOT_FOUND, str(e))
                except BatcherQueueFull as e:
                    if _PROM:
                        REJECTS.labels("grpc").inc()
                    context.abort(grpc.StatusCode.RESOURCE_EXHAUSTED,
                                  str(e))
                except Exception as e:  # noqa: BLE001
                    if _PROM:
                        ERRS.labels("grpc").inc()
                    log.exception("%s failed", fn.__name__)
                    context.abort(grpc.StatusCode.INTERNAL, str(e))
            return call

        predict = _guard(outer._handle_predict)
        get_model_metadata = _guard(outer._handle_metadata)
        get_model_status = _guard(outer._handle_status)
        handle_reload_config = _guard(outer._handle_reload)

        from .wireproto import (GetModelStatusRequest, ReloadConfigRequest)

        handler = grpc.method_handlers_generic_handler(
            "tensorflow.serving.PredictionService",
            {"Predict": grpc.unary_unary_rpc_method_handler(
                predict,
                request_deserializer=PredictRequest.decode,
                response_serializer=lambda r: r.encode()),
             "GetModelMetadata": grpc.unary_unary_rpc_method_handler(
                get_model_metadata,
                request_deserializer=GetModelMetadataRequest.decode,
                response_serializer=lambda r: r.encode())})
        model_svc = grpc.method_handlers_generic_handler(
            "tensorflow.serving.ModelService",
            {"GetModelStatus": grpc.unary_unary_rpc_method_handler(
                get_model_status,
                request_deserializer=GetModelStatusRequest.decode,
                response_serializer=lambda r: r.encode()),
             "HandleReloadConfigRequest":
                 grpc.unary_unary_rpc_method_handler(
                     handle_reload_config,
                     request_deserializer=ReloadConfigRequest.decode,
                     response_serializer=lambda r: r.encode())})
        server = grpc.server(
            cf.ThreadPoolExecutor(max_workers=16),
            options=[("grpc.max_receive_message_length", 256 * 1024 * 1024),
                     ("grpc.max_send_message_length", 256 * 1024 * 1024)])
        server.add_generic_rpc_handlers((handler, model_svc))
        bound = server.add_insecure_port(f"[::]:{self.grpc_port}")
        self.grpc_port = bound
        return server

    # ------------------------------------------------------ cpp ingress
    def _build_cpp_ingress(self):
        """Native HTTP/2 gRPC front-end (csrc/ingress.cpp).

        The TF-Serving analog of the C++ grpc daemon:
        sockets, HTTP/2 framing, HPACK and flow control run in C++
        threads with no GIL; Python is entered once per complete
        request for wireproto decode -> handler -> encode.
        """
        from .. import _ingress
        from .wireproto import (GetModelStatusRequest, ReloadConfigRequest)

        svc = "/tensorflow.serving."
        methods = {
            svc + "PredictionService/Predict":
                (PredictRequest.decode, self._handle_predict),
            svc + "PredictionService/GetModelMetadata":
                (GetModelMetadataRequest.decode, self._handle_metadata),
            svc + "ModelService/GetModelStatus":
                (GetModelStatusRequest.decode, self._handle_status),
            svc + "ModelService/HandleReloadConfigRequest":
                (ReloadConfigRequest.decode, self._handle_reload),
        }

        from .batcher import BatcherQueueFull

        def dispatch(path, payload):
            ent = methods.get(path)
            if ent is None:
                return (12, f"unknown method {path!r}", b"")
            decode, fn = ent
            try:
                return (0, "", fn(decode(payload)).encode())
            except KeyError as e:
                if _PROM:
                    ERRS.labels("grpc").inc()
                return (5, str(e), b"")  # NOT_FOUND
            except BatcherQueueFull as e:
                if _PROM:
                    REJECTS.labels("grpc").inc()
                return (8, str(e), b"")  # RESOURCE_EXHAUSTED
            except Exception as e:  # noqa: BLE001
                if _PROM:
                    ERRS.labels("grpc").inc()
                log.exception("%s failed (cpp ingress)", path)
                return (13, str(e), b"")  # INTERNAL

        srv = _ingress.IngressServer(self.grpc_port, dispatch, 16)
        return srv

    # -------------------------------------------------------------- rest
    def _build_rest_app(self):
        from fastapi import Body, FastAPI
        from fastapi.responses import JSONResponse

        app = FastAPI(title="maskrcnn_amd model server")
        outer = self

        @app.get("/v1/models/{model_name}/metadata")
        @app.get("/v1/models/{model_name}/versions/{version}/metadata")
        def model_metadata(model_name: str, version: int = None):
            state = outer.models.get(model_name)
            if state is None or state.engine is None:
                return JSONResponse({"error": "model not found"},
                                    status_code=404)
            if version is not None and version not in state.versions:
                return JSONResponse(
                    {"error": f"version {version} not loaded"},
                    status_code=404)
            sig = build_signature(state.engine.config)
            def ti(t):
                return {"dtype": "DT_FLOAT",
                        "tensor_shape": {"dim": [{"size": str(d)}
                                                 for d in t.shape]},
                        "name": t.name}
            return {"model_spec": {"name": state.name,
                                   "version": str(
                                       version if version is not None
                                       else state.version)},
                    "metadata": {"signature_def": {"signature_def": {
                        "serving_default": {
                            "inputs": {k: ti(v)
                                       for k, v in sig.inputs.items()},
                            "outputs": {k: ti(v)
                                        for k, v in sig.outputs.items()},
                            "method_name": sig.method_name}}}}}

        @app.get("/monitoring/prometheus/metrics")
        def prometheus_metrics():
            # TF-Serving's --monitoring_config_file path on :8501.
            if not _PROM:
                return JSONResponse({"error": "prometheus_client "
                                     "not installed"}, status_code=501)
            from prometheus_client import (CONTENT_TYPE_LATEST,
                                           generate_latest)
            from fastapi.responses import Response as _Resp
            return _Resp(generate_latest(),
                         media_type=CONTENT_TYPE_LATEST)

        @app.get("/v1/models/{model_name}/versions/{version}")
        def version_status(model_name: str, version: int):
            state = outer.models.get(model_name)
            ok = state is not None and version in state.versions
            return {"model_version_status": [{
                "version": str(version),
                "state": "AVAILABLE" if ok else "UNAVAILABLE",
                "status": {"error_code": "OK" if ok else "NOT_FOUND",
                           "error_message": ""}}]}

        @app.get("/v1/models/{model_name}")
        def model_status(model_name: str):
            state = outer.models.get(model_name)
            ok = state is not None and state.version is not None
            if not ok:
                return {"model_version_status": [{
                    "version": "0", "state": "UNAVAILABLE",
                    "status": {"error_code": "NOT_FOUND",
                               "error_message": ""}}]}
            return {"model_version_status": [{
                "version": str(v), "state": "AVAILABLE",
                "status": {"error_code": "OK", "error_message": ""},
            } for v in sorted(state.versions)]}

        @app.post("/v1/models/{model_and_verb:path}")
        def predict(model_and_verb: str, body: dict = Body(...)):
            # Sync handler ON PURPOSE: FastAPI runs it in its
            # threadpool, so one slow batch (fut.result blocks up to
            # minutes on a cold Find) cannot stall the event loop for
            # every other REST request.
            if _PROM:
                REQS.labels("rest").inc()
            if not model_and_verb.endswith(":predict"):
                return JSONResponse({"error": "unsupported verb"},
                                    status_code=404)
            name = model_and_verb[:-len(":predict")]
            want_version = None
            want_label = None
            if "/versions/" in name:   # /v1/models/{m}/versions/{v}:predict
                name, _, v = name.partition("/versions/")
                try:
                    want_version = int(v)
                except ValueError:
                    return JSONResponse(
                        {"error": f"bad version {v!r}"}, status_code=400)
            elif "/labels/" in name:   # /v1/models/{m}/labels/{l}:predict
                name, _, want_label = name.partition("/labels/")
            state = outer.models.get(name)
            if state is None:
                return JSONResponse(
                    {"error": f"model {name} not found"}, status_code=404)
            if want_label is not None:
                if want_label not in state.version_labels:
                    return JSONResponse(
                        {"error": f"model {name} has no version label "
                                  f"{want_label!r}"}, status_code=404)
                want_version = state.version_labels[want_label]
            if want_version is not None and \
                    want_version not in state.versions:
                return JSONResponse(
                    {"error": f"version {want_version} of {name} not "
                              f"loaded"}, status_code=404)
            sig = body.get("signature_name", "serving_default")
            if sig != "serving_default":
                return JSONResponse(
                    {"error": f"unknown signature {sig!r}"},
                    status_code=400)
            try:
                if "inputs" in body:       # columnar format
                    ins = body["inputs"]
                    images = np.asarray(ins["input_image"],
                                        dtype=np.float32)
                    metas = np.asarray(ins["input_image_meta"],
                                       dtype=np.float32)
                    anchors = (np.asarray(ins["input_anchors"],
                                          dtype=np.float32)
                               if "input_anchors" in ins else None)
                elif "instances" in body:  # row format
                    rows = body["instances"]
                    images = np.stack([np.asarray(r["input_image"],
                                                  dtype=np.float32)
                                       for r in rows])
                    metas = np.stack([np.asarray(r["input_image_meta"],
                                                 dtype=np.float32)
                                      for r in rows])
                    anchors = None
                    if rows and "input_anchors" in rows[0]:
                        anchors = np.asarray(rows[0]["input_anchors"],
                                             dtype=np.float32)[None]
                else:
                    return JSONResponse(
                        {"error": "missing inputs/instances"},
                        status_code=400)
                # Extension (mirrors the gRPC output_filter): restrict
                # which outputs are computed AND serialized — a full
                # 7-output JSON response tolist()s the 33 MB mask
                # tensor. Supports mrcnn_mask_top like gRPC.
                want = body.get("output_filter")
                if want is not None and not isinstance(want, list):
                    return JSONResponse(
                        {"error": "output_filter must be a list"},
                        status_code=400)
                eng_want = want
                if want and "mrcnn_mask_top" in want and \
                        "mrcnn_detection" not in want:
                    eng_want = want + ["mrcnn_detection"]
                t0 = time.monotonic()
                outputs, _ = state.predict(images, metas, anchors,
                                           version=want_version,
                                           outputs=eng_want)
                if want:
                    outputs = {k: v for k, v in outputs.items()
                               if k in want}
                if _PROM:
                    LAT.observe(time.monotonic() - t0)
                if "instances" in body:
                    # TF REST row schema: "instances" requests answer
                    # with "predictions" — one dict per input row.
                    B = images.shape[0]
                    preds = [{k: v[i].tolist()
                              for k, v in outputs.items()}
                             for i in range(B)]
                    return {"predictions": preds}
                return {"outputs": {k: v.tolist()
                                    for k, v in outputs.items()}}
            except Exception as e:  # noqa: BLE001
                if _PROM:
                    ERRS.labels("rest").inc()
                from .batcher import BatcherQueueFull
                if isinstance(e, BatcherQueueFull):
                    if _PROM:
                        REJECTS.labels("rest").inc()
                    return JSONResponse({"error": str(e)},
                                        status_code=429)
                log.exception("REST predict failed")
                return JSONResponse({"error": str(e)}, status_code=500)

        return app

    # --------------------------------------------------------- lifecycle
    def start(self, wait_for_model: bool = True, warmup: bool = False):
        for state in self.models.values():
            state.maybe_reload()
            if wait_for_model and state.version is None:
                raise FileNotFoundError(
                    f"no servable version under {state.base_path}")
        if warmup:
            log.info("warmup predict at max batch size...")
            for state in self.models.values():
                state.warmup()
        if self.metrics_port and _PROM:
            start_http_server(self.metrics_port)
        if self.grpc_ingress == "cpp":
            self._grpc_server = self._build_cpp_ingress()
            self._grpc_server.start()
            self.grpc_port = self._grpc_server.port
        else:
            self._grpc_server = self._build_grpc()
            self._grpc_server.start()

        if self.rest_port is not None:
            import uvicorn
            app = self._build_rest_app()
            config = uvicorn.Config(app, host="0.0.0.0",
                                    port=self.rest_port,
                                    log_level="warning")
            self._uvicorn_server = uvicorn.Server(config)
            self._rest_thread = threading.Thread(
                target=self._uvicorn_server.run, daemon=True)
            self._rest_thread.start()

        self._poll_thread = threading.Thread(target=self._poll_loop,
                                             daemon=True)
        self._poll_thread.start()
        log.info("ModelServer up: grpc=:%d rest=:%s models=%s",
                 self.grpc_port, self.rest_port,
                 {n: s.version for n, s in self.models.items()})

    def _poll_loop(self):
        while not self._stop.wait(self.poll_interval_s):
            # TF-Serving --model_config_file_poll_wait_seconds: re-read
            # the models.config file itself and apply adds/removes.
            if self.model_config_path and \
                    self.config_poll_interval_s > 0:
                now = time.monotonic()
                if now - self._last_config_poll >= \
                        self.config_poll_interval_s:
                    self._last_config_poll = now
                    try:
                        from .config_file import parse_model_config
                        self.reload_config(
                            parse_model_config(self.model_config_path))
                    except Exception:  # noqa: BLE001
                        log.exception("model_config_file poll failed")
            for state in self.models.values():
                try:
                    state.maybe_reload()
                except Exception:  # noqa: BLE001
                    log.exception("version reload failed (%s)",
                                  state.name)

    def stop(self, grace: float = 1.0):
        self._stop.set()
        if self._grpc_server is not None:
            if self.grpc_ingress == "cpp":
                self._grpc_server.stop(drain_s=grace)
            else:
                self._grpc_server.stop(grace).wait()
        if self._uvicorn_server is not None:
            self._uvicorn_server.should_exit = True
        for state in self.models.values():
            for slot in state.versions.values():
                for b in slot["batchers"]:
                    b.close()

    def wait(self):
        if self.grpc_ingress == "cpp":
            self._stop.wait()
        else:
            self._grpc_server.wait_for_termination()


def main():  # pragma: no cover
    import argparse
    ap = argparse.ArgumentParser(description="maskrcnn_amd model server")
    ap.add_argument("--model_name", default="mask_rcnn")
    ap.add_argument("--model_base_path", default=None)
    ap.add_argument("--model_config_file", default=None,
                    help="TF-Serving models.config text proto "
                         "(multi-model mode)")
    ap.add_argument("--port", type=int, default=8500)
    ap.add_argument("--rest_api_port", type=int, default=8501)
    ap.add_argument("--metrics_port", type=int, default=None)
    ap.add_argument("--enable_batching", action="store_true")
    ap.add_argument("--batching_parameters_file", default=None,
                    help="TF-Serving batching_parameters.txt text proto")
    ap.add_argument("--num_replicas", type=int, default=1,
                    help="one engine per GPU: cuda:0..cuda:N-1")
    ap.add_argument("--max_batch_size", type=int, default=32)
    ap.add_argument("--batch_timeout_micros", type=int, default=2000)
    ap.add_argument("--grpc_ingress", choices=("python", "cpp"),
                    default="python",
                    help="cpp = native HTTP/2 front-end "
                         "(csrc/ingress.cpp, no GIL on the wire path)")
    ap.add_argument("--file_system_poll_wait_seconds", type=float,
                    default=5.0,
                    help="version-directory poll interval "
                         "(TF-Serving flag; 0 disables polling)")
    ap.add_argument("--model_config_file_poll_wait_seconds", type=float,
                    default=0.0,
                    help="re-read --model_config_file at this interval "
                         "and apply adds/removes (0 = RPC reload only)")
    ap.add_argument("--enable_model_warmup",
                    action=argparse.BooleanOptionalAction, default=True,
                    help="replay assets.extra/tf_serving_warmup_requests"
                         " at version load (TF-Serving default on)")
    args = ap.parse_args()
    logging.basicConfig(level=logging.INFO)
    devices = None
    if args.num_replicas > 1:
        devices = [f"cuda:{i}" for i in range(args.num_replicas)]
    max_batch = args.max_batch_size if args.enable_batching else 1
    timeout_us = args.batch_timeout_micros
    if args.batching_parameters_file:
        from .config_file import parse_batching_parameters
        bp = parse_batching_parameters(args.batching_parameters_file)
        max_batch = bp.get("max_batch_size", max_batch)
        timeout_us = bp.get("batch_timeout_micros", timeout_us)
    model_configs = None
    if args.model_config_file:
        from .config_file import parse_model_config
        model_configs = parse_model_config(args.model_config_file)
    elif args.model_base_path is None:
        ap.error("--model_base_path or --model_config_file required")
    server = ModelServer(
        args.model_name, args.model_base_path, grpc_port=args.port,
        rest_port=args.rest_api_port or None,  # 0 = disable REST
        metrics_port=args.metrics_port,
        devices=devices, model_configs=model_configs,
        max_batch_size=max_batch, batch_timeout_micros=timeout_us,
        grpc_ingress=args.grpc_ingress,
        poll_interval_s=args.file_system_poll_wait_seconds or 3600.0,
        enable_model_warmup=args.enable_model_warmup)
    if args.model_config_file:
        server.model_config_path = args.model_config_file
        server.config_poll_interval_s = \
            args.model_config_file_poll_wait_seconds
    server.start()
    server.wait()


if __name__ == "__main__":  # pragma: no cover
    main()
