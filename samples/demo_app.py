#!/usr/bin/env python3
"""Web demo app: upload an image, get back the rendered
detections (boxes + masks via visualize.display_instances) as a PNG —
the reference stack's Flask demo in front of TF-Serving.

The detector is pluggable: a local MaskRCNN facade (default) or a
remote model server via --grpc/--rest (GrpcServingClient /
RestServingClient speak the TF-Serving wire protocols).

  python samples/demo_app.py --port 8600                 # local model
  python samples/demo_app.py --grpc localhost:8500       # via server
"""

# NOTE: no `from __future__ import annotations` here — stringized
# annotations break FastAPI's resolution of the locally-imported
# Request type (it would 422 every POST).
import argparse
import io
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

# Raw-body upload (no python-multipart in the offline image): the page
# posts the file bytes directly and swaps in the returned PNG.
_FORM = """<!doctype html>
<title>maskrcnn_amd demo</title>
<h1>Mask R-CNN (MI355X) demo</h1>
<input type="file" id="f" accept="image/*">
<button onclick="go()">Detect</button>
<p id="n"></p><img id="out" style="max-width:90%">
<script>
async function go() {
  const f = document.getElementById('f').files[0];
  if (!f) return;
  const r = await fetch('/detect', {method: 'POST', body: f});
  if (!r.ok) { document.getElementById('n').textContent =
               await r.text(); return; }
  document.getElementById('n').textContent =
      r.headers.get('X-Detections') + ' detections';
  document.getElementById('out').src =
      URL.createObjectURL(await r.blob());
}
</script>"""


def create_app(detector, class_names):
    """detector: callable(image [H,W,3] uint8) -> result dict with
    rois/masks/class_ids/scores (the facade's detect()[0] contract)."""
    import threading

    from fastapi import FastAPI, Request
    from fastapi.responses import HTMLResponse, JSONResponse, Response

    app = FastAPI(title="maskrcnn_amd demo")
    # pyplot keeps GLOBAL figure state — serialize render jobs.
    render_lock = threading.Lock()

    @app.get("/", response_class=HTMLResponse)
    def index():
        return _FORM

    @app.post("/detect")
    async def detect(request: Request):
        from PIL import Image

        from maskrcnn_amd import visualize

        try:
            data = await request.body()
            image = np.asarray(Image.open(io.BytesIO(data))
                               .convert("RGB"))
        except Exception as e:  # noqa: BLE001
            return JSONResponse({"error": f"bad image: {e}"},
                                status_code=400)
        # Inference + rendering are blocking: run off the event loop so
        # concurrent uploads and the index page stay responsive.
        import anyio

        def run():
            with render_lock:
                r = detector(image)
                buf = io.BytesIO()
                visualize.display_instances(
                    image, r["rois"], r["masks"], r["class_ids"],
                    class_names, r.get("scores"), save_path=buf)
            return r, buf

        r, buf = await anyio.to_thread.run_sync(run)
        return Response(buf.getvalue(), media_type="image/png",
                        headers={"X-Detections":
                                 str(len(r["class_ids"]))})

    return app


def main():  # pragma: no cover
    from maskrcnn_amd._rocm_env import ensure_env_or_reexec
    ensure_env_or_reexec()
    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int, default=8600)
    ap.add_argument("--weights", default=None)
    ap.add_argument("--grpc", default=None,
                    help="host:port of a model server (gRPC)")
    ap.add_argument("--rest", default=None,
                    help="http://host:port of a model server (REST)")
    ap.add_argument("--num_classes", type=int, default=81)
    args = ap.parse_args()

    from maskrcnn_amd.config import Config

    class DemoConfig(Config):
        NAME = "demo"
        NUM_CLASSES = args.num_classes
        IMAGES_PER_GPU = 1
        DETECTION_MIN_CONFIDENCE = 0.5

    config = DemoConfig()
    if args.grpc:
        from maskrcnn_amd.serving.client import GrpcServingClient
        client = GrpcServingClient(args.grpc, config)
        detector = lambda img: client.detect([img])[0]  # noqa: E731
    elif args.rest:
        from maskrcnn_amd.serving.client import RestServingClient
        client = RestServingClient(args.rest, config)
        detector = lambda img: client.detect([img])[0]  # noqa: E731
    else:
        from maskrcnn_amd.model import MaskRCNN
        model = MaskRCNN("inference", config, model_dir="logs")
        if args.weights:
            model.load_weights(args.weights, by_name=True)
        detector = lambda img: model.detect([img])[0]  # noqa: E731

    class_names = [f"class{i}" for i in range(config.NUM_CLASSES)]
    import uvicorn
    uvicorn.run(create_app(detector, class_names), host="0.0.0.0",
                port=args.port)


if __name__ == "__main__":  # pragma: no cover
    main()
