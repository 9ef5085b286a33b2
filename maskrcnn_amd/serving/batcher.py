"""Dynamic request batcher.

Replicates TF-Serving's SharedBatchScheduler/BatchingSession behavior
(SURVEY.md N11, S2 batching_parameters): concurrent Predict calls are
merged along the batch axis up to max_batch_size, or flushed after
batch_timeout_micros. Requests are bucketed by molded-image shape (with
IMAGE_RESIZE_MODE='square' every image is e.g. 1024x1024, so one bucket;
mixed-size configs get per-shape buckets, SURVEY.md §7 hard-part 5).
"""

from __future__ import annotations

import logging
import queue
import threading
import time
from concurrent.futures import Future
from typing import Optional

import numpy as np

log = logging.getLogger(__name__)

try:
    from prometheus_client import Histogram
    QUEUE_WAIT = Histogram(
        "batch_queue_wait_seconds",
        "request time spent queued before batch assembly",
        buckets=(.0005, .001, .0025, .005, .01, .025, .05, .1, .5, 2))
    INFER_TIME = Histogram(
        "batch_inference_seconds", "engine.predict time per batch",
        buckets=(.005, .01, .025, .05, .1, .25, .5, 1, 2.5, 10))
    _PROM = True
except ImportError:  # pragma: no cover
    _PROM = False


class BatcherQueueFull(RuntimeError):
    """max_enqueued_batches exceeded — TF-Serving sheds load here
    (RESOURCE_EXHAUSTED) rather than buffering unboundedly."""


class DynamicBatcher:
    def __init__(self, engine, max_batch_size: int = 32,
                 batch_timeout_micros: int = 2000,
                 max_enqueued_batches: int = 128,
                 num_batch_threads: int = 1,
                 pad_to_max: bool = True):
        self.engine = engine
        self.max_batch_size = max_batch_size
        # TF-Serving's BatchingSession pads every batch to max_batch_size
        # (SURVEY.md N11). Besides matching that behavior, padding keeps
        # the conv batch shape FIXED, which matters on MIOpen: every new
        # batch size triggers a per-shape Find (tens of seconds at
        # 1024²x32). One shape = one warmup.
        self.pad_to_max = pad_to_max
        self.batch_timeout_s = batch_timeout_micros / 1e6
        self.queue: "queue.Queue" = queue.Queue(
            maxsize=max_enqueued_batches * max(1, max_batch_size))
        self._stop = False
        self.threads = [
            threading.Thread(target=self._worker, daemon=True,
                             name=f"batcher-{i}")
            for i in range(num_batch_threads)]
        for t in self.threads:
            t.start()
        self.stats = {"batches": 0, "requests": 0, "batch_size_sum": 0}

    def submit(self, input_image: np.ndarray, input_image_meta: np.ndarray,
               input_anchors: Optional[np.ndarray] = None,
               outputs: Optional[list] = None) -> Future:
        """Enqueue one request (batch dim must be 1..max_batch_size).
        `outputs` restricts which signature outputs are computed back to
        the host (None = all 7). Returns a Future resolving to the
        outputs dict (numpy). Raises BatcherQueueFull when
        max_enqueued_batches worth of requests are already waiting
        (TF-Serving load-shedding semantics — the RPC maps it to
        RESOURCE_EXHAUSTED instead of buffering unboundedly)."""
        if self._stop:
            raise RuntimeError("batcher closed")
        fut: Future = Future()
        fut._enqueued_at = time.monotonic()
        try:
            self.queue.put_nowait((input_image, input_image_meta,
                                   input_anchors, outputs, fut))
        except queue.Full:
            raise BatcherQueueFull(
                f"batching queue full ({self.queue.maxsize} requests "
                f"waiting); retry later")
        return fut

    def close(self):
        """Graceful: stop accepting, drain what's queued (workers keep
        processing until empty — TF-Serving unloads wait for in-flight
        work), then fail any straggler enqueued during the race."""
        self._stop = True
        for t in self.threads:
            t.join(timeout=60.0)
        while True:
            try:
                item = self.queue.get_nowait()
            except queue.Empty:
                break
            if not item[-1].done():
                item[-1].set_exception(RuntimeError("batcher closed"))

    # ------------------------------------------------------------ worker
    def _worker(self):
        while True:
            try:
                first = self.queue.get(timeout=0.1)
            except queue.Empty:
                if self._stop:
                    break
                continue
            batch = [first]
            n = first[0].shape[0]
            deadline = self.batch_timeout_s
            import time
            t0 = time.monotonic()
            while n < self.max_batch_size:
                remaining = deadline - (time.monotonic() - t0)
                if remaining <= 0:
                    break
                try:
                    item = self.queue.get(timeout=remaining)
                except queue.Empty:
                    break
                if (item[0].shape[1:] != first[0].shape[1:] or
                        item[0].dtype != first[0].dtype):
                    # Different molded shape (or fp32-vs-uint8 wire
                    # format): flush current, start new — a mixed
                    # concat would silently upcast unmolded uint8.
                    # `first` MUST follow the new batch: comparing later
                    # items against the stale head let a uint8 request
                    # join an fp32 batch (upcast to raw 0..255 floats —
                    # garbage detections; caught by
                    # test_grpc_mixed_uint8_fp32_clients).
                    self._run(batch)
                    batch = [item]
                    first = item
                    n = item[0].shape[0]
                    t0 = time.monotonic()
                    continue
                batch.append(item)
                n += item[0].shape[0]
            self._run(batch)

    def _run(self, batch):
        if not batch:
            return
        try:
            # Images are passed as the REQUEST LIST: the engine
            # collates them straight into its pinned staging buffer
            # (one memcpy pass; a host-side np.concatenate of 32 x
            # 1024² requests was an extra ~100 MB / ~20 ms per batch
            # under the GIL).
            image_parts = [b[0] for b in batch]
            metas = np.concatenate([b[1] for b in batch], axis=0)
            anchors = batch[0][2]
            # Union of the batch's output filters (None = everything).
            outputs = None
            if all(b[3] is not None for b in batch):
                outputs = sorted({n for b in batch for n in b[3]})
            real_n = sum(p.shape[0] for p in image_parts)
            padded_n = real_n
            if self.pad_to_max and real_n < self.max_batch_size:
                pad = self.max_batch_size - real_n
                padded_n = self.max_batch_size
                # Padded metas replicate row 0 (real) so shape fields
                # parsed from the meta stay valid. Image pad rows are
                # zero-filled by the engine's collate.
                metas = np.concatenate(
                    [metas, np.repeat(metas[:1], pad, axis=0)], axis=0)
            t0 = time.monotonic()
            if _PROM:
                for *_r, fut in batch:
                    QUEUE_WAIT.observe(
                        t0 - getattr(fut, "_enqueued_at", t0))
            out = self.engine.predict(
                image_parts, metas, anchors, trim_to=real_n,
                outputs=outputs, pad_to=padded_n)
            dt = time.monotonic() - t0
            if _PROM:
                INFER_TIME.observe(dt)
            log.info("batch of %d (padded %d) in %.3fs", real_n,
                     padded_n, dt)
            self.stats["batches"] += 1
            self.stats["requests"] += len(batch)
            self.stats["batch_size_sum"] += padded_n
            off = 0
            for img, _, _, _, fut in batch:
                k = img.shape[0]
                sliced = {name: arr[off:off + k]
                          for name, arr in out.items()}
                fut.set_result(sliced)
                off += k
        except Exception as e:  # propagate to every waiter
            for *_rest, fut in batch:
                if not fut.done():
                    fut.set_exception(e)
