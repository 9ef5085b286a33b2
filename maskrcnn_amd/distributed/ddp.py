"""Data-parallel gradient synchronization — RCCL over xGMI.

Replaces the reference's ParallelModel (mrcnn/parallel_model.py:1-175,
in-graph tower replication with a /cpu:0 merge and no collectives) with the
MI355X-native design of SURVEY.md §2.4: one process per GPU, hand-bucketed
all-reduce overlapped with backward.

Design:
- Parameters are grouped into flat buckets in REVERSE parameter order
  (approximate backward completion order), default 25 MB per bucket —
  sized for 7 p2p xGMI links (≈153 GB/s each): big enough to amortize
  RCCL launch, small enough that the first all-reduce starts while most of
  backward is still running.
- Each parameter's post-accumulate-grad hook copies its grad into the
  bucket's flat buffer (optionally casting to bf16 for 2x fewer xGMI
  bytes); the last arrival launches an async all_reduce.
- finalize_backward() waits on all reduce works and scatters averaged
  grads back into param.grad.

Works with backend "nccl" (RCCL on ROCm) on GPU and "gloo" on CPU
(multi-process CPU tests).
"""

from __future__ import annotations

import torch
import torch.distributed as dist


class GradBucket:
    def __init__(self, params, comm_dtype, device):
        self.params = params
        self.numels = [p.numel() for p in params]
        self.offsets = []
        off = 0
        for n in self.numels:
            self.offsets.append(off)
            off += n
        self.total = off
        self.buffer = torch.zeros(self.total, dtype=comm_dtype,
                                  device=device)
        self.pending = 0
        self.work = None

    def reset(self):
        self.pending = len(self.params)
        self.work = None


class DistributedTrainer:
    """Bucketed gradient all-reduce. Use:

        ddp = DistributedTrainer(model, config)
        loss.backward(); ddp.finalize_backward(); optimizer.step()
    """

    def __init__(self, model, config=None, bucket_mb: float = 25.0,
                 comm_dtype=None, process_group=None):
        assert dist.is_initialized(), \
            "torch.distributed must be initialized first"
        self.model = model
        self.group = process_group
        self.world = dist.get_world_size(process_group)
        backend = dist.get_backend(process_group)
        if comm_dtype is None:
            # bf16 wire format on RCCL (halves xGMI bytes); fp32 on gloo
            # (gloo lacks bf16 kernels on some builds — keep CPU tests safe).
            want_bf16 = (config is None or
                         getattr(config, "COMPUTE_DTYPE", "bf16") == "bf16")
            comm_dtype = torch.bfloat16 \
                if (backend == "nccl" and want_bf16) else torch.float32
        self.comm_dtype = comm_dtype

        params = [p for _, p in model.named_parameters() if p.requires_grad]
        device = params[0].device
        # Reverse order ~= backward completion order.
        params = list(reversed(params))

        bucket_bytes = int(bucket_mb * 1024 * 1024)
        self.buckets: list[GradBucket] = []
        cur, cur_bytes = [], 0
        esize = torch.tensor([], dtype=comm_dtype).element_size()
        for p in params:
            cur.append(p)
            cur_bytes += p.numel() * esize
            if cur_bytes >= bucket_bytes:
                self.buckets.append(GradBucket(cur, comm_dtype, device))
                cur, cur_bytes = [], 0
        if cur:
            self.buckets.append(GradBucket(cur, comm_dtype, device))

        self._param_slot = {}
        for b in self.buckets:
            for i, p in enumerate(b.params):
                self._param_slot[p] = (b, i)

        self._hooks = []
        for p in params:
            self._hooks.append(
                p.register_post_accumulate_grad_hook(self._on_grad))
        self._reset_pending()

        self._sync = True  # no_sync() flips this during grad accumulation

        # Broadcast initial weights + buffers from rank 0 (reference
        # ParallelModel shares one weight copy; we replicate explicitly).
        # COALESCED: R101-FPN has ~450 state tensors; one broadcast per
        # tensor is ~450 collective launches at init. Pack same-dtype
        # tensors into <=64 MB flat chunks -> ~10 launches.
        with torch.no_grad():
            chunk, chunk_bytes, dtype = [], 0, None
            LIMIT = 64 * 1024 * 1024

            def flush():
                if not chunk:
                    return
                flat = torch.cat([t.reshape(-1) for t in chunk])
                dist.broadcast(flat, src=0, group=self.group)
                off = 0
                for t in chunk:
                    n = t.numel()
                    t.copy_(flat[off:off + n].reshape(t.shape))
                    off += n
                chunk.clear()

            for t in model.state_dict().values():
                if not (t.is_floating_point() or t.dtype in (
                        torch.int32, torch.int64)):
                    continue
                if t.dtype != dtype or chunk_bytes >= LIMIT:
                    flush()
                    dtype, chunk_bytes = t.dtype, 0
                chunk.append(t)
                chunk_bytes += t.numel() * t.element_size()
            flush()

    def _reset_pending(self):
        for b in self.buckets:
            b.reset()

    @torch.no_grad()
    def _on_grad(self, p):
        if not self._sync:
            return  # no_sync(): grads accumulate locally in p.grad
        b, i = self._param_slot[p]
        off, n = b.offsets[i], b.numels[i]
        b.buffer[off:off + n].copy_(p.grad.reshape(-1).to(self.comm_dtype))
        b.pending -= 1
        if b.pending == 0:
            b.work = dist.all_reduce(b.buffer, op=dist.ReduceOp.SUM,
                                     group=self.group, async_op=True)

    def no_sync(self):
        """Context manager: skip gradient synchronization (gradient
        accumulation). Grads accumulate in param.grad across backwards;
        the LAST backward runs outside no_sync() and its hooks copy the
        fully-accumulated grads into the buckets, so exactly one
        all-reduce happens per optimizer step (torch DDP no_sync
        semantics; the reference's ParallelModel had no equivalent)."""
        import contextlib

        @contextlib.contextmanager
        def ctx():
            prev = self._sync
            self._sync = False
            try:
                yield
            finally:
                self._sync = prev
        return ctx()

    @torch.no_grad()
    def finalize_backward(self):
        """Wait for all bucket reductions and write averaged grads back."""
        incomplete = [b for b in self.buckets if b.pending > 0]
        if incomplete:
            missing = [id(p) for b in incomplete for p in b.params]
            raise RuntimeError(
                f"{len(incomplete)} gradient buckets incomplete after "
                f"backward ({len(missing)} params without grads). All "
                "trainable parameters must receive gradients.")
        inv_world = 1.0 / self.world
        for b in self.buckets:
            b.work.wait()
            for p, off, n in zip(b.params, b.offsets, b.numels):
                p.grad.copy_(
                    b.buffer[off:off + n].view_as(p.grad).to(p.grad.dtype))
                p.grad.mul_(inv_world)
        self._reset_pending()

    def remove_hooks(self):
        for h in self._hooks:
            h.remove()
        self._hooks = []
