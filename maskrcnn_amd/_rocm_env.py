"""ROCm performance environment shared by bench / trainer / serving.

MIOpen's immediate mode (torch default) resolves most gfx950 bf16-NHWC
conv shapes to NAIVE fallback kernels — measured at 93% of the training
step (profiles/r01_naive_conv.md). Two measured fixes:

  1. MIOPEN_FIND_MODE=5 (FAST): MIOpen's heuristic solver selection picks
     the tuned CK/iGEMM solvers without any search. Must be exported
     BEFORE libMIOpen loads (i.e. before `import torch`) —
     stage_miopen_env() below is torch-free and invoked from
     maskrcnn_amd/__init__ and bench.py.
  2. torch.backends.cudnn.benchmark = True.

Measured (R101-FPN bf16 train, batch 2, 1024², 1 MI355X):
  immediate mode (naive convs):        61-64 ms/step
  FIND_MODE=5 + benchmark:             47.4 ms/step, warmup ~10 s
  exhaustive Find + benchmark:         46.5 ms/step, warmup ~100 s
A shipped user find-db was tried and REJECTED: the db recorded naive
winners for several wrw shapes and made steady state worse than the
heuristic fallback.
"""

from __future__ import annotations

import os

_staged = False
_bench_set = False


def stage_miopen_env(local_rank: int = 0) -> None:
    """Export MIOpen env knobs. Torch-free: call before `import torch`."""
    global _staged
    if _staged:
        return
    _staged = True
    os.environ.setdefault("MIOPEN_FIND_MODE", "5")


# Backwards-compatible alias (older call sites).
stage_miopen_db = stage_miopen_env


def setup_rocm_perf(local_rank: int = 0) -> None:
    """Full setup: env staging + cudnn.benchmark (call after torch import)."""
    global _bench_set
    stage_miopen_env(local_rank)
    if _bench_set:
        return
    _bench_set = True
    import torch
    if torch.cuda.is_available():
        torch.backends.cudnn.benchmark = True
