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
import sys

_staged = False
_bench_set = False

_ENV = {"MIOPEN_FIND_MODE": "5"}


def stage_miopen_env(local_rank: int = 0) -> None:
    """Export MIOpen env knobs. Torch-free: call before `import torch`."""
    global _staged
    if _staged:
        return
    _staged = True
    for k, v in _ENV.items():
        os.environ.setdefault(k, v)


def ensure_env_or_reexec() -> None:
    """Guarantee the MIOpen env is visible to libMIOpen by re-exec'ing the
    interpreter once if needed.

    Empirically (A/B on MI355X), MIOPEN_FIND_MODE set via os.environ before
    `import torch` is NOT honored (exhaustive Find still runs, +100 s
    warmup), while the same variable exported by the parent shell is. So
    entry-point scripts (bench.py, __graft_entry__) call this first: if the
    variable is missing from the inherited environment, set it and execv
    the same command line. Library imports never re-exec.
    """
    missing = [k for k in _ENV if k not in os.environ]
    if not missing:  # respect any user-set values
        return
    if os.environ.get("MASKRCNN_AMD_NO_REEXEC"):
        return
    for k in missing:
        os.environ[k] = _ENV[k]
    os.environ["MASKRCNN_AMD_NO_REEXEC"] = "1"  # loop guard
    try:
        os.execv(sys.executable, [sys.executable] + sys.argv)
    except OSError:
        pass


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
