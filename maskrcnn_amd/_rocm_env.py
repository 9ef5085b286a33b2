"""ROCm performance environment shared by bench / trainer / serving.

MIOpen's immediate mode (torch default) resolves most gfx950 bf16-NHWC
conv shapes to NAIVE fallback kernels — measured at 93% of the training
step (profiles/r01_naive_conv.md). Fixes:

  1. MIOPEN_FIND_MODE=5 (FAST): MIOpen's heuristic solver selection picks
     the tuned CK/iGEMM solvers with no search. Empirically the variable
     is only honored when it is in the environment the PROCESS started
     with (A/B on MI355X: python-set os.environ before `import torch` is
     ignored; shell-exported works). Entry-point scripts therefore call
     ensure_env_or_reexec() which re-execs the interpreter once with the
     variable exported. Library imports never re-exec and never set the
     variable (a python-side setdefault would defeat the re-exec check
     while doing nothing for libMIOpen).
  2. torch.backends.cudnn.benchmark = True (per-shape Find for any shape
     the heuristics cover badly; warmup-time cost only).

Measured (R101-FPN bf16 train, batch 2, 1024², 1 MI355X):
  immediate mode (naive convs):        61-64 ms/step
  FIND_MODE=5 + benchmark:             47.2 ms/step, warmup ~10 s
  exhaustive Find + benchmark:         46.5 ms/step, warmup ~100 s
"""

from __future__ import annotations

import os
import sys

_bench_set = False

_ENV = {"MIOPEN_FIND_MODE": "5"}
_SENTINEL = "MASKRCNN_AMD_NO_REEXEC"


def _env_to_set() -> dict:
    """_ENV plus, when the repo ships a validated MIOpen user find-db
    (miopen_db/ at the repo root), MIOPEN_USER_DB_PATH pointing at it —
    a fresh box then skips the per-shape Find (serving batch-32 1024²
    cold-start measured 3.5 min without it). The directory must be
    writable (MIOpen also records new finds there); gpurun snapshots
    are."""
    env = dict(_ENV)
    db = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "miopen_db")
    if os.path.isdir(db) and os.access(db, os.W_OK):
        env["MIOPEN_USER_DB_PATH"] = db
    return env


def ensure_env_or_reexec() -> None:
    """Re-exec the interpreter once with the MIOpen env exported.

    Call this FIRST in entry-point scripts (before importing torch or any
    module that does). No-ops when the parent shell already exported the
    variables, when re-exec already happened (sentinel), or on failure.
    """
    if os.environ.get(_SENTINEL):
        return
    _env = _env_to_set()
    missing = [k for k in _env if k not in os.environ]
    if not missing:  # parent shell set them — nothing to do
        return
    for k in missing:
        os.environ[k] = _env[k]
    os.environ[_SENTINEL] = "1"  # loop guard
    try:
        os.execv(sys.executable, [sys.executable] + sys.argv)
    except OSError:
        pass


def stage_miopen_env(local_rank: int = 0) -> None:
    """Deprecated no-op kept for import compatibility (see module doc:
    python-side env writes do not reach libMIOpen)."""


# Backwards-compatible alias (older call sites).
stage_miopen_db = stage_miopen_env


def setup_rocm_perf(local_rank: int = 0) -> None:
    """Torch-side knobs (call after torch import)."""
    global _bench_set
    if _bench_set:
        return
    _bench_set = True
    import torch
    if torch.cuda.is_available():
        # benchmark=True: exhaustive MIOpen Find per conv shape at first
        # use (~100 s warmup for the R101 train step, once per process) →
        # 46.5 ms/step steady. benchmark=False (immediate mode) resolves
        # many bf16-NHWC shapes to naive kernels → 60-64 ms/step. The
        # Find results land in MIOpen's user find-db, so later processes
        # on the same machine skip the search entirely.
        # MASKRCNN_NO_CUDNN_BENCHMARK=1 opts out.
        torch.backends.cudnn.benchmark = not bool(
            os.environ.get("MASKRCNN_NO_CUDNN_BENCHMARK"))
