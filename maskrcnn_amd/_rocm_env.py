"""ROCm performance setup shared by bench / trainer / serving.

MIOpen's immediate mode falls back to NAIVE conv kernels for gfx950
bf16-NHWC shapes absent from the system find-db — measured at 93% of the
training step (profiles/r01_naive_conv.md). Two-part fix:
  1. Ship the tuned user find-db in-repo (miopen_db/) and stage it to a
     writable dir, exporting MIOPEN_USER_DB_PATH BEFORE libMIOpen loads
     (i.e. before `import torch`) — stage_miopen_db() below is import-safe
     (no torch) and is invoked from maskrcnn_amd/__init__ and bench.py.
  2. torch.backends.cudnn.benchmark = True → per-shape Find for any shape
     still missing from the shipped db (warmup-time cost only).
"""

from __future__ import annotations

import os
import shutil
import tempfile

_staged = False
_bench_set = False


def stage_miopen_db(local_rank: int = 0) -> None:
    """Export MIOPEN_USER_DB_PATH to a writable copy of miopen_db/.

    Torch-free: safe to call before `import torch` (which loads libMIOpen).
    """
    global _staged
    if _staged or "MIOPEN_USER_DB_PATH" in os.environ:
        _staged = True
        return
    _staged = True
    repo_root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    db = os.path.join(repo_root, "miopen_db")
    if not os.path.isdir(db):
        return
    stage = os.path.join(tempfile.gettempdir(), f"miopen_db_r{local_rank}")
    try:
        if not os.path.isdir(stage):
            shutil.copytree(db, stage)
        os.environ["MIOPEN_USER_DB_PATH"] = stage
        # FAST find: consult the shipped find-db and skip the exhaustive
        # per-shape search torch's benchmark mode forces (measured: same
        # 47 ms/step steady state, warmup 139 s -> 41 s). A shape missing
        # from the db falls back to immediate mode — regenerate the db
        # (tools/tune_miopen.py) when adding new model configs.
        os.environ.setdefault("MIOPEN_FIND_MODE", "5")
    except OSError:
        pass


def setup_rocm_perf(local_rank: int = 0) -> None:
    """Full setup: db staging + cudnn.benchmark (call after torch import)."""
    global _bench_set
    stage_miopen_db(local_rank)
    if _bench_set:
        return
    _bench_set = True
    import torch
    if torch.cuda.is_available():
        torch.backends.cudnn.benchmark = True
