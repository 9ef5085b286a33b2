"""ROCm performance setup shared by bench / trainer / serving.

MIOpen's immediate mode falls back to NAIVE conv kernels for gfx950
bf16-NHWC shapes absent from the system find-db — measured at 93% of the
training step (profiles/r01_naive_conv.md). Two-part fix:
  1. torch.backends.cudnn.benchmark = True → per-shape Find at first use.
  2. Ship the tuned user find-db in-repo (miopen_db/) and stage it to a
     writable dir so a fresh box skips the Find cost.
"""

from __future__ import annotations

import os
import shutil
import tempfile

_done = False


def setup_rocm_perf(local_rank: int = 0) -> None:
    global _done
    if _done:
        return
    _done = True
    import torch
    if not torch.cuda.is_available():
        return
    torch.backends.cudnn.benchmark = True
    repo_root = os.path.dirname(os.path.dirname(
        os.path.dirname(os.path.abspath(__file__))))
    db = os.path.join(repo_root, "miopen_db")
    if os.path.isdir(db) and "MIOPEN_USER_DB_PATH" not in os.environ:
        stage = os.path.join(tempfile.gettempdir(),
                             f"miopen_db_r{local_rank}")
        if not os.path.isdir(stage):
            try:
                shutil.copytree(db, stage)
            except OSError:
                return
        os.environ["MIOPEN_USER_DB_PATH"] = stage
