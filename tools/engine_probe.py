#!/usr/bin/env python3
"""Reproduce the second-find: zeros warmup then random data, phase-logged."""
import os, sys, time, logging
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from maskrcnn_amd._rocm_env import ensure_env_or_reexec
if __name__ == "__main__":
    # re-exec only when run as a script (library imports must never
    # replace the host process; see maskrcnn_amd/_rocm_env.py)
    ensure_env_or_reexec()
def main():
    logging.basicConfig(level=logging.INFO, format="%(asctime)s %(message)s")
    import numpy as np
    import torch
    from maskrcnn_amd.config import Config

    class Cfg(Config):
        NAME = "probe"
        BACKBONE = "resnet101"
        NUM_CLASSES = 81
        IMAGES_PER_GPU = 1

    cfg = Cfg()
    from maskrcnn_amd.serving.engine import InferenceEngine
    torch.manual_seed(0)
    eng = InferenceEngine(cfg)
    eng.model.phase_log = True
    H = W = 1024
    B = 8
    metas = np.zeros((B, cfg.IMAGE_META_SIZE), np.float32)
    metas[:, 4:7] = (H, W, 3); metas[:, 7:11] = (0, 0, H, W)
    for tag, images in (("zeros", np.zeros((B, H, W, 3), np.float32)),
                        ("random", np.random.rand(B, H, W, 3).astype(np.float32) * 50),
                        ("random2", np.random.rand(B, H, W, 3).astype(np.float32) * 50)):
        t0 = time.perf_counter()
        eng.predict(images, metas)
        print(f"== {tag}: {time.perf_counter()-t0:.2f}s", flush=True)


if __name__ == "__main__":
    main()
