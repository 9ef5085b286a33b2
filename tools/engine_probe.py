#!/usr/bin/env python3
"""Time InferenceEngine.predict at serving batch sizes (GPU probe)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from maskrcnn_amd._rocm_env import ensure_env_or_reexec
ensure_env_or_reexec()
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
H = W = 1024
for B in (1, 8, 32):
    images = np.random.rand(B, H, W, 3).astype(np.float32) * 50
    metas = np.zeros((B, cfg.IMAGE_META_SIZE), np.float32)
    metas[:, 4:7] = (H, W, 3); metas[:, 7:11] = (0, 0, H, W)
    t0 = time.perf_counter()
    out = eng.predict(images, metas)
    t1 = time.perf_counter()
    out = eng.predict(images, metas)
    t2 = time.perf_counter()
    print(f"B={B}: first={t1-t0:.2f}s warm={t2-t1:.2f}s -> {1000*(t2-t1)/B:.1f} ms/img", flush=True)
