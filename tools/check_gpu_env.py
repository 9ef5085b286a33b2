#!/usr/bin/env python3
"""One-shot GPU box sanity check (run this FIRST on a fresh gpurun box).

Prints: device + arch, extension load status (fail-loudly check), a
micro numerics check of each HIP kernel family, and the MIOpen mode the
process actually sees. Exit code 0 = everything usable.
"""
import os
import subprocess
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from maskrcnn_amd._rocm_env import ensure_env_or_reexec  # noqa: E402

if __name__ == "__main__":
    ensure_env_or_reexec()


def main():
    import torch
    ok = True
    print(f"torch {torch.__version__}; cuda available: "
          f"{torch.cuda.is_available()}")
    if not torch.cuda.is_available():
        print("NO GPU — aborting")
        return 1
    print("device:", torch.cuda.get_device_name(0),
          torch.cuda.get_device_properties(0).gcnArchName)
    print("MIOPEN_FIND_MODE =", os.environ.get("MIOPEN_FIND_MODE"))
    try:
        out = subprocess.run(["rocm-smi", "--showmeminfo", "vram"],
                             capture_output=True, text=True, timeout=30)
        print(out.stdout.strip().splitlines()[-1])
    except Exception as e:  # noqa: BLE001
        print("rocm-smi unavailable:", e)

    from maskrcnn_amd.ops._ext import require_ext
    ext = require_ext()
    print("extension:", ext.__file__)

    dev = torch.device("cuda:0")
    # nms
    boxes = torch.tensor([[0, 0, 1, 1], [0, 0, 1, 1], [2, 2, 3, 3]],
                         dtype=torch.float32, device=dev)
    keep = ext.nms(boxes, torch.tensor([0.9, 0.8, 0.7], device=dev), 0.5)
    assert keep.tolist() == [0, 2], keep
    print("nms ok")
    # roi_align
    feats = torch.arange(16., device=dev).reshape(1, 1, 4, 4)
    out = ext.roi_align_forward(
        feats, torch.tensor([[0, 0, 1, 1.]], device=dev),
        torch.zeros(1, dtype=torch.int32, device=dev), 2, 2, 2, False)
    assert out.shape == (1, 1, 2, 2)
    print("roi_align ok")
    # conv3x3 v1 vs torch
    import torch.nn.functional as F
    x = torch.randn(1, 32, 8, 8, device=dev).bfloat16() \
        .to(memory_format=torch.channels_last)
    w = torch.randn(32, 32, 3, 3, device=dev) * 0.1
    from maskrcnn_amd.ops.conv3x3 import conv3x3_bias_relu
    y = conv3x3_bias_relu(x, w, torch.zeros(32, device=dev), relu=False)
    ref = F.conv2d(x.float(), w, padding=1)
    err = (y.float() - ref).abs().max().item()
    assert err < 0.5, err
    print(f"conv3x3 ok (max err {err:.3f})")
    print("ALL OK")
    return 0


if __name__ == "__main__":
    sys.exit(main())
