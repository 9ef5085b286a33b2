#!/usr/bin/env python3
"""A/B micro-benchmark: hand-written conv3x3/fpn_fuse MFMA kernels vs
MIOpen (torch F.conv2d with benchmark Find) on R101-FPN shapes."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from maskrcnn_amd._rocm_env import ensure_env_or_reexec  # noqa: E402

if __name__ == "__main__":
    # re-exec only when run as a script (library imports must never
    # replace the host process; see maskrcnn_amd/_rocm_env.py)
    ensure_env_or_reexec()

import torch  # noqa: E402
import torch.nn.functional as F  # noqa: E402


def timeit(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3  # ms


def main():
    from maskrcnn_amd._rocm_env import setup_rocm_perf
    from maskrcnn_amd.ops._ext import require_ext
    from maskrcnn_amd.ops.conv3x3 import pack_weight

    setup_rocm_perf()
    ext = require_ext()
    dev = torch.device("cuda:0")
    # (B, Cin, H, W, Cout) — the R101-FPN 3x3 shapes at batch 2 / 1024²
    # plus the head shape at 400 rois.
    shapes = [
        (2, 64, 256, 256, 64),     # res2 3x3
        (2, 128, 128, 128, 128),   # res3 3x3
        (2, 256, 64, 64, 256),     # res4 3x3
        (2, 512, 32, 32, 512),     # res5 3x3
        (2, 256, 256, 256, 256),   # fpn_p2 3x3
        (2, 256, 64, 64, 256),     # fpn_p4 3x3
        (400, 256, 14, 14, 256),   # mask head conv
        (2, 256, 256, 256, 512),   # rpn shared (P2)
    ]
    print(f"{'shape':>28}  v1 / v2 / miopen")
    for (B, Cin, H, W, Cout) in shapes:
        x = torch.randn(B, Cin, H, W, device=dev).bfloat16() \
            .to(memory_format=torch.channels_last)
        w = (torch.randn(Cout, Cin, 3, 3, device=dev) * 0.05).bfloat16() \
            .to(memory_format=torch.channels_last)
        b = torch.randn(Cout, device=dev)
        wp = pack_weight(w.float())
        bf = b.float()

        v1 = timeit(lambda: ext.conv3x3_fwd(x, wp, bf, True, 1))
        v2 = timeit(lambda: ext.conv3x3_fwd(x, wp, bf, True, 2))
        v3 = timeit(lambda: ext.conv3x3_fwd(x, wp, bf, True, 3))
        v4 = timeit(lambda: ext.conv3x3_fwd(x, wp, bf, True, 4))
        v5 = timeit(lambda: ext.conv3x3_fwd(x, wp, bf, True, 5))
        # v6 split-K targets fill-bound small-M grids only.
        v6 = timeit(lambda: ext.conv3x3_fwd(x, wp, bf, True, 6)) \
            if B * H * W <= 8192 else float("inf")
        ref = timeit(lambda: F.relu(F.conv2d(x, w, b.bfloat16(),
                                             padding=1)))
        flops = 2.0 * B * H * W * Cout * Cin * 9
        tfs = [flops / (t * 1e-3) / 1e12
               for t in (v1, v2, v3, v4, v5, v6)]
        v6s = (f"v6={v6:7.3f}ms({tfs[5]:5.0f}TF) "
               if v6 != float("inf") else "")
        print(f"{str((B, Cin, H, W, Cout)):>28} "
              f"v1={v1:7.3f}ms({tfs[0]:5.0f}TF) "
              f"v2={v2:7.3f}ms({tfs[1]:5.0f}TF) "
              f"v3={v3:7.3f}ms({tfs[2]:5.0f}TF) "
              f"v4={v4:7.3f}ms({tfs[3]:5.0f}TF) "
              f"v5={v5:7.3f}ms({tfs[4]:5.0f}TF) {v6s}"
              f"miopen={ref:7.3f}ms "
              f"best_ratio={ref / min(v1, v2, v3, v4, v5, v6):5.2f}x")

    # fpn_fuse vs unfused (conv1x1 + interpolate + add).
    print("\nfpn_fuse (1x1 + up2 + add):")
    for (B, Cin, H, W, Cout) in [(2, 2048, 32, 32, 256),
                                 (2, 1024, 64, 64, 256),
                                 (2, 512, 128, 128, 256),
                                 (2, 256, 256, 256, 256)]:
        x = torch.randn(B, Cin, H, W, device=dev).bfloat16() \
            .to(memory_format=torch.channels_last)
        w = (torch.randn(Cout, Cin, 1, 1, device=dev) * 0.05)
        wb = w.bfloat16().to(memory_format=torch.channels_last)
        b = torch.randn(Cout, device=dev)
        up = torch.randn(B, Cout, H // 2, W // 2, device=dev).bfloat16() \
            .to(memory_format=torch.channels_last)
        w2 = w.reshape(Cout, Cin).contiguous().bfloat16()

        ours = timeit(lambda: ext.fpn_fuse(x, w2, b, up))
        ref = timeit(lambda: F.conv2d(x, wb, b.bfloat16()) +
                     F.interpolate(up, scale_factor=2, mode="nearest"))
        print(f"{str((B, Cin, H, W, Cout)):>28} {ours:9.3f} {ref:10.3f} "
              f"{'':8} {ref / ours:6.2f}x")


def bench_wrw():
    """wrw kernel vs aten.convolution_backward weight grad."""
    from maskrcnn_amd.ops._ext import require_ext
    ext = require_ext()
    dev = torch.device("cuda:0")
    print("\nconv3x3 wrw (weight grad):")
    for (B, Cin, H, W, Cout) in [(2, 64, 256, 256, 64),
                                 (2, 128, 128, 128, 128),
                                 (2, 256, 64, 64, 256),
                                 (2, 512, 32, 32, 512),
                                 (2, 256, 256, 256, 256),
                                 (400, 256, 14, 14, 256)]:
        x = torch.randn(B, Cin, H, W, device=dev).bfloat16() \
            .to(memory_format=torch.channels_last)
        dy = torch.randn(B, Cout, H, W, device=dev).bfloat16() \
            .to(memory_format=torch.channels_last)
        w = torch.randn(Cout, Cin, 3, 3, device=dev).bfloat16()
        ours = timeit(lambda: ext.conv3x3_wrw(x, dy))
        ref = timeit(lambda: torch.ops.aten.convolution_backward(
            dy, x, w, None, [1, 1], [1, 1], [1, 1], False, [0, 0], 1,
            [False, True, False])[1])
        flops = 2.0 * B * H * W * Cout * Cin * 9
        print(f"{str((B, Cin, H, W, Cout)):>28} "
              f"ours={ours:7.3f}ms({flops / (ours * 1e-3) / 1e12:5.0f}TF)"
              f" aten={ref:7.3f}ms {ref / ours:5.2f}x")


def bench_fp8():
    """fp8-compute conv (incl. on-the-fly act quant) vs bf16 v2."""
    from maskrcnn_amd.ops._ext import require_ext
    from maskrcnn_amd.ops.conv3x3 import (conv3x3_fp8, pack_weight,
                                          pack_weight_fp8)
    ext = require_ext()
    dev = torch.device("cuda:0")
    print("\nconv3x3 fp8 e4m3 (incl. activation quant) vs bf16 v2:")
    for (B, Cin, H, W, Cout) in [(2, 256, 64, 64, 256),
                                 (2, 256, 256, 256, 256),
                                 (400, 256, 14, 14, 256),
                                 (16, 256, 84, 84, 256)]:  # 1344/16 maps
        x = torch.randn(B, Cin, H, W, device=dev).bfloat16() \
            .to(memory_format=torch.channels_last)
        w = (torch.randn(Cout, Cin, 3, 3, device=dev) * 0.05)
        b = torch.randn(Cout, device=dev)
        qp, scale = pack_weight_fp8(w)
        wp = pack_weight(w)
        f8 = timeit(lambda: conv3x3_fp8(x, qp, scale, b, relu=True))
        bf = timeit(lambda: ext.conv3x3_fwd(x, wp, b.float(), True, 2))
        flops = 2.0 * B * H * W * Cout * Cin * 9
        print(f"{str((B, Cin, H, W, Cout)):>28} "
              f"fp8={f8:7.3f}ms({flops / (f8 * 1e-3) / 1e12:5.0f}TF) "
              f"bf16v2={bf:7.3f}ms {bf / f8:5.2f}x")


def bench_mx():
    """MX block-scaled e4m3 conv (K=128 mfma_scale) vs the best bf16
    hand kernel for the shape. Reports both end-to-end (incl. the
    on-the-fly activation quant pass) and kernel-only time."""
    from maskrcnn_amd.ops._ext import require_ext
    from maskrcnn_amd.ops.conv3x3 import (conv3x3_mx, pack_weight,
                                          pack_weight_mx,
                                          quantize_activations_mx)
    ext = require_ext()
    dev = torch.device("cuda:0")
    print("\nconv3x3 MX e4m3 (K=128 block-scaled) vs bf16 best:")
    for (B, Cin, H, W, Cout) in [(2, 256, 64, 64, 256),
                                 (2, 256, 256, 256, 256),
                                 (400, 256, 14, 14, 256),
                                 (16, 256, 84, 84, 256)]:
        x = torch.randn(B, Cin, H, W, device=dev).bfloat16() \
            .to(memory_format=torch.channels_last)
        w = (torch.randn(Cout, Cin, 3, 3, device=dev) * 0.05)
        b = torch.randn(Cout, device=dev)
        wq, ws = pack_weight_mx(w)
        wp = pack_weight(w)
        bf = b.float()
        mx = timeit(lambda: conv3x3_mx(x, wq, ws, b, relu=True))
        xq, xs = quantize_activations_mx(x)
        mxk = timeit(lambda: ext.conv3x3_mx(xq, xs, wq, ws, bf, True))
        M = B * H * W
        ver = 5 if M >= 131072 else (3 if M >= 32768 else 1)
        bf16 = timeit(lambda: ext.conv3x3_fwd(x, wp, bf, True, ver))
        flops = 2.0 * B * H * W * Cout * Cin * 9
        print(f"{str((B, Cin, H, W, Cout)):>28} "
              f"mx+q={mx:7.3f}ms({flops / (mx * 1e-3) / 1e12:5.0f}TF) "
              f"mx-k={mxk:7.3f}ms({flops / (mxk * 1e-3) / 1e12:5.0f}TF) "
              f"bf16v{ver}={bf16:7.3f}ms ratio={bf16 / mx:5.2f}x "
              f"(kernel-only {bf16 / mxk:5.2f}x)")


if __name__ == "__main__":
    main()
    bench_wrw()
    bench_fp8()
    bench_mx()
