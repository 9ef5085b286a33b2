#!/usr/bin/env python3
"""Build the in-tree HIP extension maskrcnn_amd/_C.so for gfx950.

Drives hipcc directly (one compile per TU, parallel, then link) so the
.so lands inside the package and ships to the GPU box with the repo
snapshot. Cross-compiles fine on a GPU-less box.
"""

from __future__ import annotations

import os
import subprocess
import sysconfig
from concurrent.futures import ThreadPoolExecutor
from pathlib import Path

ROOT = Path(__file__).resolve().parent
CSRC = ROOT / "csrc"

ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
# MASKRCNN_ASAN=1: host-side AddressSanitizer build (device code is not
# instrumented — -fsanitize applies to the host pass only). Run tests
# with LD_PRELOAD of libasan (tools/asan_check.sh). Objects/so land in
# separate paths so a sanitized build never ships by accident.
ASAN = os.environ.get("MASKRCNN_ASAN", "0") == "1"


def torch_paths():
    import torch
    from torch.utils import cpp_extension as ce
    return ce.include_paths(), ce.library_paths(), \
        torch._C._GLIBCXX_USE_CXX11_ABI


def build(verbose: bool = True) -> Path:
    BUILD = ROOT / "build" / ("ext_asan" if ASAN else "ext")
    OUT = ROOT / "maskrcnn_amd" / ("_C_asan.so" if ASAN else "_C.so")
    inc, libs, cxx11 = torch_paths()
    py_inc = sysconfig.get_paths()["include"]
    BUILD.mkdir(parents=True, exist_ok=True)

    common = [
        "-O3", "-std=c++17", "-fPIC", f"--offload-arch={ARCH}",
        "-DUSE_ROCM=1", "-D__HIP_PLATFORM_AMD__=1",
        "-DHIP_ENABLE_WARP_SYNC_BUILTINS=1",
        f"-D_GLIBCXX_USE_CXX11_ABI={1 if cxx11 else 0}",
        "-DTORCH_EXTENSION_NAME=_C",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-fno-gpu-rdc",
        "-Wno-deprecated-declarations",
        "-Wno-unused-result",
    ]
    if ASAN:
        common += ["-fsanitize=address", "-shared-libasan", "-g"]
    incs = [f"-I{p}" for p in inc] + [f"-I{py_inc}", f"-I{CSRC}"]

    # ingress.cpp is a SEPARATE host-only product (_ingress.so via
    # build_ingress — needs -lnghttp2); sweeping it into _C.so leaves
    # an undefined nghttp2 symbol that breaks the core import.
    sources = sorted(CSRC.glob("hip/*.hip")) + \
        sorted(p for p in CSRC.glob("*.cpp") if p.name != "ingress.cpp")
    objs = []

    def compile_one(src: Path) -> Path:
        obj = BUILD / (src.stem + ".o")
        # Skip recompilation if up to date.
        deps = [src] + list(CSRC.glob("hip/*.h"))
        if obj.exists() and all(obj.stat().st_mtime > d.stat().st_mtime
                                for d in deps):
            return obj
        cmd = ["hipcc", "-c", str(src), "-o", str(obj)] + common + incs
        if src.suffix == ".cpp":
            cmd += ["-x", "hip"]  # bindings include hip_runtime
        if verbose:
            print(" ".join(cmd[:4]), "...")
        subprocess.run(cmd, check=True)
        return obj

    with ThreadPoolExecutor(max_workers=8) as pool:
        objs = list(pool.map(compile_one, sources))

    link = ["hipcc", "-shared", "-fPIC", "-o", str(OUT)] + \
        [str(o) for o in objs] + \
        [f"-L{p}" for p in libs] + \
        ["-ltorch", "-ltorch_cpu", "-ltorch_python", "-lc10",
         "-lc10_hip", "-ltorch_hip",
         f"-Wl,-rpath,{libs[0]}"]
    if ASAN:
        link += ["-fsanitize=address", "-shared-libasan"]
    if verbose:
        print(" ".join(link[:6]), "...")
    subprocess.run(link, check=True)
    print(f"built {OUT}")
    return OUT


def build_ingress(verbose: bool = True) -> Path:
    """Build maskrcnn_amd/_ingress.so — the native HTTP/2 gRPC ingress.

    Host-only C++ (no HIP, no torch): pybind11 + the system
    libnghttp2.so.14 (HPACK). Linked with -l:NAME.so.14 because the
    image ships the runtime library but not the dev symlink/headers.
    """
    import pybind11
    OUT = ROOT / "maskrcnn_amd" / "_ingress.so"
    src = CSRC / "ingress.cpp"
    if OUT.exists() and OUT.stat().st_mtime > src.stat().st_mtime:
        return OUT
    py_inc = sysconfig.get_paths()["include"]
    cmd = ["g++", "-O2", "-std=c++17", "-shared", "-fPIC",
           str(src), "-o", str(OUT),
           f"-I{pybind11.get_include()}", f"-I{py_inc}",
           "-l:libnghttp2.so.14", "-lpthread",
           "-fvisibility=hidden"]
    if verbose:
        print(" ".join(cmd[:6]), "...")
    subprocess.run(cmd, check=True)
    print(f"built {OUT}")
    return OUT


if __name__ == "__main__":
    build()
    build_ingress()
