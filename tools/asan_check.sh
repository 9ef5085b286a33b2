#!/bin/bash
# Host-ASAN pass over the extension's CPU-reachable paths (SURVEY §5
# sanitizers row / VERDICT r1 item 8): builds maskrcnn_amd/_C_asan.so
# with -fsanitize=address (host side; device code is uninstrumented),
# preloads clang's shared libasan, and runs the CPU test suite with the
# sanitized extension swapped in.
#
#   bash tools/asan_check.sh [pytest args...]
set -e
cd "$(dirname "$0")/.."
MASKRCNN_ASAN=1 python build_ext.py
ASAN_RT=$(hipcc -print-file-name=libclang_rt.asan-x86_64.so)
cp maskrcnn_amd/_C.so /tmp/_C_real.so
cp maskrcnn_amd/_C_asan.so maskrcnn_amd/_C.so
trap 'cp /tmp/_C_real.so maskrcnn_amd/_C.so' EXIT
# detect_leaks off: python+torch leak-at-exit noise buries real reports
LD_PRELOAD="$ASAN_RT" ASAN_OPTIONS=detect_leaks=0:abort_on_error=1 \
    python -m pytest tests/test_ops_cpu.py tests/test_eval.py \
    -q -m "not gpu" "$@"
echo "ASAN pass clean"
