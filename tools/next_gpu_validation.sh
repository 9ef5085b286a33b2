#!/bin/bash
# Runbook for the FIRST GPU access after the round-2 continuation
# session (which landed CPU-validated work after the GPU budget
# closed). Run blocks in order; each is independently useful.
# Usage: bash tools/next_gpu_validation.sh [block1|block2|block3|all]
set -x
cd "$(dirname "$0")/.." || exit 1
mkdir -p gpurun_out

block1() {  # split-K conv (csrc/hip/conv3x3_splitk.hip, version=6)
  # numerics vs fp32 reference + determinism
  MASKRCNN_HIP_CONV_SPLITK=1 timeout 600 python -m pytest \
    tests/test_conv3x3.py -k splitk -m gpu -x -q \
    2>&1 | tee gpurun_out/v6_numerics.log
  # A/B vs MIOpen on the fill-bound target shape
  timeout 900 python tools/conv_bench.py \
    2>&1 | tee gpurun_out/v6_ab.log
  # If v6 >= 1.0x MIOpen on (2,512,32,32,512): flip the auto-dispatch
  # in csrc/bindings.cpp (conv3x3_fwd ver==0 branch) to route
  # M <= 8192 && Cout >= 512 -> 6, rebuild, re-run tests + bench.py.
}

block2() {  # native C++ gRPC ingress full-stack serving number
  timeout 1800 python bench/serving_bench.py --grpc-ingress cpp \
    --requests 192 --client-procs 8 --client-threads 4 \
    2>&1 | tee gpurun_out/serving_cpp_ingress.log
  # compare against the python-ingress 28.6 img/s record
  # (profiles/r02_conv_ab.md); if CPU-side 1.3-2.1x holds, expect
  # 35-60 img/s. Also worth one --server-procs 2 run.
}

block3() {  # serving soak on the GPU (cpp ingress, real model)
  timeout 900 python tools/soak_serving.py --seconds 300 --threads 8 \
    --grpc-ingress cpp --image-size 512 \
    2>&1 | tee gpurun_out/soak_gpu.log
}

case "${1:-all}" in
  block1) block1 ;;
  block2) block2 ;;
  block3) block3 ;;
  all) block1; block2; block3 ;;
esac
