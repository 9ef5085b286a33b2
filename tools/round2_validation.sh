#!/bin/bash
# Round-2 opening GPU session plan — spend the fresh 90 GPU-min budget
# front-loaded. Run pieces via gpurun, NOT all at once; each block is
# independently useful. See NOTES_ROUND2.md for context.
#
#   gpurun --timeout 1200 -- 'bash tools/round2_validation.sh block1'
set -x
cd "${GRAFT_REPO_ROOT:-/root/repo}"
mkdir -p gpurun_out

case "${1:-block1}" in
block1)  # v3 + wrw + fp8 kernel validation + conv A/B (~5 min)
  timeout 300 python tools/check_gpu_env.py || exit 1
  timeout 500 python -m pytest tests/test_conv3x3.py tests/test_ops_gpu.py \
      -m gpu -q --runxfail > gpurun_out/r2_conv_tests.log 2>&1
  tail -4 gpurun_out/r2_conv_tests.log
  timeout 400 python tools/conv_bench.py > gpurun_out/r2_conv_ab.log 2>&1
  grep -v libdrm gpurun_out/r2_conv_ab.log
  # hipGraph serving replay (engine._forward capture path)
  timeout 300 python -m pytest \
      tests/test_zz_hipgraph_gpu.py::test_engine_hipgraph_matches_eager \
      -m gpu -q --runxfail >> gpurun_out/r2_conv_tests.log 2>&1
  tail -2 gpurun_out/r2_conv_tests.log
  ;;
block2)  # training-path A/Bs (one box; ~12 min incl. one find phase)
  run() { timeout 500 python bench.py --steps 15 --warmup 5 \
          > "gpurun_out/r2_$1.json" 2>/dev/null; cat "gpurun_out/r2_$1.json"; }
  echo "=== baseline"; run base
  echo "=== fused fpn";        MASKRCNN_TRAIN_FUSED_FPN=1 run fpn
  echo "=== hip conv";         MASKRCNN_TRAIN_HIP_CONV=1 run conv
  echo "=== cl roi";           MASKRCNN_TRAIN_CL_ROI=1 run clroi
  echo "=== all three";        MASKRCNN_TRAIN_FUSED_FPN=1 \
      MASKRCNN_TRAIN_HIP_CONV=1 MASKRCNN_TRAIN_CL_ROI=1 run all
  echo "=== vectorized targets"; MASKRCNN_VECTORIZED_TARGETS=1 run vtgt
  echo "=== hip wrw";          MASKRCNN_TRAIN_HIP_CONV=1 \
      MASKRCNN_HIP_WRW=1 run wrw
  ;;
block3)  # serving bench (~8 min incl. 265 s batch-32 find)
  # Capture the MIOpen user find-db in gpurun_out so it merges back and
  # can be committed as repo miopen_db/ (fresh boxes then skip the
  # 265 s batch-32 Find; _rocm_env exports MIOPEN_USER_DB_PATH).
  mkdir -p gpurun_out/miopen_db
  if [ -d miopen_db ] && [ -n "$(ls miopen_db 2>/dev/null)" ]; then
    cp miopen_db/* gpurun_out/miopen_db/ 2>/dev/null
  fi
  export MIOPEN_USER_DB_PATH="$PWD/gpurun_out/miopen_db"
  export MIOPEN_FIND_MODE=5
  timeout 900 python bench/serving_bench.py --requests 96 \
      --client-procs 4 --client-threads 8 --warmup 16 \
      > gpurun_out/r2_serving.json 2> gpurun_out/r2_serving.err
  cat gpurun_out/r2_serving.json
  grep -E "WARNING|batch of" gpurun_out/r2_serving.err | tail -5
  ls -la gpurun_out/miopen_db/
  # A/B: hipGraph replay of the engine forward (validated in block1)
  MASKRCNN_SERVING_HIPGRAPH=1 timeout 700 python bench/serving_bench.py \
      --requests 96 --client-procs 4 --client-threads 8 --warmup 16 \
      > gpurun_out/r2_serving_graph.json 2> gpurun_out/r2_serving_graph.err
  cat gpurun_out/r2_serving_graph.json
  tail -12 gpurun_out/r2_serving_graph.err
  ;;
block4)  # steady-state rocprof of the tuned step (profile AFTER warmup
         # by profiling a second process on the warm box: find-db is
         # box-local, so run bench once unprofiled first)
  cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
  timeout 400 python bench.py --steps 3 --warmup 2 > /dev/null 2>&1
  mkdir -p gpurun_out/r2_prof
  timeout 500 rocprofv3 --kernel-trace --stats --output-format csv \
      -d gpurun_out/r2_prof -o r2 -- \
      python bench.py --steps 15 --warmup 3 \
      > gpurun_out/r2_prof_bench.json 2>/dev/null
  cat gpurun_out/r2_prof_bench.json
  python tools/gap_analysis.py "gpurun_out/r2_prof/*kernel_trace.csv"
  find gpurun_out/r2_prof -name "*kernel_trace*" -delete
  python tools/prof_summary.py "gpurun_out/r2_prof/*kernel_stats.csv" 25
  ;;
block5)  # PMC counters for the hand kernels (own run, never with traces)
  cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
  mkdir -p gpurun_out/r2_pmc
  timeout 500 rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES,SQ_WAVE_CYCLES,SQ_LDS_BANK_CONFLICT \
      --output-format csv -d gpurun_out/r2_pmc -o pmc -- \
      python tools/conv_bench.py > gpurun_out/r2_pmc.log 2>&1
  ls gpurun_out/r2_pmc
  ;;
esac
if [ "${1:-}" = "block7" ]; then
  # Final default-flag sweep at 30 steps (block2's 15-step singles were
  # within ±0.5 ms of each other; this pins the shipping combination).
  run7() { timeout 500 python bench.py --steps 30 --warmup 5 \
           > "gpurun_out/r7_$1.json" 2>/dev/null; cat "gpurun_out/r7_$1.json"; }
  export MIOPEN_FIND_MODE=5
  echo "=== all-on (shipping defaults)"; run7 all
  echo "=== vtgt only"; MASKRCNN_TRAIN_FUSED_FPN=0 MASKRCNN_TRAIN_CL_ROI=0 run7 vtgt
  echo "=== vtgt+fpn"; MASKRCNN_TRAIN_CL_ROI=0 run7 vtgt_fpn
  echo "=== vtgt+clroi"; MASKRCNN_TRAIN_FUSED_FPN=0 run7 vtgt_clroi
fi

# (appended) block6: fp8-weight large-image serving (BASELINE config #5,
# 1333 -> padded 1344; DP=8 scale-out is the driver's 8-GPU tier — this
# measures the per-GPU replica)
if [ "${1:-}" = "block6" ]; then
  timeout 1100 python bench/serving_bench.py --requests 48 \
      --client-procs 4 --client-threads 8 --warmup 8 \
      --image-size 1344 --weight-dtype fp8 --max-batch 16 \
      > gpurun_out/r2_fp8_serving.json 2> gpurun_out/r2_fp8_serving.err
  cat gpurun_out/r2_fp8_serving.json
  # A/B: fp8 COMPUTE in the mask head (kernel validated in block1)
  MASKRCNN_FP8_COMPUTE=1 timeout 800 python bench/serving_bench.py \
      --requests 48 --client-procs 4 --client-threads 8 --warmup 8 \
      --image-size 1344 --weight-dtype fp8 --max-batch 16 \
      > gpurun_out/r2_fp8c_serving.json 2>/dev/null
  cat gpurun_out/r2_fp8c_serving.json
fi
