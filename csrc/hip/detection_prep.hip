// Detection post-process prep — fused argmax/gather/refine/clip (N9).
//
// Replaces the ~8 elementwise/gather torch launches in front of the
// detection NMS (DetectionLayer: argmax class per roi, class-specific
// delta * BBOX_STD_DEV, apply to roi, clip to the per-image window,
// background/score masking) with ONE kernel, one thread per (b, n) roi.
// Feeds nms directly: scores of dropped rois are zeroed, which cannot
// perturb NMS of positive-score boxes (see model/detection.py
// _forward_syncfree — same zero-score masking contract).
//
// All math fp32 (probs/deltas may arrive bf16 from the autocast
// classifier head; refined boxes feed IoU tests, so fp32 here).
#include "common.h"

namespace {

template <typename T>
__global__ void detection_prep_kernel(
    const float* __restrict__ rois,    // [B,N,4] normalized (ALWAYS
                                       // fp32: bf16 coords lose ~2 px
                                       // at 1024² — probs/deltas come
                                       // from the bf16 head, rois from
                                       // the fp32 proposal layer)
    const T* __restrict__ probs,       // [B,N,C]
    const T* __restrict__ deltas,      // [B,N,C,4]
    const float* __restrict__ windows, // [B,4] normalized y1x1y2x2
    const float* __restrict__ std_dev, // [4]
    float* __restrict__ refined,       // [B,N,4]
    int* __restrict__ class_ids,       // [B,N]
    float* __restrict__ scores,        // [B,N] (0 for dropped rois)
    int B, int N, int C, float min_conf) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= (long)B * N) return;
  const int b = (int)(i / N);

  // argmax over classes.
  const T* p = probs + i * C;
  int best = 0;
  float best_v = ld(p);
  for (int c = 1; c < C; ++c) {
    const float v = ld(p + c);
    if (v > best_v) { best_v = v; best = c; }
  }

  // class-specific delta * std_dev applied to the roi.
  const T* d = deltas + (i * C + best) * 4;
  const float dy = ld(d + 0) * std_dev[0];
  const float dx = ld(d + 1) * std_dev[1];
  const float dh = ld(d + 2) * std_dev[2];
  const float dw = ld(d + 3) * std_dev[3];
  const float* r = rois + i * 4;
  float y1 = r[0], x1 = r[1], y2 = r[2], x2 = r[3];
  float h = y2 - y1, w = x2 - x1;
  float cy = y1 + 0.5f * h + dy * h;
  float cx = x1 + 0.5f * w + dx * w;
  h *= __expf(dh);
  w *= __expf(dw);
  y1 = cy - 0.5f * h; x1 = cx - 0.5f * w;
  y2 = y1 + h;        x2 = x1 + w;

  // clip to this image's window.
  const float wy1 = windows[b * 4 + 0], wx1 = windows[b * 4 + 1];
  const float wy2 = windows[b * 4 + 2], wx2 = windows[b * 4 + 3];
  y1 = fminf(fmaxf(y1, wy1), wy2);
  x1 = fminf(fmaxf(x1, wx1), wx2);
  y2 = fminf(fmaxf(y2, wy1), wy2);
  x2 = fminf(fmaxf(x2, wx1), wx2);

  refined[i * 4 + 0] = y1;
  refined[i * 4 + 1] = x1;
  refined[i * 4 + 2] = y2;
  refined[i * 4 + 3] = x2;
  class_ids[i] = best;
  const bool keep = best > 0 && (min_conf <= 0.f || best_v >= min_conf);
  scores[i] = keep ? best_v : 0.f;
}

}  // namespace

extern "C" {

void detection_prep_launch(const void* rois, const void* probs,
                           const void* deltas, const float* windows,
                           const float* std_dev, float* refined,
                           int* class_ids, float* scores, int B, int N,
                           int C, float min_conf, bool bf16,
                           hipStream_t stream) {
  const long total = (long)B * N;
  dim3 grid((unsigned)ceil_div((int)total, 256));
  if (bf16) {
    hipLaunchKernelGGL(detection_prep_kernel<__hip_bfloat16>, grid,
                       dim3(256), 0, stream, (const float*)rois,
                       (const __hip_bfloat16*)probs,
                       (const __hip_bfloat16*)deltas, windows, std_dev,
                       refined, class_ids, scores, B, N, C, min_conf);
  } else {
    hipLaunchKernelGGL(detection_prep_kernel<float>, grid, dim3(256), 0,
                       stream, (const float*)rois, (const float*)probs,
                       (const float*)deltas, windows, std_dev, refined,
                       class_ids, scores, B, N, C, min_conf);
  }
}

}  // extern "C"
