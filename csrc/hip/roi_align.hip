// RoIAlign / crop_and_resize forward + backward — CDNA4 (gfx950) kernels.
//
// Replaces the reference's tf.image.crop_and_resize CUDA kernel (SURVEY.md
// N2; used by PyramidROIAlign and mask-target cropping). Two modes:
//   sampling == 0: TF crop_and_resize parity — corner-aligned PHxPW grid,
//                  bilinear, out-of-bounds corners contribute zero
//                  (grid_sample align_corners=True + zeros padding).
//   sampling >= 1: true RoIAlign — sampling^2 bin-centered samples averaged.
// Boxes are normalized [y1,x1,y2,x2]; pixel = norm * (dim-1).
//
// Layout-polymorphic via strides: supports NCHW and NHWC (channels_last)
// without copies. Thread order puts the fastest-varying *memory* axis on
// consecutive lanes (c for NHWC, pw for NCHW) so the 4 bilinear-corner
// gathers coalesce into full 64-lane lines in the NHWC path.
//
// Memory-bound gather: the roofline is HBM/L2 bandwidth, not MFMA.
#include "common.h"

namespace {

struct Strides {
  long sB, sC, sH, sW;  // input feature strides (elements)
};

template <typename T>
__device__ inline float bilinear_zero_pad(const T* feat, int H, int W,
                                          const Strides s, float y, float x) {
  // grid_sample(align_corners=True, padding_mode="zeros") semantics:
  // corners outside the image contribute 0 with unchanged weights.
  int y0 = static_cast<int>(floorf(y));
  int x0 = static_cast<int>(floorf(x));
  int y1 = y0 + 1, x1 = x0 + 1;
  float ly = y - y0, lx = x - x0;
  float hy = 1.f - ly, hx = 1.f - lx;
  float v = 0.f;
  bool y0i = (y0 >= 0 && y0 < H), y1i = (y1 >= 0 && y1 < H);
  bool x0i = (x0 >= 0 && x0 < W), x1i = (x1 >= 0 && x1 < W);
  if (y0i && x0i) v += hy * hx * ld(feat + y0 * s.sH + x0 * s.sW);
  if (y0i && x1i) v += hy * lx * ld(feat + y0 * s.sH + x1 * s.sW);
  if (y1i && x0i) v += ly * hx * ld(feat + y1 * s.sH + x0 * s.sW);
  if (y1i && x1i) v += ly * lx * ld(feat + y1 * s.sH + x1 * s.sW);
  return v;
}

// One output element per thread. `cl` selects the lane->element mapping.
template <typename T, bool CL>
__global__ void roi_align_fwd_kernel(
    const T* __restrict__ feats, const float* __restrict__ boxes,
    const int* __restrict__ box_ind, T* __restrict__ out,
    int N, int C, int H, int W, int PH, int PW,
    Strides fs, Strides os, int sampling) {
  long total = (long)N * C * PH * PW;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    int n, c, ph, pw;
    if (CL) {  // c innermost (NHWC memory): lanes stride the C axis
      c = idx % C;
      long r = idx / C;
      pw = r % PW; r /= PW;
      ph = r % PH; n = r / PH;
    } else {   // pw innermost (NCHW memory)
      pw = idx % PW;
      long r = idx / PW;
      ph = r % PH; r /= PH;
      c = r % C;  n = r / C;
    }
    const float y1 = boxes[n * 4 + 0], x1 = boxes[n * 4 + 1];
    const float y2 = boxes[n * 4 + 2], x2 = boxes[n * 4 + 3];
    const int b = box_ind[n];
    const T* f = feats + (long)b * fs.sB + (long)c * fs.sC;

    float acc;
    if (sampling == 0) {
      float yy = (PH > 1) ? (y1 + (y2 - y1) * ph / (PH - 1))
                          : 0.5f * (y1 + y2);
      float xx = (PW > 1) ? (x1 + (x2 - x1) * pw / (PW - 1))
                          : 0.5f * (x1 + x2);
      acc = bilinear_zero_pad<T>(f, H, W, fs, yy * (H - 1), xx * (W - 1));
    } else {
      const float y1p = y1 * (H - 1), x1p = x1 * (W - 1);
      const float rh = (y2 - y1) * (H - 1), rw = (x2 - x1) * (W - 1);
      const int S = sampling;
      float sum = 0.f;
      for (int sy = 0; sy < S; ++sy) {
        float yy = y1p + rh * (ph * S + sy + 0.5f) / (PH * S);
        for (int sx = 0; sx < S; ++sx) {
          float xx = x1p + rw * (pw * S + sx + 0.5f) / (PW * S);
          sum += bilinear_zero_pad<T>(f, H, W, fs, yy, xx);
        }
      }
      acc = sum / (S * S);
    }
    out[(long)n * os.sB + (long)c * os.sC + (long)ph * os.sH +
        (long)pw * os.sW] = static_cast<T>(acc);
  }
}

template <typename T, bool CL>
__global__ void roi_align_bwd_kernel(
    const T* __restrict__ grad_out, const float* __restrict__ boxes,
    const int* __restrict__ box_ind, float* __restrict__ grad_in,
    int N, int C, int H, int W, int PH, int PW,
    Strides gs, Strides os, int sampling) {
  long total = (long)N * C * PH * PW;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    int n, c, ph, pw;
    if (CL) {
      c = idx % C;
      long r = idx / C;
      pw = r % PW; r /= PW;
      ph = r % PH; n = r / PH;
    } else {
      pw = idx % PW;
      long r = idx / PW;
      ph = r % PH; r /= PH;
      c = r % C;  n = r / C;
    }
    const float y1 = boxes[n * 4 + 0], x1 = boxes[n * 4 + 1];
    const float y2 = boxes[n * 4 + 2], x2 = boxes[n * 4 + 3];
    const int b = box_ind[n];
    float* g = grad_in + (long)b * gs.sB + (long)c * gs.sC;
    const float go = ld(grad_out + (long)n * os.sB + (long)c * os.sC +
                        (long)ph * os.sH + (long)pw * os.sW);
    if (go == 0.f) continue;

    const int S = (sampling == 0) ? 1 : sampling;
    const float w_s = (sampling == 0) ? 1.f : 1.f / (S * S);
    for (int sy = 0; sy < S; ++sy) {
      float yy;
      if (sampling == 0) {
        yy = ((PH > 1) ? (y1 + (y2 - y1) * ph / (PH - 1))
                       : 0.5f * (y1 + y2)) * (H - 1);
      } else {
        yy = y1 * (H - 1) + (y2 - y1) * (H - 1) *
                                (ph * S + sy + 0.5f) / (PH * S);
      }
      for (int sx = 0; sx < S; ++sx) {
        float xx;
        if (sampling == 0) {
          xx = ((PW > 1) ? (x1 + (x2 - x1) * pw / (PW - 1))
                         : 0.5f * (x1 + x2)) * (W - 1);
        } else {
          xx = x1 * (W - 1) + (x2 - x1) * (W - 1) *
                                  (pw * S + sx + 0.5f) / (PW * S);
        }
        int y0 = static_cast<int>(floorf(yy));
        int x0 = static_cast<int>(floorf(xx));
        int y1i = y0 + 1, x1i = x0 + 1;
        float ly = yy - y0, lx = xx - x0;
        float hy = 1.f - ly, hx = 1.f - lx;
        const float gw = go * w_s;
        if (y0 >= 0 && y0 < H && x0 >= 0 && x0 < W)
          atomicAdd(g + y0 * gs.sH + x0 * gs.sW, gw * hy * hx);
        if (y0 >= 0 && y0 < H && x1i >= 0 && x1i < W)
          atomicAdd(g + y0 * gs.sH + x1i * gs.sW, gw * hy * lx);
        if (y1i >= 0 && y1i < H && x0 >= 0 && x0 < W)
          atomicAdd(g + y1i * gs.sH + x0 * gs.sW, gw * ly * hx);
        if (y1i >= 0 && y1i < H && x1i >= 0 && x1i < W)
          atomicAdd(g + y1i * gs.sH + x1i * gs.sW, gw * ly * lx);
      }
    }
  }
}

// Deterministic-reduction backward (SURVEY §5 sanitizers row): the float
// atomicAdd scatter above is order-dependent (fp addition is not
// associative), so grads can differ bitwise between runs. This variant
// accumulates FIXED-POINT contributions with integer atomics — integer
// addition is associative and commutative, so the result is bitwise
// identical for every scheduling order. scale is chosen by the host so
// the worst-case sum stays far below 2^63.
template <typename T, bool CL>
__global__ void roi_align_bwd_det_kernel(
    const T* __restrict__ grad_out, const float* __restrict__ boxes,
    const int* __restrict__ box_ind, unsigned long long* __restrict__ acc,
    int N, int C, int H, int W, int PH, int PW,
    Strides gs, Strides os, int sampling, float scale) {
  long total = (long)N * C * PH * PW;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    int n, c, ph, pw;
    if (CL) {
      c = idx % C;
      long r = idx / C;
      pw = r % PW; r /= PW;
      ph = r % PH; n = r / PH;
    } else {
      pw = idx % PW;
      long r = idx / PW;
      ph = r % PH; r /= PH;
      c = r % C;  n = r / C;
    }
    const float y1 = boxes[n * 4 + 0], x1 = boxes[n * 4 + 1];
    const float y2 = boxes[n * 4 + 2], x2 = boxes[n * 4 + 3];
    const int b = box_ind[n];
    unsigned long long* g = acc + (long)b * gs.sB + (long)c * gs.sC;
    const float go = ld(grad_out + (long)n * os.sB + (long)c * os.sC +
                        (long)ph * os.sH + (long)pw * os.sW);
    if (go == 0.f) continue;

    const int S = (sampling == 0) ? 1 : sampling;
    const float w_s = (sampling == 0) ? 1.f : 1.f / (S * S);
    for (int sy = 0; sy < S; ++sy) {
      float yy;
      if (sampling == 0) {
        yy = ((PH > 1) ? (y1 + (y2 - y1) * ph / (PH - 1))
                       : 0.5f * (y1 + y2)) * (H - 1);
      } else {
        yy = y1 * (H - 1) + (y2 - y1) * (H - 1) *
                                (ph * S + sy + 0.5f) / (PH * S);
      }
      for (int sx = 0; sx < S; ++sx) {
        float xx;
        if (sampling == 0) {
          xx = ((PW > 1) ? (x1 + (x2 - x1) * pw / (PW - 1))
                         : 0.5f * (x1 + x2)) * (W - 1);
        } else {
          xx = x1 * (W - 1) + (x2 - x1) * (W - 1) *
                                  (pw * S + sx + 0.5f) / (PW * S);
        }
        int y0 = static_cast<int>(floorf(yy));
        int x0 = static_cast<int>(floorf(xx));
        int y1i = y0 + 1, x1i = x0 + 1;
        float ly = yy - y0, lx = xx - x0;
        float hy = 1.f - ly, hx = 1.f - lx;
        const float gw = go * w_s;
#define DET_ADD(ptr, v)                                                     \
  atomicAdd((ptr), (unsigned long long)(long long)llrintf((v) * scale))
        if (y0 >= 0 && y0 < H && x0 >= 0 && x0 < W)
          DET_ADD(g + y0 * gs.sH + x0 * gs.sW, gw * hy * hx);
        if (y0 >= 0 && y0 < H && x1i >= 0 && x1i < W)
          DET_ADD(g + y0 * gs.sH + x1i * gs.sW, gw * hy * lx);
        if (y1i >= 0 && y1i < H && x0 >= 0 && x0 < W)
          DET_ADD(g + y1i * gs.sH + x0 * gs.sW, gw * ly * hx);
        if (y1i >= 0 && y1i < H && x1i >= 0 && x1i < W)
          DET_ADD(g + y1i * gs.sH + x1i * gs.sW, gw * ly * lx);
#undef DET_ADD
      }
    }
  }
}

__global__ void fixed_to_float_kernel(const unsigned long long* __restrict__
                                          acc,
                                      float* __restrict__ out, long total,
                                      float inv_scale) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x)
    out[i] = (float)(long long)acc[i] * inv_scale;
}

inline int grid_for(long total, int block) {
  long g = (total + block - 1) / block;
  // >> 256 workgroups to fill 8 XCDs; cap and grid-stride the rest.
  if (g > 8192) g = 8192;
  return (int)g;
}

}  // namespace

extern "C" {

void roi_align_fwd_launch(const void* feats, const float* boxes,
                          const int* box_ind, void* out, int N, int C,
                          int H, int W, int PH, int PW, long fsB, long fsC,
                          long fsH, long fsW, long osB, long osC, long osH,
                          long osW, int sampling, bool channels_last,
                          bool is_bf16, hipStream_t stream) {
  Strides fs{fsB, fsC, fsH, fsW}, os{osB, osC, osH, osW};
  long total = (long)N * C * PH * PW;
  if (total == 0) return;
  const int block = 256;
  dim3 grid(grid_for(total, block));
#define LAUNCH(T, CL)                                                      \
  hipLaunchKernelGGL((roi_align_fwd_kernel<T, CL>), grid, dim3(block), 0,  \
                     stream, (const T*)feats, boxes, box_ind, (T*)out, N,  \
                     C, H, W, PH, PW, fs, os, sampling)
  if (is_bf16) {
    if (channels_last) LAUNCH(__hip_bfloat16, true);
    else LAUNCH(__hip_bfloat16, false);
  } else {
    if (channels_last) LAUNCH(float, true);
    else LAUNCH(float, false);
  }
#undef LAUNCH
}

void roi_align_bwd_launch(const void* grad_out, const float* boxes,
                          const int* box_ind, float* grad_in, int N, int C,
                          int H, int W, int PH, int PW, long gsB, long gsC,
                          long gsH, long gsW, long osB, long osC, long osH,
                          long osW, int sampling, bool channels_last,
                          bool is_bf16, hipStream_t stream) {
  Strides gs{gsB, gsC, gsH, gsW}, os{osB, osC, osH, osW};
  long total = (long)N * C * PH * PW;
  if (total == 0) return;
  const int block = 256;
  dim3 grid(grid_for(total, block));
#define LAUNCH(T, CL)                                                      \
  hipLaunchKernelGGL((roi_align_bwd_kernel<T, CL>), grid, dim3(block), 0,  \
                     stream, (const T*)grad_out, boxes, box_ind, grad_in,  \
                     N, C, H, W, PH, PW, gs, os, sampling)
  if (is_bf16) {
    if (channels_last) LAUNCH(__hip_bfloat16, true);
    else LAUNCH(__hip_bfloat16, false);
  } else {
    if (channels_last) LAUNCH(float, true);
    else LAUNCH(float, false);
  }
#undef LAUNCH
}

void roi_align_bwd_det_launch(const void* grad_out, const float* boxes,
                              const int* box_ind,
                              unsigned long long* acc_ws, float* grad_in,
                              int N, int C, int H, int W, int PH, int PW,
                              long gsB, long gsC, long gsH, long gsW,
                              long osB, long osC, long osH, long osW,
                              int sampling, bool channels_last, bool is_bf16,
                              float scale, hipStream_t stream) {
  Strides gs{gsB, gsC, gsH, gsW}, os{osB, osC, osH, osW};
  long total = (long)N * C * PH * PW;
  if (total == 0) return;
  const int block = 256;
  dim3 grid(grid_for(total, block));
#define LAUNCH(T, CL)                                                       \
  hipLaunchKernelGGL((roi_align_bwd_det_kernel<T, CL>), grid, dim3(block),  \
                     0, stream, (const T*)grad_out, boxes, box_ind,         \
                     acc_ws, N, C, H, W, PH, PW, gs, os, sampling, scale)
  if (is_bf16) {
    if (channels_last) LAUNCH(__hip_bfloat16, true);
    else LAUNCH(__hip_bfloat16, false);
  } else {
    if (channels_last) LAUNCH(float, true);
    else LAUNCH(float, false);
  }
#undef LAUNCH
}

void fixed_to_float_launch(const unsigned long long* acc, float* out,
                           long total, float inv_scale,
                           hipStream_t stream) {
  if (total == 0) return;
  const int block = 256;
  dim3 grid(grid_for(total, block));
  hipLaunchKernelGGL(fixed_to_float_kernel, grid, dim3(block), 0, stream,
                     acc, out, total, inv_scale);
}

}  // extern "C"
