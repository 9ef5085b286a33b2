// 3x3 conv (stride 1, pad 1) WEIGHT gradient — CDNA4 MFMA kernel.
//
// SURVEY.md N1 (training backward): wrw was 47% of the original MIOpen
// step profile (profiles/r01_naive_conv.md). GEMM view per tap (r,s):
//   dW[co,ci] = sum_pixels dy[p, co] * x[p + shift(r,s), ci]
// i.e. D = A^T B with A = dy [K x Cout], B = shifted x [K x Cin], both
// channels-innermost (NHWC/channels_last) so K (pixels) is the outer
// stride — staged transposed into LDS as [channel][k] rows and fed to
// v_mfma_f32_16x16x32_bf16 (m = co, n = ci, k = 32 pixels of one image
// row segment).
//
// Tap grouping: blockIdx.y encodes (ci_tile, dr). A workgroup stages ONE
// x input row (h + dr - 1, with 1-column halo) and ONE dy row segment,
// and accumulates the 3 taps sharing that dr (dc = 0..2 are column
// shifts into the staged halo row). Global traffic is 3x the tensors
// instead of 9x; accumulators stay at 3 taps x 2x2 fragments = 48 VGPRs
// per wave. Workgroups split the (image-row-segment) K space via
// blockIdx.z and combine with fp32 atomicAdd into dw (caller zero-fills).
//
// Correctness-first structure (single-buffered LDS, scalar B-fragment
// packing for the misaligned dc shifts); the tuning ladder (glds,
// counted vmcnt, aligned triple-staging of the dc shifts, XCD swizzle)
// is round-2 work after GPU validation.
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) short bf16x8;

namespace {

constexpr int CO_T = 64;     // M tile (Cout)
constexpr int CI_T = 64;     // N tile (Cin)
constexpr int SW = 32;       // K chunk: 32 pixels of one image row
constexpr int XCOLS = SW + 2;  // staged x row: SW + 1-col halo each side
constexpr int KP = 40;       // LDS k-pitch (shorts): 80 B, 16-B aligned

__global__ __launch_bounds__(256, 2)
void conv3x3_wrw_kernel(const __hip_bfloat16* __restrict__ x,   // [B,H,W,Cin]
                        const __hip_bfloat16* __restrict__ dy,  // [B,H,W,Cout]
                        float* __restrict__ dw,  // [9,Cout,Cin] zero-filled
                        int Bn, int H, int W, int Cin, int Cout) {
  const int co0 = blockIdx.x * CO_T;
  const int ci_tile = blockIdx.y / 3;
  const int dr = blockIdx.y % 3;          // input row offset - 1
  const int ci0 = ci_tile * CI_T;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wave_m = wave >> 1;           // co quadrant
  const int wave_n = wave & 1;            // ci quadrant

  // lds_dy[co][k]: transposed stage of dy[w0+k][co0+co].
  // lds_x[ci][c]:  transposed stage of x row h+dr-1, c -> gx = w0-1+c.
  __shared__ short lds_dy[CO_T * KP];
  __shared__ short lds_x[CI_T * KP];

  f32x4 acc[3][2][2] = {};  // [dc][mi][ni]

  const int spr = (W + SW - 1) / SW;      // segments per image row
  const long total_segs = (long)Bn * H * spr;

  for (long seg = blockIdx.z; seg < total_segs; seg += gridDim.z) {
    const int w0 = (int)(seg % spr) * SW;
    const int h = (int)((seg / spr) % H);
    const long n = seg / ((long)spr * H);
    const int gy = h + dr - 1;            // x source row
    if (gy < 0 || gy >= H) continue;      // zero B => zero contribution

    // Stage dy segment: thread -> (k = tid>>3, 8 channels at (tid&7)*8).
    {
      const int k = tid >> 3;
      const int c8 = (tid & 7) * 8;
      short4* dst = (short4*)&lds_dy[0];
      const int gw = w0 + k;
      short v[8] = {};
      if (gw < W && co0 + c8 < Cout) {
        const short* src = (const short*)dy +
            (((n * H + h) * W + gw) * (long)Cout + co0 + c8);
        *(short4*)&v[0] = ((const short4*)src)[0];
        *(short4*)&v[4] = ((const short4*)src)[1];
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) lds_dy[(c8 + j) * KP + k] = v[j];
      (void)dst;
    }
    // Stage x halo row: units of 8 channels x 1 column; 34 cols x 8
    // units = 272 units over 256 threads (one loop iteration + tail).
    for (int u = tid; u < XCOLS * 8; u += 256) {
      const int c8 = (u & 7) * 8;
      const int col = u >> 3;
      const int gx = w0 - 1 + col;
      short v[8] = {};
      if (gx >= 0 && gx < W && ci0 + c8 < Cin) {
        const short* src = (const short*)x +
            (((n * H + gy) * W + gx) * (long)Cin + ci0 + c8);
        *(short4*)&v[0] = ((const short4*)src)[0];
        *(short4*)&v[4] = ((const short4*)src)[1];
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) lds_x[(c8 + j) * KP + col] = v[j];
    }
    __syncthreads();

    const int kf = (lane >> 4) * 8;
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
      const int co_r = wave_m * 32 + mi * 16 + (lane & 15);
      const bf16x8 a = *(const bf16x8*)&lds_dy[co_r * KP + kf];
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        const int ci_r = wave_n * 32 + ni * 16 + (lane & 15);
        const short* xrow = &lds_x[ci_r * KP];
#pragma unroll
        for (int dc = 0; dc < 3; ++dc) {
          // B[k][ci] = x[w0 + k + dc - 1][ci] -> staged col k + dc.
          bf16x8 b;
#pragma unroll
          for (int j = 0; j < 8; ++j)
            ((short*)&b)[j] = xrow[kf + j + dc];
          acc[dc][mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a, b, acc[dc][mi][ni], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // Combine: D fragment row = co = (lane>>4)*4 + r, col = ci = lane&15.
#pragma unroll
  for (int dc = 0; dc < 3; ++dc) {
    const int tap = dr * 3 + dc;
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        const int ci = ci0 + wave_n * 32 + ni * 16 + (lane & 15);
        if (ci >= Cin) continue;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int co = co0 + wave_m * 32 + mi * 16 + (lane >> 4) * 4 + r;
          if (co >= Cout) continue;
          atomicAdd(&dw[((long)tap * Cout + co) * Cin + ci],
                    acc[dc][mi][ni][r]);
        }
      }
    }
  }
}

}  // namespace

extern "C" {

void conv3x3_wrw_launch(const void* x, const void* dy, float* dw, int B,
                        int H, int W, int Cin, int Cout,
                        hipStream_t stream) {
  const int co_t = ceil_div(Cout, CO_T);
  const int ci_t = ceil_div(Cin, CI_T);
  const int spr = ceil_div(W, SW);
  const long total_segs = (long)B * H * spr;
  // Fill 256 CUs (8 XCDs): >= ~1024 workgroups when K allows.
  int split = (int)(2048 / ((long)co_t * ci_t * 3) + 1);
  if (split > total_segs) split = (int)total_segs;
  if (split < 1) split = 1;
  dim3 grid(co_t, ci_t * 3, split);
  hipLaunchKernelGGL(conv3x3_wrw_kernel, grid, dim3(256), 0, stream,
                     (const __hip_bfloat16*)x, (const __hip_bfloat16*)dy,
                     dw, B, H, W, Cin, Cout);
}

}  // extern "C"
