// 3x3 conv (stride 1, pad 1) + bias + optional ReLU — CDNA4 MFMA kernel.
//
// The SURVEY.md N1 centerpiece: an implicit-GEMM NHWC bf16 convolution on
// v_mfma_f32_16x16x32_bf16 with LDS-staged tiles. Frozen BatchNorm is
// folded OUTSIDE the kernel (w_eff = w*scale, bias_eff = shift — the
// reference always runs TRAIN_BN=False, N5), so the epilogue is just
// bias (+ ReLU).
//
// GEMM view: M = B*H*W output pixels, N = Cout, K = 9*Cin, iterated as
// 9 taps x Cin/32 k-chunks. Per 256-thread block (4 waves as 2x2):
// output tile 64(M) x 64(N); per wave 32x32 = 2x2 mfma fragments.
// A-tile staging gathers 64 pixels' shifted input rows (zero-filled at
// image borders) as 16-B chunks; weights are pre-laid-out [3][3][Cin][Cout]
// so the B-tile read is contiguous (tap-major, k-inner, n-outer...).
//
// Correctness-first structure (single-buffered LDS, one barrier pair per
// k-step); the tuning ladder (glds, 128x128 tiles, counted vmcnt, XCD
// swizzle) comes after rocprof evidence.
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) short bf16x8;

namespace {

constexpr int BM = 64;
constexpr int BN = 64;
constexpr int BK = 32;
constexpr int LDS_K = BK + 8;  // 80 B rows: 16-B aligned, bank-spread

// Weights are passed as [3][3][Cin][Cout] (tap-major, Cout innermost)…
// but the B-fragment wants B[k][n] with k = channel chunk. With Cout
// innermost a [BK][BN] LDS tile stages contiguously: rows k (channels),
// cols n. Fragment read then needs B[k=(l>>4)*8+j][col=l&15]: 8
// CONSECUTIVE k for one lane = 8 rows -> strided LDS reads. Instead we
// pre-transpose weights host-side to [3][3][Cout][Cin] so the LDS tile is
// [BN][BK] (n rows, k inner) and each lane reads 8 consecutive k — same
// pattern as fpn_fuse.
__global__ __launch_bounds__(256, 2)
void conv3x3_fwd_kernel(const __hip_bfloat16* __restrict__ x,  // [B,H,W,Cin]
                        const __hip_bfloat16* __restrict__ w,  // [9,Cout,Cin]
                        const float* __restrict__ bias,        // [Cout]
                        __hip_bfloat16* __restrict__ out,      // [B,H,W,Cout]
                        int Bn, int H, int W, int Cin, int Cout,
                        int relu) {
  const long M = (long)Bn * H * W;
  const long m0 = (long)blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wave_m = wave >> 1;
  const int wave_n = wave & 1;

  __shared__ short a_tile[BM * LDS_K];
  __shared__ short b_tile[BN * LDS_K];

  f32x4 acc[2][2] = {};

  const int lrow = tid >> 2;        // 0..63
  const int lcol = (tid & 3) * 8;   // 0/8/16/24

  // Decompose this thread's staged pixel once.
  const long m_load = m0 + lrow;
  int px_x = 0, px_y = 0;
  long px_b = 0;
  bool m_valid = m_load < M;
  if (m_valid) {
    px_x = (int)(m_load % W);
    long rest = m_load / W;
    px_y = (int)(rest % H);
    px_b = rest / H;
  }

  for (int tap = 0; tap < 9; ++tap) {
    const int dy = tap / 3 - 1;
    const int dx = tap % 3 - 1;
    const int sy = px_y + dy;
    const int sx = px_x + dx;
    const bool in_img = m_valid && sy >= 0 && sy < H && sx >= 0 && sx < W;
    const long src_pix = ((px_b * H + sy) * W + sx);
    for (int k0 = 0; k0 < Cin; k0 += BK) {
      {
        short4* dst = (short4*)&a_tile[lrow * LDS_K + lcol];
        if (in_img) {
          const short* src = (const short*)x + src_pix * Cin + k0 + lcol;
          dst[0] = ((const short4*)src)[0];
          dst[1] = ((const short4*)src)[1];
        } else {
          dst[0] = short4{0, 0, 0, 0};
          dst[1] = short4{0, 0, 0, 0};
        }
      }
      {
        const int n = n0 + lrow;
        short4* dst = (short4*)&b_tile[lrow * LDS_K + lcol];
        if (n < Cout) {
          const short* src = (const short*)w +
              ((long)tap * Cout + n) * Cin + k0 + lcol;
          dst[0] = ((const short4*)src)[0];
          dst[1] = ((const short4*)src)[1];
        } else {
          dst[0] = short4{0, 0, 0, 0};
          dst[1] = short4{0, 0, 0, 0};
        }
      }
      __syncthreads();

      const int kfrag = (lane >> 4) * 8;
#pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
        const int arow = wave_m * 32 + mi * 16 + (lane & 15);
        bf16x8 a = *(const bf16x8*)&a_tile[arow * LDS_K + kfrag];
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          const int bcol = wave_n * 32 + ni * 16 + (lane & 15);
          bf16x8 b = *(const bf16x8*)&b_tile[bcol * LDS_K + kfrag];
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a, b, acc[mi][ni], 0, 0, 0);
        }
      }
      __syncthreads();
    }
  }

#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      const int ncol = n0 + wave_n * 32 + ni * 16 + (lane & 15);
      if (ncol >= Cout) continue;
      const float bv = bias ? bias[ncol] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long m = m0 + wave_m * 32 + mi * 16 + (lane >> 4) * 4 + r;
        if (m >= M) continue;
        float v = acc[mi][ni][r] + bv;
        if (relu) v = fmaxf(v, 0.f);
        out[m * Cout + ncol] = __float2bfloat16(v);
      }
    }
  }
}

}  // namespace

extern "C" {

void conv3x3_fwd_launch(const void* x, const void* w, const float* bias,
                        void* out, int B, int H, int W, int Cin, int Cout,
                        bool relu, hipStream_t stream) {
  dim3 grid(ceil_div(B * H * W, BM), ceil_div(Cout, BN));
  hipLaunchKernelGGL(conv3x3_fwd_kernel, grid, dim3(256), 0, stream,
                     (const __hip_bfloat16*)x, (const __hip_bfloat16*)w,
                     bias, (__hip_bfloat16*)out, B, H, W, Cin, Cout,
                     relu ? 1 : 0);
}

}  // extern "C"
