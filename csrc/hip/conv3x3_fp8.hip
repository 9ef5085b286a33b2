// 3x3 conv (stride 1, pad 1), fp8 e4m3 in / bf16 out — CDNA4 MFMA.
//
// The fp8 COMPUTE path for BASELINE config #5 (weight-only fp8 landed
// first; this kernel runs the MACs at the fp8 MFMA rate and halves the
// activation HBM traffic too). Same implicit-GEMM structure as
// conv3x3.hip v1 — M = B*H*W pixels, N = Cout, K = 9 taps x Cin — but
// operands are OCP e4m3 bytes: v_mfma_f32_16x16x32_fp8_fp8 takes 8
// bytes per lane per fragment (one `long`), fp32 accumulate, and the
// C/D layout is the same as bf16 (dtype-independent on gfx950).
//
// Dequantization is a pure epilogue scale: out = acc * (a_scale *
// w_scale[co]) + bias. a_scale is a 1-element DEVICE tensor (the
// per-tensor activation scale computed on-GPU by the wrapper — no host
// sync); w_scale is the per-output-channel absmax/448 scale from
// utils/fp8.py. Weights are pre-laid-out [9][Cout][Cin] tap-major (same
// pack as the bf16 kernel, bytes instead of shorts).
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;

namespace {

constexpr int BM = 64;
constexpr int BN = 64;
constexpr int BK = 32;      // fp8 channels per k-chunk (one MFMA K)
constexpr int LDS_K = 40;   // byte pitch: 8-B aligned rows, bank-spread

__global__ __launch_bounds__(256, 2)
void conv3x3_fp8_kernel(const unsigned char* __restrict__ x,  // [B,H,W,Cin] e4m3
                        const unsigned char* __restrict__ w,  // [9,Cout,Cin] e4m3
                        const float* __restrict__ bias,       // [Cout]
                        const float* __restrict__ w_scale,    // [Cout]
                        const float* __restrict__ a_scale,    // [1] device
                        __hip_bfloat16* __restrict__ out,     // [B,H,W,Cout]
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

  __shared__ unsigned char a_tile[BM * LDS_K];
  __shared__ unsigned char b_tile[BN * LDS_K];

  f32x4 acc[2][2] = {};

  const int lrow = tid >> 2;        // 0..63
  const int lcol = (tid & 3) * 8;   // 0/8/16/24: 8 bytes per thread

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
        unsigned long long* dst =
            (unsigned long long*)&a_tile[lrow * LDS_K + lcol];
        *dst = in_img
            ? *(const unsigned long long*)(x + src_pix * Cin + k0 + lcol)
            : 0ull;
      }
      {
        const int n = n0 + lrow;
        unsigned long long* dst =
            (unsigned long long*)&b_tile[lrow * LDS_K + lcol];
        *dst = (n < Cout)
            ? *(const unsigned long long*)(
                  w + ((long)tap * Cout + n) * Cin + k0 + lcol)
            : 0ull;
      }
      __syncthreads();

      const int kfrag = (lane >> 4) * 8;
#pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
        const int arow = wave_m * 32 + mi * 16 + (lane & 15);
        const long a = *(const long*)&a_tile[arow * LDS_K + kfrag];
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          const int bcol = wave_n * 32 + ni * 16 + (lane & 15);
          const long b = *(const long*)&b_tile[bcol * LDS_K + kfrag];
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
              a, b, acc[mi][ni], 0, 0, 0);
        }
      }
      __syncthreads();
    }
  }

  const float as = a_scale[0];
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      const int ncol = n0 + wave_n * 32 + ni * 16 + (lane & 15);
      if (ncol >= Cout) continue;
      const float sc = as * w_scale[ncol];
      const float bv = bias ? bias[ncol] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long m = m0 + wave_m * 32 + mi * 16 + (lane >> 4) * 4 + r;
        if (m >= M) continue;
        float v = acc[mi][ni][r] * sc + bv;
        if (relu) v = fmaxf(v, 0.f);
        out[m * Cout + ncol] = __float2bfloat16(v);
      }
    }
  }
}

}  // namespace

extern "C" {

void conv3x3_fp8_launch(const void* x, const void* w, const float* bias,
                        const float* w_scale, const float* a_scale,
                        void* out, int B, int H, int W, int Cin, int Cout,
                        bool relu, hipStream_t stream) {
  dim3 grid(ceil_div(B * H * W, BM), ceil_div(Cout, BN));
  hipLaunchKernelGGL(conv3x3_fp8_kernel, grid, dim3(256), 0, stream,
                     (const unsigned char*)x, (const unsigned char*)w,
                     bias, w_scale, a_scale, (__hip_bfloat16*)out,
                     B, H, W, Cin, Cout, relu ? 1 : 0);
}

}  // extern "C"
