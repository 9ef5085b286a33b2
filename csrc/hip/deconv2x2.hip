// 2x2 stride-2 transposed conv + bias + ReLU — CDNA4 MFMA (NHWC bf16).
//
// The mask-head upsample (reference Conv2DTranspose 2x2/2,
// mrcnn/model.py:~1000 build_fpn_mask_graph). A 2x2/2 deconv has no
// overlapping taps, so it is exactly a GEMM
//     C[M, 4*Cout] = X[M, Cin] x W[Cin, 4*Cout],   M = N*Hin*Win,
// where column (tap*Cout + co), tap = di*2 + dj, is output pixel
// (2i+di, 2j+dj). The kernel reuses the conv3x3_v2 machinery (128x128
// tile, K=32 steps, double-buffered glds with XOR swizzle) with the
// 9-tap halo staging replaced by a plain K-walk and the epilogue store
// scattered to the 2x-upsampled NHWC layout.
//
// Weights are packed [4, Cout, Cin] tap-major (pack_weight_deconv in
// ops/deconv.py), identical addressing to conv3x3's [9, Cout, Cin].
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) short bf16x8;

namespace {

constexpr int BM = 128;
constexpr int BN = 128;
constexpr int BK = 32;                   // shorts (bf16) per k-step
constexpr int ROW_B = BK * 2;            // 64 bytes per LDS row
constexpr int TILE_B = BM * ROW_B;       // 8 KiB per tile

__device__ inline int swz(int row, int chunk) {
  return chunk ^ ((row >> 2) & 3);
}

__global__ __launch_bounds__(256)
void deconv2x2_kernel(const __hip_bfloat16* __restrict__ x,  // [N,Hi,Wi,Cin]
                      const __hip_bfloat16* __restrict__ w,  // [4,Cout,Cin]
                      const float* __restrict__ bias,
                      const __hip_bfloat16* __restrict__ zeros,  // >=16 B
                      __hip_bfloat16* __restrict__ out,  // [N,2Hi,2Wi,Cout]
                      int Bn, int H, int W, int Cin, int Cout,
                      int relu) {
  const long M = (long)Bn * H * W;
  const int Ncols = 4 * Cout;
  const long m0 = (long)blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wave_m = wave >> 1;
  const int wave_n = wave & 1;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  short* const lds = (short*)smem;
  constexpr int TILE_S = TILE_B / 2;
  auto a_off = [](int buf) { return 2 * buf * TILE_S; };
  auto b_off = [](int buf) { return 2 * buf * TILE_S + TILE_S; };

  const int s_row[2] = {wave * 32 + (lane >> 2),
                        wave * 32 + 16 + (lane >> 2)};

  f32x4 acc[4][4] = {};
  const int n_steps = Cin / BK;

  auto stage = [&](int step, int buf) {
    const int k0 = step * BK;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int row = s_row[i];
      const int chunk = swz(row, lane & 3);
      const long m = m0 + row;
      const short* asrc = (m < M)
          ? (const short*)x + m * (long)Cin + k0 + chunk * 8
          : (const short*)zeros;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)asrc,
          (__attribute__((address_space(3))) unsigned int*)
              &lds[a_off(buf) + (wave * 32 + i * 16) * BK],
          16, 0, 0);
      // B column n = tap*Cout + co -> packed w row n (tap-major).
      const int n = n0 + row;
      const short* bsrc = (n < Ncols)
          ? (const short*)w + (long)n * Cin + k0 + chunk * 8
          : (const short*)zeros;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)bsrc,
          (__attribute__((address_space(3))) unsigned int*)
              &lds[b_off(buf) + (wave * 32 + i * 16) * BK],
          16, 0, 0);
    }
  };

  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  for (int t = 0; t < n_steps; ++t) {
    const int cur = t & 1;
    if (t + 1 < n_steps) stage(t + 1, cur ^ 1);

    const int kchunk = (lane >> 4);
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      const int arow = wave_m * 64 + mi * 16 + (lane & 15);
      bf16x8 a = *(const bf16x8*)&lds[
          a_off(cur) + arow * BK + swz(arow, kchunk) * 8];
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const int bcol = wave_n * 64 + ni * 16 + (lane & 15);
        bf16x8 b = *(const bf16x8*)&lds[
            b_off(cur) + bcol * BK + swz(bcol, kchunk) * 8];
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a, b, acc[mi][ni], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  const int Ho = 2 * H, Wo = 2 * W;
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      const int ncol = n0 + wave_n * 64 + ni * 16 + (lane & 15);
      if (ncol >= Ncols) continue;
      const int tap = ncol / Cout;
      const int co = ncol - tap * Cout;
      const int di = tap >> 1, dj = tap & 1;
      const float bv = bias ? bias[co] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long m = m0 + wave_m * 64 + mi * 16 + (lane >> 4) * 4 + r;
        if (m >= M) continue;
        const int j = (int)(m % W);
        const long rest = m / W;
        const int i = (int)(rest % H);
        const long n = rest / H;
        float v = acc[mi][ni][r] + bv;
        if (relu) v = fmaxf(v, 0.f);
        out[(((n * Ho + 2 * i + di) * (long)Wo) + 2 * j + dj) *
                (long)Cout + co] = __float2bfloat16(v);
      }
    }
  }
}

}  // namespace

extern "C" {

void deconv2x2_launch(const void* x, const void* w, const float* bias,
                      const void* zeros, void* out, int B, int H, int W,
                      int Cin, int Cout, bool relu, hipStream_t stream) {
  dim3 grid(ceil_div(B * H * W, BM), ceil_div(4 * Cout, BN));
  const size_t lds = 4 * TILE_B;
  hipLaunchKernelGGL(deconv2x2_kernel, grid, dim3(256), lds, stream,
                     (const __hip_bfloat16*)x, (const __hip_bfloat16*)w,
                     bias, (const __hip_bfloat16*)zeros,
                     (__hip_bfloat16*)out, B, H, W, Cin, Cout,
                     relu ? 1 : 0);
}

}  // extern "C"
