// 3x3 conv s1p1 + bias/ReLU — SPLIT-K variant for fill-bound shapes.
//
// The one conv shape where the hand kernels still lose to MIOpen is
// (2,512,32²,512): M = B*H*W = 2048 rows. A 64x64-tile grid is only
// 32x8 = 256 workgroups — 1 per CU, half the chip's block slots empty,
// and each block serializes K = 9*512 = 4608 (144 BK-chunks). This
// variant adds a grid.z split over the 144 (tap, k-chunk) reduction
// steps: z takes chunks c ≡ z (mod SPLIT) and accumulates fp32
// partials into its own workspace slab [z, M, N] (NO atomics — the
// reduce epilogue sums slabs in a fixed order, so the result is
// deterministic). SPLIT=8 → 2048 blocks.
//
// The tile/staging/fragment math is byte-identical to conv3x3.hip v1
// (HW-validated); only the chunk decomposition, the z-strided loop and
// the fp32 workspace epilogue are new — those are what
// tests/test_splitk_sim.py simulates.
//
// STATUS: compiled for gfx950, CPU-sim-validated; NOT yet executed on
// hardware (this round's GPU budget closed before it landed) — gated
// behind MASKRCNN_HIP_CONV_SPLITK=1.
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) short bf16x8;

namespace {

constexpr int BM = 64;
constexpr int BN = 64;
constexpr int BK = 32;
constexpr int LDS_K = BK + 8;

__global__ __launch_bounds__(256, 2)
void conv3x3_splitk_kernel(const __hip_bfloat16* __restrict__ x,
                           const __hip_bfloat16* __restrict__ w,  // [9,Cout,Cin]
                           float* __restrict__ ws,  // [SPLIT, M, Cout]
                           int Bn, int H, int W, int Cin, int Cout,
                           int split) {
  const long M = (long)Bn * H * W;
  const long m0 = (long)blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const int z = blockIdx.z;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wave_m = wave >> 1;
  const int wave_n = wave & 1;

  __shared__ short a_tile[BM * LDS_K];
  __shared__ short b_tile[BN * LDS_K];

  f32x4 acc[2][2] = {};

  const int lrow = tid >> 2;
  const int lcol = (tid & 3) * 8;

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

  const int KC = Cin / BK;          // k-chunks per tap (Cin % 32 == 0)
  const int CHUNKS = 9 * KC;
  for (int c = z; c < CHUNKS; c += split) {
    const int tap = c / KC;
    const int k0 = (c % KC) * BK;
    const int dy = tap / 3 - 1;
    const int dx = tap % 3 - 1;
    const int sy = px_y + dy;
    const int sx = px_x + dx;
    const bool in_img = m_valid && sy >= 0 && sy < H && sx >= 0 && sx < W;
    const long src_pix = ((px_b * H + sy) * W + sx);
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

  // fp32 partials into this z's slab (no atomics).
  float* slab = ws + (long)z * M * Cout;
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      const int ncol = n0 + wave_n * 32 + ni * 16 + (lane & 15);
      if (ncol >= Cout) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long m = m0 + wave_m * 32 + mi * 16 + (lane >> 4) * 4 + r;
        if (m >= M) continue;
        slab[m * Cout + ncol] = acc[mi][ni][r];
      }
    }
  }
}

// Reduce slabs in fixed z-order + bias + ReLU -> bf16. One thread per
// output element (M*Cout = 1M at the target shape: trivially parallel).
__global__ __launch_bounds__(256)
void splitk_reduce_kernel(const float* __restrict__ ws,
                          const float* __restrict__ bias,
                          __hip_bfloat16* __restrict__ out,
                          long MN, int Cout, int split, int relu) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= MN) return;
  float v = 0.f;
  for (int z = 0; z < split; ++z) v += ws[(long)z * MN + i];
  if (bias) v += bias[i % Cout];
  if (relu) v = fmaxf(v, 0.f);
  out[i] = __float2bfloat16(v);
}

}  // namespace

extern "C" {

void conv3x3_splitk_launch(const void* x, const void* w,
                           const float* bias, void* workspace, void* out,
                           int B, int H, int W, int Cin, int Cout,
                           int split, bool relu, hipStream_t stream) {
  const long M = (long)B * H * W;
  dim3 grid((unsigned)((M + BM - 1) / BM), ceil_div(Cout, BN), split);
  hipLaunchKernelGGL(conv3x3_splitk_kernel, grid, dim3(256), 0, stream,
                     (const __hip_bfloat16*)x, (const __hip_bfloat16*)w,
                     (float*)workspace, B, H, W, Cin, Cout, split);
  const long MN = M * Cout;
  hipLaunchKernelGGL(splitk_reduce_kernel,
                     dim3((unsigned)((MN + 255) / 256)), dim3(256), 0,
                     stream, (const float*)workspace, bias,
                     (__hip_bfloat16*)out, MN, Cout, split, relu ? 1 : 0);
}

}  // extern "C"
