// Fused FPN lateral + top-down step — CDNA4 (gfx950) MFMA kernel.
//
// Computes P_i = Conv1x1(C_i; W, bias) + NearestUpsample2x(P_{i+1}) in ONE
// kernel (reference: fpn_c*p* lateral convs + UpSampling2D +
// Add, mrcnn/model.py:~1900-1950). The 1x1 conv over NHWC is a GEMM
// [M = B*H*W pixels, N = C_out, K = C_in] and runs on
// v_mfma_f32_16x16x32_bf16 (64-wide wavefronts, LDS-staged A/B tiles);
// the upsample-add and bias ride in the epilogue, saving the two
// elementwise kernels and one full P-map round trip through HBM.
//
// Geometry: 256-thread blocks (4 waves as 2x2), block tile 64x64,
// K-step 32. Per wave: 32x32 output = 2x2 mfma fragments.
// A-fragment (M=16,K=32): lane l holds A[row=l&15][k=(l>>4)*8 + 0..7].
// B-fragment (K=32,N=16): lane l holds B[k=(l>>4)*8 + 0..7][col=l&15].
// C/D: col=lane&15, row=(lane>>4)*4 + reg (cdna_hip_programming.md §3).
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) short bf16x8;

namespace {

constexpr int BM = 64;   // pixel tile
constexpr int BN = 64;   // out-channel tile
constexpr int BK = 32;   // k step
// LDS row stride in shorts; +8 pad breaks the 64-bank power-of-2 stride
// for the 16-B ds_reads (Guideline 4).
constexpr int LDS_K = BK + 8;

__global__ __launch_bounds__(256, 2)
void fpn_fuse_kernel(const __hip_bfloat16* __restrict__ x,   // [B,H,W,Cin]
                     const __hip_bfloat16* __restrict__ w,   // [Cout,Cin]
                     const float* __restrict__ bias,         // [Cout]
                     const __hip_bfloat16* __restrict__ up,  // [B,Hu,Wu,Cout] or null
                     __hip_bfloat16* __restrict__ out,       // [B,H,W,Cout]
                     int Bn, int H, int W, int Cin, int Cout) {
  const long M = (long)Bn * H * W;
  const int m0 = blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;       // 0..3
  const int wave_m = wave >> 1;    // 0..1
  const int wave_n = wave & 1;     // 0..1

  __shared__ short a_tile[BM * LDS_K];
  __shared__ short b_tile[BN * LDS_K];

  f32x4 acc[2][2] = {};

  const int lrow = tid >> 2;          // 0..63: tile row loaded
  const int lcol = (tid & 3) * 8;     // 0,8,16,24: k offset (8 shorts)

  for (int k0 = 0; k0 < Cin; k0 += BK) {
    // Stage A: 64 pixels x 32 channels; each thread one 16-B chunk.
    {
      long m = m0 + lrow;
      const short* src = (const short*)x + m * Cin + k0 + lcol;
      short4* dst = (short4*)&a_tile[lrow * LDS_K + lcol];
      if (m < M) {
        dst[0] = ((const short4*)src)[0];
        dst[1] = ((const short4*)src)[1];
      } else {
        dst[0] = short4{0, 0, 0, 0};
        dst[1] = short4{0, 0, 0, 0};
      }
    }
    // Stage B: 64 out-channels x 32 k.
    {
      int n = n0 + lrow;
      const short* src = (const short*)w + (long)n * Cin + k0 + lcol;
      short4* dst = (short4*)&b_tile[lrow * LDS_K + lcol];
      if (n < Cout) {
        dst[0] = ((const short4*)src)[0];
        dst[1] = ((const short4*)src)[1];
      } else {
        dst[0] = short4{0, 0, 0, 0};
        dst[1] = short4{0, 0, 0, 0};
      }
    }
    __syncthreads();

    const int kfrag = (lane >> 4) * 8;  // this lane's 8-element k offset
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

  // Epilogue: bias + nearest-up2 add, bf16 store.
  const int Hu = H >> 1, Wu = W >> 1;
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
        if (up) {
          const int xw = (int)(m % W);
          const long rest = m / W;
          const int y = (int)(rest % H);
          const long b_img = rest / H;
          const long uidx = (((b_img * Hu) + (y >> 1)) * Wu + (xw >> 1))
                            * Cout + ncol;
          v += __bfloat162float(up[uidx]);
        }
        out[m * Cout + ncol] = __float2bfloat16(v);
      }
    }
  }
}

}  // namespace

extern "C" {

void fpn_fuse_launch(const void* x, const void* w, const float* bias,
                     const void* up, void* out, int B, int H, int W,
                     int Cin, int Cout, hipStream_t stream) {
  dim3 grid(ceil_div(B * H * W, BM), ceil_div(Cout, BN));
  hipLaunchKernelGGL(fpn_fuse_kernel, grid, dim3(256), 0, stream,
                     (const __hip_bfloat16*)x, (const __hip_bfloat16*)w,
                     bias, (const __hip_bfloat16*)up,
                     (__hip_bfloat16*)out, B, H, W, Cin, Cout);
}

}  // extern "C"
