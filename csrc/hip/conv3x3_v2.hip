// 3x3 conv (s1 p1) + bias + ReLU — CDNA4 MFMA, 128x128 tile + glds.
//
// The tuned variant of csrc/hip/conv3x3.hip for GEMM-heavy shapes,
// following the cdna_hip_programming.md §5 ladder: 128x128 block tile
// (4 waves, 64x64 each), K-step 32, double-buffered LDS filled by
// global_load_lds_dwordx4 (16-B lane-linear DMA), XOR-swizzled on the
// SOURCE address + matching XOR on the ds_read (rule 21) so the
// ds_read_b128 fragment reads are bank-conflict-free. Out-of-image /
// out-of-range lanes DMA from a zero page (glds cannot predicate).
//
// One barrier pair per K-step; the glds for step t+1 issues before the
// MFMAs of step t (the +40% "2-buf overlap" regime).
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) short bf16x8;

namespace {

constexpr int BM = 128;
constexpr int BN = 128;
constexpr int BK = 32;                   // shorts (bf16) per k-step
constexpr int ROW_B = BK * 2;            // 64 bytes per LDS row
constexpr int TILE_B = BM * ROW_B;       // 8 KiB per tile

// Physical 16-B chunk for logical (row, chunk): XOR spreads the fragment
// read's 16-row lane group over 4 bank slots AND 4 row-phases.
__device__ inline int swz(int row, int chunk) {
  return chunk ^ ((row >> 2) & 3);
}

__global__ __launch_bounds__(256)
void conv3x3_v2_kernel(const __hip_bfloat16* __restrict__ x,  // [B,H,W,Cin]
                       const __hip_bfloat16* __restrict__ w,  // [9,Cout,Cin]
                       const float* __restrict__ bias,
                       const __hip_bfloat16* __restrict__ zeros,  // >=16 B
                       __hip_bfloat16* __restrict__ out,      // [B,H,W,Cout]
                       int Bn, int H, int W, int Cin, int Cout,
                       int relu) {
  const long M = (long)Bn * H * W;
  const long m0 = (long)blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wave_m = wave >> 1;   // 0..1 : 64-row half
  const int wave_n = wave & 1;    // 0..1 : 64-col half

  // [2 bufs][A|B], 16-B aligned. One flat array + short offsets (an
  // array of __shared__ pointers trips an addrspacecast static-init
  // error on hipcc; and multiple __shared__ objects de-pipeline glds —
  // §5 'Three .s-level traps').
  extern __shared__ __attribute__((aligned(16))) char smem[];
  short* const lds = (short*)smem;
  constexpr int TILE_S = TILE_B / 2;  // shorts per tile
  auto a_off = [](int buf) { return 2 * buf * TILE_S; };
  auto b_off = [](int buf) { return 2 * buf * TILE_S + TILE_S; };

  // ---- staging geometry: per wave 2 glds instructions per tile;
  // instruction i covers rows [wave*32 + i*16, +16), lane l -> row
  // +(l>>2), chunk l&3 (pre-swizzled on the source).
  const int s_row[2] = {wave * 32 + (lane >> 2),
                        wave * 32 + 16 + (lane >> 2)};
  // This lane's A-tile pixel decomposition for both rows (fixed).
  int apx_x[2], apx_y[2];
  long apx_b[2];
  bool am_ok[2];
  for (int i = 0; i < 2; ++i) {
    long m = m0 + s_row[i];
    am_ok[i] = m < M;
    long mm = am_ok[i] ? m : 0;
    apx_x[i] = (int)(mm % W);
    long rest = mm / W;
    apx_y[i] = (int)(rest % H);
    apx_b[i] = rest / H;
  }

  f32x4 acc[4][4] = {};

  const int n_steps = 9 * (Cin / BK);

  auto stage = [&](int step, int buf) {
    const int tap = step / (Cin / BK);
    const int k0 = (step % (Cin / BK)) * BK;
    const int dy = tap / 3 - 1, dx = tap % 3 - 1;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int row = s_row[i];
      const int chunk = swz(row, lane & 3);
      // A source.
      const int sy = apx_y[i] + dy, sx = apx_x[i] + dx;
      const bool ok = am_ok[i] && sy >= 0 && sy < H && sx >= 0 && sx < W;
      const short* asrc = ok
          ? (const short*)x +
                (((apx_b[i] * H + sy) * W + sx) * (long)Cin + k0 +
                 chunk * 8)
          : (const short*)zeros;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)asrc,
          (__attribute__((address_space(3))) unsigned int*)
              &lds[a_off(buf) + (wave * 32 + i * 16) * BK],
          16, 0, 0);
      // B source: w[tap][n0+row][k0 + chunk*8].
      const int n = n0 + row;
      const short* bsrc = (n < Cout)
          ? (const short*)w + ((long)tap * Cout + n) * (long)Cin + k0 +
                chunk * 8
          : (const short*)zeros;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)bsrc,
          (__attribute__((address_space(3))) unsigned int*)
              &lds[b_off(buf) + (wave * 32 + i * 16) * BK],
          16, 0, 0);
    }
  };

  stage(0, 0);
  // Drain the prologue DMA, then enter the loop.
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  for (int t = 0; t < n_steps; ++t) {
    const int cur = t & 1;
    if (t + 1 < n_steps) stage(t + 1, cur ^ 1);

    const int kf = (lane >> 4) * 8;            // fragment k offset
    const int kchunk = kf / 8;                 // 16-B chunk index 0..3
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
    // Next tile's DMA must have landed before we overwrite/read it.
    __syncthreads();
  }

#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      const int ncol = n0 + wave_n * 64 + ni * 16 + (lane & 15);
      if (ncol >= Cout) continue;
      const float bv = bias ? bias[ncol] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long m = m0 + wave_m * 64 + mi * 16 + (lane >> 4) * 4 + r;
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

void conv3x3_v2_launch(const void* x, const void* w, const float* bias,
                       const void* zeros, void* out, int B, int H, int W,
                       int Cin, int Cout, bool relu, hipStream_t stream) {
  dim3 grid(ceil_div(B * H * W, BM), ceil_div(Cout, BN));
  const size_t lds = 4 * TILE_B;
  hipLaunchKernelGGL(conv3x3_v2_kernel, grid, dim3(256), lds, stream,
                     (const __hip_bfloat16*)x, (const __hip_bfloat16*)w,
                     bias, (const __hip_bfloat16*)zeros,
                     (__hip_bfloat16*)out, B, H, W, Cin, Cout,
                     relu ? 1 : 0);
}

}  // extern "C"
