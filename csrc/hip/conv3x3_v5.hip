// 3x3 conv (s1 p1) + bias + ReLU — CDNA4 MFMA, v4: 256x256 tile.
//
// v3 (128x128 tile, 4 waves) tops out ~641-664 TF on the big-map
// 256-ch shapes: per 16-MFMA barrier pair it stages 16 KB of LDS-DMA
// (32 FLOP per staged byte). v4 doubles the tile to 256x256 with 8
// waves (512 threads, 2 per SIMD): 64 FLOP per staged byte, half the
// barrier density (2 barriers per 32 MFMAs), and 2x the B-panel L2
// reuse — the cdna_hip_programming.md §5 "256² template" geometry, but
// keeping v3's GPU-validated 3-buffer counted-vmcnt(4) ring (the
// template's 2-buffer 3-halves-in-flight form would overwrite a buffer
// its partner K-tile still reads when K-tiles alternate buffers).
//
// Geometry: BM=BN=256, BK=32 shorts; wave (wm, wn) = (w>>2, w&3) owns
// a 128x64 output block = acc[8][4] fragments of 16x16. Per K-tile:
// phase A stages the 16 KB A panel of tile t+2 and computes mi 0..3;
// phase B stages the B panel and computes mi 4..7. LDS = 3 bufs x
// (16+16) KB = 96 KB -> one workgroup per CU.
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) short bf16x8;

namespace {

constexpr int BM = 256;
constexpr int BN = 256;
constexpr int BK = 32;                   // shorts per k-step
constexpr int ROW_B = BK * 2;            // 64 B per LDS row
constexpr int PANEL_B = BM * ROW_B;      // 16 KiB per A (or B) panel
constexpr int NBUF = 3;
constexpr int NXCD = 8;

// st_16x32 swizzle (cdna_hip_programming.md §5 template form): XOR
// byte-bit-5 with byte-bit-9. 64-B rows: byte = row*64 + chunk*16, so
// bit9 = (row>>3)&1 and bit5 = chunk bit 1 -> chunk ^= ((row>>3)&1)<<1.
// (v2/v3/v4 use a 2-bit XOR; PMC measured 50% LDS conflict overhead
// there — this variant A/Bs the guide's exact form.)
__device__ inline int swz(int row, int chunk) {
  return chunk ^ (((row >> 3) & 1) << 1);
}

__global__ __launch_bounds__(512, 2)
void conv3x3_v5_kernel(const __hip_bfloat16* __restrict__ x,  // [B,H,W,Cin]
                       const __hip_bfloat16* __restrict__ w,  // [9,Cout,Cin]
                       const float* __restrict__ bias,
                       const __hip_bfloat16* __restrict__ zeros,
                       __hip_bfloat16* __restrict__ out,      // [B,H,W,Cout]
                       int Bn, int H, int W, int Cin, int Cout,
                       int relu) {
  const long M = (long)Bn * H * W;
  const int nwg = gridDim.x;
  int bid = blockIdx.x;
  {  // bijective XCD swizzle (T1)
    const int q = nwg / NXCD, r = nwg % NXCD;
    const int xcd = bid % NXCD, pos = bid / NXCD;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const long m0 = (long)bid * BM;
  const int n0 = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;               // 0..7
  const int wave_m = wave >> 2;            // 0..1 : 128-row half
  const int wave_n = wave & 3;             // 0..3 : 64-col quarter

  extern __shared__ __attribute__((aligned(16))) char smem[];
  short* const lds = (short*)smem;
  constexpr int PANEL_S = PANEL_B / 2;     // shorts per panel
  auto a_off = [](int buf) { return 2 * buf * PANEL_S; };
  auto b_off = [](int buf) { return 2 * buf * PANEL_S + PANEL_S; };

  // Staging rows for this lane: 2 glds per panel per wave, instruction
  // i covers rows [wave*32 + i*16, +16), lane l -> row +(l>>2),
  // chunk l&3 (pre-swizzled source).
  const int s_row[2] = {wave * 32 + (lane >> 2),
                        wave * 32 + 16 + (lane >> 2)};
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

  f32x4 acc[8][4] = {};
  const int n_steps = 9 * (Cin / BK);

  auto stage_a = [&](int step) {
    const int buf = step % NBUF;
    const int tap = step / (Cin / BK);
    const int k0 = (step % (Cin / BK)) * BK;
    const int dy = tap / 3 - 1, dx = tap % 3 - 1;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int row = s_row[i];
      const int chunk = swz(row, lane & 3);
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
    }
  };
  auto stage_b = [&](int step) {
    const int buf = step % NBUF;
    const int tap = step / (Cin / BK);
    const int k0 = (step % (Cin / BK)) * BK;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int row = s_row[i];
      const int chunk = swz(row, lane & 3);
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

  // Prologue: tiles 0 and 1 in flight (4 glds per wave each).
  stage_a(0); stage_b(0);
  if (n_steps > 1) { stage_a(1); stage_b(1); }

  for (int t = 0; t < n_steps; ++t) {
    const int cur = t % NBUF;
    // Tile t's 4 glds (per wave) landed; tile t+1's stay in flight.
    if (t + 1 < n_steps) {
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    // Stage t+2 into the ring slot every wave finished reading at t-1.
    const bool more = (t + 2 < n_steps);
    if (more) stage_a(t + 2);

    const int kchunk = lane >> 4;          // 16-B chunk of the 64-B row
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      const int arow = wave_m * 128 + mi * 16 + (lane & 15);
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
    if (more) stage_b(t + 2);
#pragma unroll
    for (int mi = 4; mi < 8; ++mi) {
      const int arow = wave_m * 128 + mi * 16 + (lane & 15);
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
    // Close the read phase before t+3 may overwrite slot cur.
    __builtin_amdgcn_s_barrier();
  }

#pragma unroll
  for (int mi = 0; mi < 8; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      const int ncol = n0 + wave_n * 64 + ni * 16 + (lane & 15);
      if (ncol >= Cout) continue;
      const float bv = bias ? bias[ncol] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long m = m0 + wave_m * 128 + mi * 16 + (lane >> 4) * 4 + r;
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

void conv3x3_v5_launch(const void* x, const void* w, const float* bias,
                       const void* zeros, void* out, int B, int H, int W,
                       int Cin, int Cout, bool relu, hipStream_t stream) {
  dim3 grid(ceil_div(B * H * W, BM), ceil_div(Cout, BN));
  const size_t lds = 2 * NBUF * PANEL_B;
  hipLaunchKernelGGL(conv3x3_v5_kernel, grid, dim3(512), lds, stream,
                     (const __hip_bfloat16*)x, (const __hip_bfloat16*)w,
                     bias, (const __hip_bfloat16*)zeros,
                     (__hip_bfloat16*)out, B, H, W, Cin, Cout,
                     relu ? 1 : 0);
}

}  // extern "C"
