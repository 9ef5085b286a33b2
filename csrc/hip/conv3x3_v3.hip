// 3x3 conv (s1 p1) + bias + ReLU — CDNA4 MFMA, v3: deep glds pipeline.
//
// v2 (conv3x3_v2.hip) drains every DMA at its per-iteration
// __syncthreads (the workgroup release waits vmcnt(0) while glds is in
// flight — cdna_hip_programming.md §5 'Pipelining across barriers').
// v3 keeps ONE tile always in flight: 3 LDS buffer pairs, raw
// s_barrier + COUNTED s_waitcnt vmcnt(4) (4 glds per wave per tile),
// stage(t+2) issued after the barrier so it overwrites only a buffer
// every wave has finished reading. Measured on GEMM this structure is
// +83% over serial staging vs +40% for the v2 form.
//
// Also: XCD-aware bijective blockIdx swizzle (T1) so neighbouring
// output tiles (sharing B-panels) land on the same XCD's L2.
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) short bf16x8;

namespace {

constexpr int BM = 128;
constexpr int BN = 128;
constexpr int BK = 32;                   // shorts per k-step
constexpr int ROW_B = BK * 2;            // 64 B per LDS row
constexpr int TILE_B = BM * ROW_B;       // 8 KiB per tile
constexpr int NBUF = 3;
constexpr int NXCD = 8;

__device__ inline int swz(int row, int chunk) {
  return chunk ^ ((row >> 2) & 3);
}

__global__ __launch_bounds__(256)
void conv3x3_v3_kernel(const __hip_bfloat16* __restrict__ x,  // [B,H,W,Cin]
                       const __hip_bfloat16* __restrict__ w,  // [9,Cout,Cin]
                       const float* __restrict__ bias,
                       const __hip_bfloat16* __restrict__ zeros,
                       __hip_bfloat16* __restrict__ out,      // [B,H,W,Cout]
                       int Bn, int H, int W, int Cin, int Cout,
                       int relu) {
  const long M = (long)Bn * H * W;
  // Bijective XCD swizzle (cdna_hip_programming.md §5: nwg%8!=0-safe
  // form): consecutive swizzled ids stay on one XCD's L2.
  const int nwg = gridDim.x;
  int bid = blockIdx.x;
  {
    const int q = nwg / NXCD, r = nwg % NXCD;
    const int xcd = bid % NXCD, pos = bid / NXCD;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const long m0 = (long)bid * BM;
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

  auto stage = [&](int step) {
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

  // Prologue: tiles 0 and 1 in flight.
  stage(0);
  if (n_steps > 1) stage(1);

  for (int t = 0; t < n_steps; ++t) {
    const int cur = t % NBUF;
    // Wait for tile t's 4 glds; leave tile t+1's (if any) in flight.
    if (t + 1 < n_steps) {
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    // Raw barrier: all waves' tile-t DMA landed; does NOT drain the
    // in-flight tile t+1 (a __syncthreads here would).
    __builtin_amdgcn_s_barrier();
    // Stage t+2 into the buffer every wave finished reading at t-1
    // (safe: issued after the barrier above).
    if (t + 2 < n_steps) stage(t + 2);

    const int kfrag = (lane >> 4) * 8;
    const int kchunk = kfrag / 8;
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
    // Close the read phase before anyone overwrites buf cur (t+3 lands
    // here only after every wave passes this barrier at t+1).
    __builtin_amdgcn_s_barrier();
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

void conv3x3_v3_launch(const void* x, const void* w, const float* bias,
                       const void* zeros, void* out, int B, int H, int W,
                       int Cin, int Cout, bool relu, hipStream_t stream) {
  dim3 grid(ceil_div(B * H * W, BM), ceil_div(Cout, BN));
  const size_t lds = 2 * NBUF * TILE_B;
  hipLaunchKernelGGL(conv3x3_v3_kernel, grid, dim3(256), lds, stream,
                     (const __hip_bfloat16*)x, (const __hip_bfloat16*)w,
                     bias, (const __hip_bfloat16*)zeros,
                     (__hip_bfloat16*)out, B, H, W, Cin, Cout,
                     relu ? 1 : 0);
}

}  // extern "C"
