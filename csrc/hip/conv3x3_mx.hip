// 3x3 conv (s1 p1) + bias + ReLU — MX block-scaled fp8 MFMA (gfx950).
//
// The 2x-rate fp8 path (NOTES_ROUND2 item 10): non-block-scaled fp8
// MFMA (16x16x32) runs at the BF16 rate on gfx950, so conv3x3_fp8.hip
// can never beat bf16 on compute; the MX instruction
// v_mfma_scale_f32_16x16x128_f8f6f4 runs K=128 per issue (~5 PF dense,
// 4647 TF measured on MX GEMM) with block-scaled e8m0 exponents.
//
// Layout (pinned EMPIRICALLY by tests/test_fp8.py probes on MI355X,
// gpurun_out/r10_probe.log): operand lane l holds row l%16, K-block
// l/16 (32 consecutive e4m3 bytes); C fragment is the usual 16x16
// layout (c[r]: row (l/16)*4+r, col l%16). The SCALE operand at
// OPSEL=(0,0) is read from byte 0 of lane `row` for k in [0,64) and
// byte 0 of lane `16+row` for k in [64,128) — ONE e8m0 per 64
// consecutive k (not per 32); quantization uses block=64 to match
// (utils/fp8.py mx_quantize(block=64), mx_scale_src).
//
// Structure: conv3x3_v2's 128x128-tile double-buffered glds skeleton
// with BK=128 fp8 bytes per K-step; A/B data staged via width-16 glds,
// per-row scale bytes via width-2 glds.
//
// Inputs: xq [B,H,W,Cin] e4m3 bytes (channels_last), xs [B,H,W,Cin/64]
// e8m0 bytes; wq [9,Cout,Cin] e4m3 tap-major, ws [9,Cout,Cin/64] e8m0.
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) int i32x8;
typedef __attribute__((ext_vector_type(4))) int i32x4;

namespace {

constexpr int BM = 128;
constexpr int BN = 128;
constexpr int BK = 128;                  // fp8 bytes per k-step
constexpr int ROW_B = BK;                // 128 B per LDS row
constexpr int TILE_B = BM * ROW_B;       // 16 KiB per tile
constexpr int SCL_B = BM * 2;            // 2 block-64 scales per row

__global__ __launch_bounds__(256)
void conv3x3_mx_kernel(const unsigned char* __restrict__ xq,
                       const unsigned char* __restrict__ xs,
                       const unsigned char* __restrict__ wq,
                       const unsigned char* __restrict__ ws,
                       const float* __restrict__ bias,
                       const unsigned char* __restrict__ zeros,  // >=64 B
                       __hip_bfloat16* __restrict__ out,  // [B,H,W,Cout]
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
  const int KS = Cin / 64;               // block-64 scale bytes/pixel

  extern __shared__ __attribute__((aligned(16))) char smem[];
  // [buf][A 16K | B 16K | As 512 | Bs 512]
  auto a_off = [](int buf) { return buf * (2 * TILE_B + 2 * SCL_B); };
  auto b_off = [&](int buf) { return a_off(buf) + TILE_B; };
  auto as_off = [&](int buf) { return a_off(buf) + 2 * TILE_B; };
  auto bs_off = [&](int buf) { return a_off(buf) + 2 * TILE_B + SCL_B; };

  // Data staging: 4 glds per matrix per wave; instr i covers rows
  // [wave*32 + i*8, +8), lane l -> row +(l>>3), 16-B chunk l&7.
  const int d_row = (lane >> 3);         // 0..7 within the instr block
  const int d_chunk = lane & 7;
  // This lane's A pixels for the 4 data instrs (rows wave*32+i*8+d_row).
  int apx_x[4], apx_y[4];
  long apx_b[4];
  bool am_ok[4];
  for (int i = 0; i < 4; ++i) {
    long m = m0 + wave * 32 + i * 8 + d_row;
    am_ok[i] = m < M;
    long mm = am_ok[i] ? m : 0;
    apx_x[i] = (int)(mm % W);
    long rest = mm / W;
    apx_y[i] = (int)(rest % H);
    apx_b[i] = rest / H;
  }
  // Scale staging (width-2 glds): wave 0 stages A scales, wave 1 B
  // scales; 2 instrs of 64 rows each. Lane l -> row l.
  int spx_x[2], spx_y[2];
  long spx_b[2];
  bool sm_ok[2];
  for (int i = 0; i < 2; ++i) {
    long m = m0 + i * 64 + lane;
    sm_ok[i] = m < M;
    long mm = sm_ok[i] ? m : 0;
    spx_x[i] = (int)(mm % W);
    long rest = mm / W;
    spx_y[i] = (int)(rest % H);
    spx_b[i] = rest / H;
  }

  f32x4 acc[4][4] = {};
  const int n_steps = 9 * (Cin / BK);

  auto stage = [&](int step, int buf) {
    const int tap = step / (Cin / BK);
    const int k0 = (step % (Cin / BK)) * BK;
    const int dy = tap / 3 - 1, dx = tap % 3 - 1;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int row = wave * 32 + i * 8 + d_row;
      const int sy = apx_y[i] + dy, sx = apx_x[i] + dx;
      const bool ok = am_ok[i] && sy >= 0 && sy < H && sx >= 0 && sx < W;
      const unsigned char* asrc = ok
          ? xq + ((apx_b[i] * H + sy) * (long)W + sx) * Cin + k0 +
                d_chunk * 16
          : zeros;
      // LDS dest must be WAVE-UNIFORM: the DMA writes lane-linear
      // (dest + lane*16), which lands lane l at row +(l>>3),
      // chunk l&7 — exactly the per-lane source coordinates above.
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)asrc,
          (__attribute__((address_space(3))) unsigned int*)
              &smem[a_off(buf) + (wave * 32 + i * 8) * ROW_B],
          16, 0, 0);
      const int n = n0 + row;
      const unsigned char* bsrc = (n < Cout)
          ? wq + ((long)tap * Cout + n) * Cin + k0 + d_chunk * 16
          : zeros;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)bsrc,
          (__attribute__((address_space(3))) unsigned int*)
              &smem[b_off(buf) + (wave * 32 + i * 8) * ROW_B],
          16, 0, 0);
    }
    // Scales: plain global loads + ds_write (tiny traffic; the
    // per-step __syncthreads drains lgkm+vmcnt, so no glds discipline
    // needed). Threads 0..127: A rows; 128..255: B rows.
    {
      const int r = tid & 127;
      if (tid < 128) {
        const int i = r >> 6;
        const int sy = spx_y[i] + dy, sx = spx_x[i] + dx;
        const bool ok =
            sm_ok[i] && sy >= 0 && sy < H && sx >= 0 && sx < W;
        const unsigned char* src = ok
            ? xs + ((spx_b[i] * H + sy) * (long)W + sx) * KS + k0 / 64
            : zeros;
        *(unsigned short*)&smem[as_off(buf) + r * 2] =
            *(const unsigned short*)src;
      } else {
        const int n = n0 + r;
        const unsigned char* src = (n < Cout)
            ? ws + ((long)tap * Cout + n) * KS + k0 / 64
            : zeros;
        *(unsigned short*)&smem[bs_off(buf) + r * 2] =
            *(const unsigned short*)src;
      }
    }
  };

  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  const int frag_row = lane & 15;        // row/col within the fragment
  const int frag_kb = lane >> 4;         // K-block 0..3 (32 B each)

  for (int t = 0; t < n_steps; ++t) {
    const int cur = t & 1;
    if (t + 1 < n_steps) stage(t + 1, cur ^ 1);

    // Scale registers: the instruction reads byte 0 of lanes 0..31
    // only — lane r carries row r's k[0,64) scale, lane 16+r row r's
    // k[64,128) scale (probe-measured; lanes 32..63 ignored). So each
    // lane loads the scale for the (row, half) it CARRIES, not its own
    // fragment's.
    const int s_carry_row = lane & 15;
    const int s_carry_half = (lane >> 4) & 1;
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      const int arow = wave_m * 64 + mi * 16 + frag_row;
      const char* ap = &smem[a_off(cur) + arow * ROW_B + frag_kb * 32];
      i32x4 alo = *(const i32x4*)ap;
      i32x4 ahi = *(const i32x4*)(ap + 16);
      i32x8 a = {alo[0], alo[1], alo[2], alo[3],
                 ahi[0], ahi[1], ahi[2], ahi[3]};
      const int sa_row = wave_m * 64 + mi * 16 + s_carry_row;
      const int sa = (unsigned char)
          smem[as_off(cur) + sa_row * 2 + s_carry_half];
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const int bcol = wave_n * 64 + ni * 16 + frag_row;
        const char* bp = &smem[b_off(cur) + bcol * ROW_B + frag_kb * 32];
        i32x4 blo = *(const i32x4*)bp;
        i32x4 bhi = *(const i32x4*)(bp + 16);
        i32x8 b = {blo[0], blo[1], blo[2], blo[3],
                   bhi[0], bhi[1], bhi[2], bhi[3]};
        const int sb_col = wave_n * 64 + ni * 16 + s_carry_row;
        const int sb = (unsigned char)
            smem[bs_off(cur) + sb_col * 2 + s_carry_half];
        acc[mi][ni] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
            a, b, acc[mi][ni], 0 /*cbsz: e4m3*/, 0 /*blgp: e4m3*/,
            0 /*OpselA*/, sa, 0 /*OpselB*/, sb);
      }
    }
    __syncthreads();
  }

#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      const int ncol = n0 + wave_n * 64 + ni * 16 + frag_row;
      if (ncol >= Cout) continue;
      const float bv = bias ? bias[ncol] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long m = m0 + wave_m * 64 + mi * 16 + frag_kb * 4 + r;
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

void conv3x3_mx_launch(const void* xq, const void* xs, const void* wq,
                       const void* ws, const float* bias,
                       const void* zeros, void* out, int B, int H, int W,
                       int Cin, int Cout, bool relu, hipStream_t stream) {
  dim3 grid(ceil_div(B * H * W, BM), ceil_div(Cout, BN));
  const size_t lds = 2 * (2 * TILE_B + 2 * SCL_B);
  hipLaunchKernelGGL(conv3x3_mx_kernel, grid, dim3(256), lds, stream,
                     (const unsigned char*)xq, (const unsigned char*)xs,
                     (const unsigned char*)wq, (const unsigned char*)ws,
                     bias, (const unsigned char*)zeros,
                     (__hip_bfloat16*)out, B, H, W, Cin, Cout,
                     relu ? 1 : 0);
}

}  // extern "C"
