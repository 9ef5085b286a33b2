// Layout probe for v_mfma_scale_f32_16x16x128_f8f6f4 (gfx950).
//
// Constructs the A/B operands IN REGISTERS per lane under an assumed
// layout (lane l holds row l%16, K-block l/16, 32 contiguous e4m3
// bytes), runs ONE scaled MFMA, and writes each lane's 4 accumulators
// out. The host sets a single A one-hot at (m0,k0) and B at (n0,k0)
// and checks where the 1.0 lands in C — pinning the operand and
// accumulator layouts and the scale-byte OPSEL semantics empirically
// (tests/test_fp8.py::test_gpu_mx_probe_layout).
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) int i32x8;

namespace {

// a_scales / b_scales are FULL per-lane i32s ([64] u32, all 4 bytes
// under the lane's control) so the host can map empirically which
// (source lane, byte) the hardware reads each block's e8m0 from.
__global__ __launch_bounds__(64)
void mx_probe_kernel(const unsigned char* __restrict__ a_bytes,  // [16][128]
                     const unsigned char* __restrict__ b_bytes,  // [16][128]
                     const unsigned int* __restrict__ a_scales,  // [64]
                     const unsigned int* __restrict__ b_scales,  // [64]
                     float* __restrict__ c_out) {                // [64][4]
  const int l = threadIdx.x;
  const int row = l & 15;
  const int kb = l >> 4;
  union {
    i32x8 v;
    unsigned char b[32];
  } a, b;
#pragma unroll
  for (int j = 0; j < 32; ++j) {
    a.b[j] = a_bytes[row * 128 + kb * 32 + j];
    b.b[j] = b_bytes[row * 128 + kb * 32 + j];
  }
  const int sa = (int)a_scales[l];
  const int sb = (int)b_scales[l];
  f32x4 c = {0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
      a.v, b.v, c, 0, 0, 0, sa, 0, sb);
#pragma unroll
  for (int r = 0; r < 4; ++r) c_out[l * 4 + r] = c[r];
}

}  // namespace

extern "C" {

void mx_probe_launch(const void* a, const void* b, const unsigned int* as,
                     const unsigned int* bs, float* c,
                     hipStream_t stream) {
  hipLaunchKernelGGL(mx_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (const unsigned char*)a, (const unsigned char*)b,
                     as, bs, c);
}

}  // extern "C"
