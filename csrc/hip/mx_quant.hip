// MX block-64 activation quantization — bf16 NHWC -> (e4m3 bytes,
// e8m0 scales), one pass.
//
// The torch-side quantize (amax/div/cast as ~6 eager kernels over the
// full activation) costs 2-3x the MX conv itself (r13_conv_ab); this
// kernel does it in ONE bandwidth-bound pass: each thread owns one
// 64-channel block of one pixel: read 64 bf16 (128 B), reduce |max|,
// derive the e8m0 exponent (floor(log2 amax) - 8, matching
// utils/fp8.py mx_quantize), scale by 2^-e via exponent arithmetic and
// convert pairs with v_cvt_pk_fp8_f32.
#include "common.h"

typedef __attribute__((ext_vector_type(2))) float f32x2;

namespace {

__global__ __launch_bounds__(256)
void mx_quant_kernel(const __hip_bfloat16* __restrict__ x,  // [M, C]
                     unsigned char* __restrict__ q,         // [M, C]
                     unsigned char* __restrict__ s,         // [M, C/64]
                     long nblocks, int KS) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= nblocks) return;
  const long m = i / KS;
  const int kb = (int)(i % KS);
  const __hip_bfloat16* src = x + m * (long)KS * 64 + kb * 64;

  float v[64];
  float amax = 0.f;
#pragma unroll
  for (int j = 0; j < 64; ++j) {
    v[j] = __bfloat162float(src[j]);
    amax = fmaxf(amax, fabsf(v[j]));
  }
  amax = fmaxf(amax, 1.1754944e-38f);  // 2^-126 floor (fp8.py clamp)
  int ee;
  frexpf(amax, &ee);                    // amax = f * 2^ee, f in [0.5,1)
  int e = (ee - 1) - 8;                 // floor(log2 amax) - E4M3_EMAX
  if (e < -127) e = -127;
  if (e > 127) e = 127;
  s[i] = (unsigned char)(e + 127);
  const float inv = exp2f((float)-e);
  unsigned int packed[16];
#pragma unroll
  for (int j = 0; j < 32; ++j) {
    float a = v[2 * j] * inv, b = v[2 * j + 1] * inv;
    // clamp to e4m3 range (448) like utils/fp8.py
    a = fminf(fmaxf(a, -448.f), 448.f);
    b = fminf(fmaxf(b, -448.f), 448.f);
    unsigned int w = __builtin_amdgcn_cvt_pk_fp8_f32(a, b, 0, false);
    if (j & 1) {
      packed[j >> 1] |= (w & 0xffffu) << 16;
    } else {
      packed[j >> 1] = w & 0xffffu;
    }
  }
  unsigned int* dst = (unsigned int*)(q + m * (long)KS * 64 + kb * 64);
#pragma unroll
  for (int j = 0; j < 16; ++j) dst[j] = packed[j];
}

}  // namespace

extern "C" {

void mx_quant_launch(const void* x, void* q, void* s, long nblocks,
                     int KS, hipStream_t stream) {
  const int block = 256;
  const long grid = (nblocks + block - 1) / block;
  hipLaunchKernelGGL(mx_quant_kernel, dim3((unsigned)grid), dim3(block),
                     0, stream, (const __hip_bfloat16*)x,
                     (unsigned char*)q, (unsigned char*)s, nblocks, KS);
}

}  // extern "C"
