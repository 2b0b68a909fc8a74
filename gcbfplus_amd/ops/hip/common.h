// Common device helpers for the gfx950 (CDNA4) kernels.
// Wavefront = 64 lanes; MFMA v_mfma_f32_16x16x32_bf16 fragments:
//   A: lane l holds A[row = l&15][k = (l>>4)*8 + j], j = 0..7 (one bf16x8)
//   B: lane l holds B[k = (l>>4)*8 + j][col = l&15]
//   C/D: lane l holds C[row = (l>>4)*4 + r][col = l&15], r = 0..3 (f32x4)
#pragma once
#include <hip/hip_runtime.h>

typedef __bf16 bf16_t;
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef __bf16 bf16x2 __attribute__((ext_vector_type(2)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define WAVE 64

// activation codes (keep in sync with gcbfplus_amd/ops/__init__.py)
#define ACT_NONE 0
#define ACT_RELU 1
#define ACT_TANH 2

__device__ __forceinline__ float apply_act(float x, int act) {
  if (act == ACT_RELU) return x > 0.f ? x : 0.f;
  if (act == ACT_TANH) return tanhf(x);
  return x;
}

// gradient of act given the OUTPUT y = act(z)
__device__ __forceinline__ float act_grad_from_out(float y, int act) {
  if (act == ACT_RELU) return y > 0.f ? 1.f : 0.f;
  if (act == ACT_TANH) return 1.f - y * y;
  return 1.f;
}

__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}
