// K13: fused optimizer path over FLAT parameter/grad/state buffers.
//
//   grad_norm_sq_kernel : ||g||^2 partial reduction (deterministic order)
//   adamw_flat_kernel   : global-norm clip + AdamW/Adam step + finite guard,
//                         all device-side (no host sync, HIP-graph safe).
//
// Semantics mirror the reference's optimizer stack exactly:
//   clip = max_norm / max(max_norm, ||g||)   (trainer/utils.py:66-75)
//   optax.apply_if_finite: if ||g|| is non-finite, skip the whole update
//   (no m/v/t advance)                        (gcbf.py:102,119)
//   adamw(lr, b1=.9, b2=.999, eps=1e-8, wd)   (gcbf_plus.py:109,127)
// The step counter t lives on device and is advanced by the kernel itself
// (block 0 thread 0 via a separate t_next write after all blocks read t —
// we instead pass t as a device scalar updated by a tiny follow-up kernel).
#include "common.h"

__global__ void grad_norm_sq_partial_kernel(const float* __restrict__ g, long n,
                                            float* __restrict__ partial) {
  __shared__ float red[4];
  float acc = 0.f;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    const float v = g[i];
    acc += v * v;
  }
  acc = wave_reduce_sum(acc);
  const int w = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) red[w] = acc;
  __syncthreads();
  if (threadIdx.x == 0) partial[blockIdx.x] = red[0] + red[1] + red[2] + red[3];
}

__global__ void reduce_norm_kernel(const float* __restrict__ partial, int nb,
                                   float* __restrict__ norm_out) {
  // single block; deterministic fixed-order sum
  __shared__ float red[4];
  float acc = 0.f;
  for (int i = threadIdx.x; i < nb; i += 256) acc += partial[i];
  acc = wave_reduce_sum(acc);
  const int w = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) red[w] = acc;
  __syncthreads();
  if (threadIdx.x == 0) norm_out[0] = sqrtf(red[0] + red[1] + red[2] + red[3]);
}

// p, g, m, v: flat f32 buffers of length n. t: device int32 step counter
// (pre-increment semantics: kernel uses t+1). norm: device scalar ||g||.
// pbf: bf16 shadow of p kept in lock-step so fused_linear never re-casts
// weights on the hot path (one cast kernel per Dense per minibatch saved).
__global__ void adamw_flat_kernel(float* __restrict__ p, const float* __restrict__ g,
                                  float* __restrict__ m, float* __restrict__ v,
                                  bf16_t* __restrict__ pbf,
                                  const float* __restrict__ norm, const int* __restrict__ t,
                                  float lr, float b1, float b2, float eps, float wd,
                                  float max_norm, long n) {
  const float nv = norm[0];
  if (!isfinite(nv)) return;  // apply_if_finite: skip everything
  const float clip = max_norm / fmaxf(max_norm, nv);
  const int tt = t[0] + 1;
  const float bc1 = 1.f - __powf(b1, tt);
  const float bc2 = 1.f - __powf(b2, tt);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    const float gi = g[i] * clip;
    const float mi = b1 * m[i] + (1.f - b1) * gi;
    const float vi = b2 * v[i] + (1.f - b2) * gi * gi;
    m[i] = mi;
    v[i] = vi;
    const float upd = (mi / bc1) / (sqrtf(vi / bc2) + eps) + wd * p[i];
    const float pn = p[i] - lr * upd;
    p[i] = pn;
    pbf[i] = (bf16_t)pn;
  }
}

__global__ void advance_step_kernel(int* __restrict__ t, const float* __restrict__ norm) {
  if (threadIdx.x == 0 && blockIdx.x == 0 && isfinite(norm[0])) t[0] += 1;
}
