// K18: one-kernel minibatch gather — the five per-minibatch index_selects
// (states / masks / safe / unsafe / u_qp) into the HIP-graph static input
// buffers become ONE launch. One workgroup per minibatch row.
#include "common.h"

__launch_bounds__(256) __global__
void mb_gather_kernel(const float* __restrict__ states, const bool* __restrict__ masks,
                      const bool* __restrict__ safe, const bool* __restrict__ unsafe,
                      const float* __restrict__ u_qp, const long* __restrict__ idx,
                      float* __restrict__ o_states, bool* __restrict__ o_masks,
                      bool* __restrict__ o_safe, bool* __restrict__ o_unsafe,
                      float* __restrict__ o_uqp, int n_state, int n_mask, int n_flag,
                      int n_uqp) {
  const int b = blockIdx.x;
  const long s = idx[b];
  const int tid = threadIdx.x;
  const float4* src4 = (const float4*)(states + (long)s * n_state);
  float4* dst4 = (float4*)(o_states + (long)b * n_state);
  for (int i = tid; i < n_state / 4; i += 256) dst4[i] = src4[i];
  for (int i = tid; i < n_mask; i += 256)
    o_masks[(long)b * n_mask + i] = masks[(long)s * n_mask + i];
  for (int i = tid; i < n_flag; i += 256) {
    o_safe[(long)b * n_flag + i] = safe[(long)s * n_flag + i];
    o_unsafe[(long)b * n_flag + i] = unsafe[(long)s * n_flag + i];
  }
  for (int i = tid; i < n_uqp; i += 256)
    o_uqp[(long)b * n_uqp + i] = u_qp[(long)s * n_uqp + i];
}
