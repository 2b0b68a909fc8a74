// Fused GCBF+ minibatch prologue for the DoubleIntegrator family:
//   u_ref (clipped-error LQR)  ->  action = clamp(2*actor_raw + u_ref)
//   next_agent = clip_state(euler(agent, action))
//   big_states = [ states ; next_states ]   (the 2B-batched CBF input)
// Replaces ~20 eager kernels per minibatch. Backward chains to actor_raw
// only (mb states are leaves): dactor = 2 * m_act o (daction +
// (dt/m) * m_vel o dnext_vel).
#include "common.h"

__launch_bounds__(256) __global__
void di_loss_prep_fwd_kernel(const float* __restrict__ states,  // (B,V,4)
                             const float* __restrict__ raw,     // (B,N,2)
                             const float* __restrict__ Kmat,    // (2,4)
                             float* __restrict__ action,        // (B,N,2)
                             float* __restrict__ big,           // (2B,V,4)
                             int B, int V, int N, float dt, float inv_m,
                             float comm, float vmax) {
  const long rows = (long)2 * B * V;
  for (long row = (long)blockIdx.x * blockDim.x + threadIdx.x; row < rows;
       row += (long)gridDim.x * blockDim.x) {
    const int v = row % V;
    const long bb = row / V;
    const int half = bb / B;     // 0: current, 1: next
    const int b = bb % B;
    const float* src = states + ((long)b * V + v) * 4;
    float* dst = big + row * 4;
    if (half == 0 || v >= N) {
      *(float4*)dst = *(const float4*)src;
      continue;
    }
    // next agent state for agent i = v
    const int i = v;
    const float* gl = states + ((long)b * V + N + i) * 4;
    float err[4], nrm = 0.f;
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      err[s] = gl[s] - src[s];
      nrm += err[s] * err[s];
    }
    nrm = fmaxf(sqrtf(nrm), 1e-9f);
    float uref[2];
#pragma unroll
    for (int u = 0; u < 2; ++u) {
      const int s0 = u * 4;
      float e2[4];
#pragma unroll
      for (int s = 0; s < 4; ++s) {
        const float emax = fabsf(err[s] / nrm * comm);
        e2[s] = fminf(fmaxf(err[s], -emax), emax);
      }
      float acc = 0.f;
#pragma unroll
      for (int s = 0; s < 4; ++s) acc += Kmat[s0 + s] * e2[s];
      uref[u] = fminf(fmaxf(acc, -1.f), 1.f);
    }
    float act[2];
#pragma unroll
    for (int u = 0; u < 2; ++u) {
      act[u] = fminf(fmaxf(2.f * raw[((long)b * N + i) * 2 + u] + uref[u], -1.f), 1.f);
      action[((long)b * N + i) * 2 + u] = act[u];
    }
    dst[0] = src[0] + src[2] * dt;
    dst[1] = src[1] + src[3] * dt;
    dst[2] = fminf(fmaxf(src[2] + act[0] * inv_m * dt, -vmax), vmax);
    dst[3] = fminf(fmaxf(src[3] + act[1] * inv_m * dt, -vmax), vmax);
  }
}

__launch_bounds__(256) __global__
void di_loss_prep_bwd_kernel(const float* __restrict__ states, const float* __restrict__ raw,
                             const float* __restrict__ Kmat, const float* __restrict__ action,
                             const float* __restrict__ daction,  // (B,N,2)
                             const float* __restrict__ dbig,     // (2B,V,4)
                             float* __restrict__ draw,           // (B,N,2)
                             int B, int V, int N, float dt, float inv_m, float comm,
                             float vmax) {
  const long total = (long)B * N;
  for (long it = (long)blockIdx.x * blockDim.x + threadIdx.x; it < total;
       it += (long)gridDim.x * blockDim.x) {
    const int i = it % N;
    const int b = it / N;
    const float* src = states + ((long)b * V + i) * 4;
    const float* gl = states + ((long)b * V + N + i) * 4;
    const float* dnx = dbig + (((long)(B + b)) * V + i) * 4;
    // recompute u_ref for the pre-clamp action (clamp-mask needs it)
    float err[4], nrm = 0.f;
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      err[s] = gl[s] - src[s];
      nrm += err[s] * err[s];
    }
    nrm = fmaxf(sqrtf(nrm), 1e-9f);
    float e2[4];
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      const float emax = fabsf(err[s] / nrm * comm);
      e2[s] = fminf(fmaxf(err[s], -emax), emax);
    }
#pragma unroll
    for (int u = 0; u < 2; ++u) {
      float uref = 0.f;
#pragma unroll
      for (int s = 0; s < 4; ++s) uref += Kmat[u * 4 + s] * e2[s];
      uref = fminf(fmaxf(uref, -1.f), 1.f);
      const float pre = 2.f * raw[it * 2 + u] + uref;
      const float m_act = (pre >= -1.f && pre <= 1.f) ? 1.f : 0.f;
      const float a = action[it * 2 + u];
      const float vel_pre = src[2 + u] + a * inv_m * dt;
      const float m_vel = (fabsf(vel_pre) <= vmax) ? 1.f : 0.f;
      const float dact = daction[it * 2 + u] + dnx[2 + u] * m_vel * inv_m * dt;
      draw[it * 2 + u] = 2.f * m_act * dact;
    }
  }
}
