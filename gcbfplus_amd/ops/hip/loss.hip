// K10: fused GCBF+ loss — all hinge terms + action MSE in one kernel pair
// (reference gcbf_plus.py:364-431; the eager chain is ~160 4.7us kernels).
//
// fwd inputs (flattened over B*N agents):
//   h       = CBF(graph)            h_next = CBF(next graph)
//   h_ng    = CBF_detached(next graph)     (equal in VALUE to h_next)
//   safe/unsafe labels; action & u_qp over (B*N, nu)
// outputs: out[0..9] = total, loss_action, loss_unsafe, loss_safe,
//   loss_h_dot, acc_unsafe, acc_safe, acc_h_dot, unsafe_ratio, cnt pack
// bwd: analytic gradients to h, h_next, h_ng, action.
#include "common.h"

__launch_bounds__(256) __global__
void gcbf_loss_fwd_kernel(const float* __restrict__ h, const float* __restrict__ h_next,
                          const float* __restrict__ h_ng, const float* __restrict__ action,
                          const float* __restrict__ u_qp, const bool* __restrict__ safe,
                          const bool* __restrict__ unsafe, float* __restrict__ out,
                          long n, int nu, float dt, float alpha, float eps,
                          float c_act, float c_unsafe, float c_safe, float c_hdot) {
  // single workgroup: n ~ mb*N (a few thousand)
  __shared__ float red[4][10];
  float s_unsafe = 0, s_safe = 0, s_hdot = 0, s_act = 0;
  float cnt_u = 0, cnt_s = 0, acc_u = 0, acc_s = 0, acc_hd = 0;
  for (long i = threadIdx.x; i < n; i += 256) {
    const float hi = h[i];
    const float hd = (h_next[i] - hi) / dt;
    const float hd_ng = (h_ng[i] - hi) / dt;
    const bool u = unsafe[i], sf = safe[i];
    cnt_u += u;
    cnt_s += sf;
    const float h_u = u ? hi : -2.f * eps;
    s_unsafe += fmaxf(h_u + eps, 0.f);
    acc_u += (u ? hi : 1.f) < 0.f;
    const float h_s = sf ? hi : 2.f * eps;
    s_safe += fmaxf(-h_s + eps, 0.f);
    acc_s += (sf ? hi : -1.f) > 0.f;
    const float val = fmaxf(-hd - alpha * hi + eps, 0.f);
    const float val_ng = fmaxf(-hd_ng - alpha * hi + eps, 0.f);
    s_hdot += (u || sf) ? val : val_ng;
    acc_hd += (hd + alpha * hi) > 0.f;
    float a2 = 0.f;
    for (int c = 0; c < nu; ++c) {
      const float d = action[i * nu + c] - u_qp[i * nu + c];
      a2 += d * d;
    }
    s_act += a2;
  }
  float vals[9] = {s_unsafe, s_safe, s_hdot, s_act, cnt_u, cnt_s, acc_u, acc_s, acc_hd};
#pragma unroll
  for (int v = 0; v < 9; ++v) {
    float x = wave_reduce_sum(vals[v]);
    if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6][v] = x;
    __syncthreads();
    if (threadIdx.x == 0) vals[v] = red[0][v] + red[1][v] + red[2][v] + red[3][v];
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    const float loss_unsafe = vals[0] / (vals[4] + 1e-6f);
    const float loss_safe = vals[1] / (vals[5] + 1e-6f);
    const float loss_hdot = vals[2] / n;
    const float loss_act = vals[3] / n;
    out[0] = c_act * loss_act + c_unsafe * loss_unsafe + c_safe * loss_safe +
             c_hdot * loss_hdot;
    out[1] = loss_act;
    out[2] = loss_unsafe;
    out[3] = loss_safe;
    out[4] = loss_hdot;
    out[5] = (vals[6] + 1e-6f) / (vals[4] + 1e-6f);  // acc_unsafe
    out[6] = (vals[7] + 1e-6f) / (vals[5] + 1e-6f);  // acc_safe
    out[7] = vals[8] / n;                            // acc_h_dot
    out[8] = vals[4] / n;                            // unsafe data ratio
    out[9] = vals[5];                                // cnt_safe (for bwd)
    out[10] = vals[4];                               // cnt_unsafe
  }
}

__launch_bounds__(256) __global__
void gcbf_loss_bwd_kernel(const float* __restrict__ h, const float* __restrict__ h_next,
                          const float* __restrict__ h_ng, const float* __restrict__ action,
                          const float* __restrict__ u_qp, const bool* __restrict__ safe,
                          const bool* __restrict__ unsafe, const float* __restrict__ out,
                          const float* __restrict__ gscale, float* __restrict__ dh,
                          float* __restrict__ dh_next, float* __restrict__ dh_ng,
                          float* __restrict__ daction, long n, int nu, float dt,
                          float alpha, float eps, float c_act, float c_unsafe,
                          float c_safe, float c_hdot) {
  const float g = gscale[0];
  const float cnt_u = out[10], cnt_s = out[9];
  const float wu = g * c_unsafe / (cnt_u + 1e-6f);
  const float ws = g * c_safe / (cnt_s + 1e-6f);
  const float wh = g * c_hdot / n;
  const float wa = g * c_act / n;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    const float hi = h[i];
    const float hd = (h_next[i] - hi) / dt;
    const float hd_ng = (h_ng[i] - hi) / dt;
    const bool u = unsafe[i], sf = safe[i];
    float d_h = 0.f, d_hn = 0.f, d_hng = 0.f;
    if (u && (hi + eps > 0.f)) d_h += wu;
    if (sf && (-hi + eps > 0.f)) d_h -= ws;
    const bool labeled = u || sf;
    if (labeled) {
      if (-hd - alpha * hi + eps > 0.f) {
        d_h += wh * (1.f / dt - alpha);
        d_hn -= wh / dt;
      }
    } else {
      if (-hd_ng - alpha * hi + eps > 0.f) {
        d_h += wh * (-alpha);  // only the alpha*h term sees grad (h detached in hd_ng)
        d_hng -= wh / dt;
      }
    }
    dh[i] = d_h;
    dh_next[i] = d_hn;
    dh_ng[i] = d_hng;
    for (int c = 0; c < nu; ++c)
      daction[i * nu + c] = wa * 2.f * (action[i * nu + c] - u_qp[i * nu + c]);
  }
}
