// K2/K15: fused first-layer GNN input builder.
//
// fwd: states (B,V,S) f32 -> X (B*N*D, KP) bf16 where each slot row is
//   [ clip(recv - send) (E=S) | sender one-hot (3) | recv one-hot (3) | 0pad ]
// with the position clip of reference add_edge_feats
// (double_integrator.py:275-286): p' = p * comm/max(|p|, comm) via
// n = sqrt(1e-6 + |p|^2). KP is padded to a multiple of 32 so the MFMA GEMM
// consumes it directly. Replaces ~10 eager elementwise/gather/cat/cast
// kernels on the minibatch critical path.
//
// bwd: dX -> dstates via the clip-jacobian vjp, ATOMIC-FREE: each node
// gathers its fixed contribution set (receiver side: own row; sender side:
// slots that point at it) in a deterministic order.
#include "common.h"

// slot layout: d in [0,N) sender=agent d | d==N sender=goal i | else lidar
__device__ __forceinline__ int sender_node(int i, int d, int N, int R) {
  if (d < N) return d;
  if (d == N) return N + i;
  return 2 * N + i * R + (d - N - 1);
}

// edge-state transform: mode 0 = identity; mode 1 = DubinsCar
// [x, y, v cos(theta), v sin(theta)] (reference dubins_car.py:263-275)
__device__ __forceinline__ void edge_state_of(const float* st, float* es, int S, int mode) {
  if (mode == 1) {
    es[0] = st[0];
    es[1] = st[1];
    es[2] = st[3] * __cosf(st[2]);
    es[3] = st[3] * __sinf(st[2]);
  } else {
    for (int s = 0; s < S; ++s) es[s] = st[s];
  }
}

__launch_bounds__(256) __global__
void edge_msg_in_fwd_kernel(const float* __restrict__ states, bf16_t* __restrict__ X,
                            int B, int N, int R, int S, int pdim, int KP, float comm,
                            int mode) {
  const int D = N + 1 + R;
  const int V = 2 * N + N * R;
  const long total = (long)B * N * D;
  for (long row = (long)blockIdx.x * blockDim.x + threadIdx.x; row < total;
       row += (long)gridDim.x * blockDim.x) {
    const int d = row % D;
    const int i = (row / D) % N;
    const int b = row / ((long)N * D);
    const float* recv = states + ((long)b * V + i) * S;
    const float* send = states + ((long)b * V + sender_node(i, d, N, R)) * S;
    bf16_t out[64];
    float p2 = 1e-6f;
    float e[16], er[16], es_[16];
    edge_state_of(recv, er, S, mode);
    edge_state_of(send, es_, S, mode);
    for (int s = 0; s < S; ++s) {
      e[s] = er[s] - es_[s];
      if (s < pdim) p2 += e[s] * e[s];
    }
    const float n = sqrtf(p2);
    const float coef = (n > comm) ? comm / n : 1.f;
    for (int s = 0; s < S; ++s) out[s] = (bf16_t)(s < pdim ? e[s] * coef : e[s]);
    // sender one-hot: agent 001, goal 010, obs 100 (double_integrator.py:288-295)
    const int stype = (d < N) ? 0 : (d == N ? 1 : 2);
    out[S + 0] = (bf16_t)(stype == 2 ? 1.f : 0.f);
    out[S + 1] = (bf16_t)(stype == 1 ? 1.f : 0.f);
    out[S + 2] = (bf16_t)(stype == 0 ? 1.f : 0.f);
    out[S + 3] = (bf16_t)0.f;  // recv one-hot: always agent = 001
    out[S + 4] = (bf16_t)0.f;
    out[S + 5] = (bf16_t)1.f;
    for (int s = S + 6; s < KP; ++s) out[s] = (bf16_t)0.f;
    bf16_t* dst = X + row * KP;
    for (int s = 0; s < KP; s += 8) *(bf16x8*)(dst + s) = *(bf16x8*)(out + s);
  }
}

// S=4 / mode 0 / pdim 2 / KP 32 specialization (DoubleIntegrator, the
// benchmark env): float4 state loads instead of 8 scalar dwords per node —
// the generic kernel's runtime-S copy loops serialize into scalar loads
// that dominate at ~3 waves/SIMD occupancy.
__launch_bounds__(256) __global__
void edge_msg_in_fwd_s4_kernel(const float* __restrict__ states, bf16_t* __restrict__ X,
                               int B, int N, int R, float comm) {
  const int D = N + 1 + R;
  const int V = 2 * N + N * R;
  const long total = (long)B * N * D;
  for (long row = (long)blockIdx.x * blockDim.x + threadIdx.x; row < total;
       row += (long)gridDim.x * blockDim.x) {
    const int d = row % D;
    const int i = (row / D) % N;
    const int b = row / ((long)N * D);
    const float4 rv = *(const float4*)(states + ((long)b * V + i) * 4);
    const float4 sv = *(const float4*)(states + ((long)b * V + sender_node(i, d, N, R)) * 4);
    const float e0 = rv.x - sv.x, e1 = rv.y - sv.y;
    const float e2 = rv.z - sv.z, e3 = rv.w - sv.w;
    const float n = sqrtf(1e-6f + e0 * e0 + e1 * e1);
    const float coef = (n > comm) ? comm / n : 1.f;
    const int stype = (d < N) ? 0 : (d == N ? 1 : 2);
    bf16_t out[16];
    out[0] = (bf16_t)(e0 * coef);
    out[1] = (bf16_t)(e1 * coef);
    out[2] = (bf16_t)e2;
    out[3] = (bf16_t)e3;
    out[4] = (bf16_t)(stype == 2 ? 1.f : 0.f);
    out[5] = (bf16_t)(stype == 1 ? 1.f : 0.f);
    out[6] = (bf16_t)(stype == 0 ? 1.f : 0.f);
    out[7] = (bf16_t)0.f;
    out[8] = (bf16_t)0.f;
    out[9] = (bf16_t)1.f;
#pragma unroll
    for (int s = 10; s < 16; ++s) out[s] = (bf16_t)0.f;
    bf16_t* dst = X + row * 32;
    *(bf16x8*)dst = *(bf16x8*)out;
    *(bf16x8*)(dst + 8) = *(bf16x8*)(out + 8);
    const bf16x8 z8 = {};
    *(bf16x8*)(dst + 16) = z8;
    *(bf16x8*)(dst + 24) = z8;
  }
}

// S=4 / mode 0 / pdim 2 / KP 32 backward specialization: float4 state loads,
// one bf16x8 load per dX row (only the first 4 lanes carry gradient — the
// one-hot / pad columns are constants).
__device__ __forceinline__ void slot_vjp_s4(const float4 rv, const float4 sv,
                                            const bf16_t* __restrict__ dx,
                                            float* __restrict__ acc, float sign,
                                            float comm) {
  const bf16x8 gv = *(const bf16x8*)dx;
  const float g0 = (float)gv[0], g1 = (float)gv[1];
  const float g2 = (float)gv[2], g3 = (float)gv[3];
  const float e0 = rv.x - sv.x, e1 = rv.y - sv.y;
  const float n = sqrtf(1e-6f + e0 * e0 + e1 * e1);
  if (n > comm) {
    const float inv_n = 1.f / n;
    const float gdotp = g0 * e0 + g1 * e1;
    const float c3 = gdotp * inv_n * inv_n * inv_n;
    acc[0] += sign * comm * (g0 * inv_n - e0 * c3);
    acc[1] += sign * comm * (g1 * inv_n - e1 * c3);
  } else {
    acc[0] += sign * g0;
    acc[1] += sign * g1;
  }
  acc[2] += sign * g2;
  acc[3] += sign * g3;
}

__launch_bounds__(256) __global__
void edge_msg_in_bwd_s4_kernel(const float* __restrict__ states,
                               const bf16_t* __restrict__ dX,
                               float* __restrict__ dstates, int B, int N, int R,
                               float comm) {
  const int D = N + 1 + R;
  const int V = 2 * N + N * R;
  const long total = (long)B * V;
  for (long node = (long)blockIdx.x * blockDim.x + threadIdx.x; node < total;
       node += (long)gridDim.x * blockDim.x) {
    const int v = node % V;
    const int b = node / V;
    const float* st = states + (long)b * V * 4;
    const bf16_t* dxb = dX + (long)b * N * D * 32;
    const float4 own = *(const float4*)(st + (long)v * 4);
    float acc[4] = {0.f, 0.f, 0.f, 0.f};
    if (v < N) {
      const int j = v;
      for (int d = 0; d < D; ++d) {
        const float4 sv = *(const float4*)(st + (long)sender_node(j, d, N, R) * 4);
        slot_vjp_s4(own, sv, dxb + ((long)j * D + d) * 32, acc, 1.f, comm);
      }
      for (int i = 0; i < N; ++i) {
        const float4 rv = *(const float4*)(st + (long)i * 4);
        slot_vjp_s4(rv, own, dxb + ((long)i * D + j) * 32, acc, -1.f, comm);
      }
    } else if (v < 2 * N) {
      const int j = v - N;
      const float4 rv = *(const float4*)(st + (long)j * 4);
      slot_vjp_s4(rv, own, dxb + ((long)j * D + N) * 32, acc, -1.f, comm);
    } else {
      const int h = v - 2 * N;
      const int j = h / R;
      const int r = h % R;
      const float4 rv = *(const float4*)(st + (long)j * 4);
      slot_vjp_s4(rv, own, dxb + ((long)j * D + N + 1 + r) * 32, acc, -1.f, comm);
    }
    *(float4*)(dstates + ((long)b * V + v) * 4) =
        float4{acc[0], acc[1], acc[2], acc[3]};
  }
}

// vjp of one slot's edge features wrt the raw diff v (recomputed forward)
__device__ __forceinline__ void slot_vjp(const float* recv_es, const float* send_raw,
                                         const bf16_t* dx, float* acc, float sign,
                                         int S, int pdim, float comm, int mode) {
  float p2 = 1e-6f;
  float e[16], g[16], se[16];
  const float* recv = recv_es;
  const float* send = send_raw;
  if (mode != 0) {
    edge_state_of(send_raw, se, S, mode);
    send = se;
  }
  for (int s = 0; s < S; ++s) g[s] = (float)dx[s];
  for (int s = 0; s < S; ++s) {
    e[s] = recv[s] - send[s];
    if (s < pdim) p2 += e[s] * e[s];
  }
  const float n = sqrtf(p2);
  if (n > comm) {
    float gdotp = 0.f;
    for (int s = 0; s < pdim; ++s) gdotp += g[s] * e[s];
    const float inv_n = 1.f / n;
    for (int s = 0; s < pdim; ++s)
      acc[s] += sign * comm * (g[s] * inv_n - e[s] * gdotp * inv_n * inv_n * inv_n);
  } else {
    for (int s = 0; s < pdim; ++s) acc[s] += sign * g[s];
  }
  for (int s = pdim; s < S; ++s) acc[s] += sign * g[s];
}

__launch_bounds__(256) __global__
void edge_msg_in_bwd_kernel(const float* __restrict__ states, const bf16_t* __restrict__ dX,
                            float* __restrict__ dstates, int B, int N, int R, int S,
                            int pdim, int KP, float comm, int mode) {
  const int D = N + 1 + R;
  const int V = 2 * N + N * R;
  const long total = (long)B * V;
  for (long node = (long)blockIdx.x * blockDim.x + threadIdx.x; node < total;
       node += (long)gridDim.x * blockDim.x) {
    const int v = node % V;
    const int b = node / V;
    const float* st = states + (long)b * V * S;
    const bf16_t* dxb = dX + (long)b * N * D * KP;
    float acc[16];
    for (int s = 0; s < S; ++s) acc[s] = 0.f;
    const float* own = st + (long)v * S;
    float own_es_buf[16];
    const float* own_es = own;
    if (mode != 0) {
      edge_state_of(own, own_es_buf, S, mode);
      own_es = own_es_buf;
    }
    if (v < N) {
      // agent j: receiver side over its D slots, sender side in others' rows
      const int j = v;
      for (int d = 0; d < D; ++d) {
        const float* send = st + (long)sender_node(j, d, N, R) * S;
        slot_vjp(own_es, send, dxb + ((long)j * D + d) * KP, acc, 1.f, S, pdim, comm, mode);
      }
      for (int i = 0; i < N; ++i) {
        const float* ri = st + (long)i * S;
        float r2[16];
        if (mode != 0) {
          edge_state_of(ri, r2, S, mode);
          ri = r2;
        }
        slot_vjp(ri, own, dxb + ((long)i * D + j) * KP, acc, -1.f, S, pdim, comm, mode);
      }
    } else if (v < 2 * N) {
      const int j = v - N;  // goal j: sender in slot (j, N)
      const float* rj = st + (long)j * S;
      float rb[16];
      if (mode != 0) { edge_state_of(rj, rb, S, mode); rj = rb; }
      slot_vjp(rj, own, dxb + ((long)j * D + N) * KP, acc, -1.f, S, pdim, comm, mode);
    } else {
      const int h = v - 2 * N;  // lidar hit (j, r): sender in slot (j, N+1+r)
      const int j = h / R;
      const int r = h % R;
      const float* rj = st + (long)j * S;
      float rb[16];
      if (mode != 0) { edge_state_of(rj, rb, S, mode); rj = rb; }
      slot_vjp(rj, own, dxb + ((long)j * D + N + 1 + r) * KP, acc, -1.f, S, pdim, comm, mode);
    }
    float* out = dstates + ((long)b * V + v) * S;
    if (mode == 1) {
      // chain through d(es)/d(state): [x, y, v cos, v sin]
      const float th = own[2], vv = own[3];
      out[0] = acc[0];
      out[1] = acc[1];
      out[2] = acc[2] * (-vv * __sinf(th)) + acc[3] * (vv * __cosf(th));
      out[3] = acc[2] * __cosf(th) + acc[3] * __sinf(th);
    } else {
      for (int s = 0; s < S; ++s) out[s] = acc[s];
    }
  }
}
