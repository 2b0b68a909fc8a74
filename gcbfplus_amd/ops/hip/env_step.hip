// K5-K8: fused DoubleIntegrator environment step.
//
// One workgroup per env world; replaces ~60 eager torch kernels per rollout
// step with one launch. Computes (reference env/double_integrator.py):
//   u_ref (clipped-error LQR, :332-338)        reward = -mean|a - u_ref|^2
//   clip_action + euler + clip_state (:128-143) cost (:183-198)
//   LiDAR re-scan from the NEXT positions (:288-320, via raytrace math)
//   edge-slot mask (aa dist / goal / lidar, :223-264)
// Outputs: next_states (B,V,4), mask (B,N,D), reward (B,), cost (B,).
#include "common.h"

__launch_bounds__(256) __global__
void di_env_step_kernel(const float* __restrict__ states,   // (B, V, 4)
                        const float* __restrict__ action,   // (B, N, 2)
                        const float* __restrict__ points,   // (B, K, 4, 2)
                        const float* __restrict__ Kmat,     // (2, 4) LQR gain
                        float* __restrict__ next_states,    // (B, V, 4)
                        bool* __restrict__ mask,            // (B, N, D)
                        float* __restrict__ reward,         // (B,)
                        float* __restrict__ cost,           // (B,)
                        int N, int K, int R, float dt, float inv_m, float comm,
                        float car_r, float vmax) {
  extern __shared__ float smem[];
  float* sNext = smem;             // [N][4] next agent states
  float* sCur = sNext + N * 4;     // [N][2] current agent pos
  float* sObs = sCur + N * 2;      // [K][4][2] obstacle corners
  __shared__ float red[4][2];      // reward / cost partials
  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  const int V = 2 * N + N * R;
  const int D = N + 1 + R;
  const float* st = states + (long)b * V * 4;
  float* nx = next_states + (long)b * V * 4;

  for (int i = tid; i < K * 8; i += 256) sObs[i] = points[(long)b * K * 8 + i];

  // ---- phase 1: per-agent dynamics + reward terms ------------------------
  float r_part = 0.f, c_part = 0.f;
  for (int i = tid; i < N; i += 256) {
    const float* a = st + (long)i * 4;        // agent state
    const float* gl = st + (long)(N + i) * 4; // goal state
    sCur[i * 2] = a[0];
    sCur[i * 2 + 1] = a[1];
    // u_ref (reference :332-338)
    float err[4], nrm = 0.f;
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      err[s] = gl[s] - a[s];
      nrm += err[s] * err[s];
    }
    nrm = fmaxf(sqrtf(nrm), 1e-9f);
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      const float emax = fabsf(err[s] / nrm * comm);
      err[s] = fminf(fmaxf(err[s], -emax), emax);
    }
    float uref[2];
#pragma unroll
    for (int u = 0; u < 2; ++u) {
      float acc = 0.f;
#pragma unroll
      for (int s = 0; s < 4; ++s) acc += Kmat[u * 4 + s] * err[s];
      uref[u] = fminf(fmaxf(acc, -1.f), 1.f);
    }
    const float ax = fminf(fmaxf(action[((long)b * N + i) * 2], -1.f), 1.f);
    const float ay = fminf(fmaxf(action[((long)b * N + i) * 2 + 1], -1.f), 1.f);
    r_part += (ax - uref[0]) * (ax - uref[0]) + (ay - uref[1]) * (ay - uref[1]);
    // euler + state clip (:128-143)
    sNext[i * 4 + 0] = a[0] + a[2] * dt;
    sNext[i * 4 + 1] = a[1] + a[3] * dt;
    sNext[i * 4 + 2] = fminf(fmaxf(a[2] + ax * inv_m * dt, -vmax), vmax);
    sNext[i * 4 + 3] = fminf(fmaxf(a[3] + ay * inv_m * dt, -vmax), vmax);
  }
  __syncthreads();

  // ---- phase 2: cost on the CURRENT state (:183-198) ---------------------
  for (int i = tid; i < N; i += 256) {
    const float xi = sCur[i * 2], yi = sCur[i * 2 + 1];
    bool coll = false;
    for (int j = 0; j < N; ++j) {
      if (j == i) continue;
      const float dx = xi - sCur[j * 2], dy = yi - sCur[j * 2 + 1];
      coll = coll || (dx * dx + dy * dy < 4.f * car_r * car_r);
    }
    bool inside = false;
    for (int k = 0; k < K; ++k) {
      const float* q = sObs + k * 8;
      const float cx = 0.5f * (q[0] + q[4]), cy = 0.5f * (q[1] + q[5]);
      const float ux = q[0] - q[2], uy = q[1] - q[3];
      const float vx = q[2] - q[4], vy = q[3] - q[5];
      const float w = sqrtf(ux * ux + uy * uy), h = sqrtf(vx * vx + vy * vy);
      const float rx = xi - cx, ry = yi - cy;
      const float du = fabsf(rx * ux + ry * uy) / w - 0.5f * w;
      const float dv = fabsf(rx * vx + ry * vy) / h - 0.5f * h;
      // rounded-corner inflation by car_r (reference obstacle.py:52-63)
      const bool in_down = (du < car_r) && (dv < 0.f);
      const bool in_up = (du < 0.f) && (dv < car_r);
      const bool corner = (du > 0.f) && (dv > 0.f) &&
                          (sqrtf(du * du + dv * dv) < car_r);
      inside = inside || in_down || in_up || corner;
    }
    c_part += (coll ? 1.f : 0.f) + (inside ? 1.f : 0.f);
  }

  // write next agent + copy goal states
  for (int i = tid; i < N * 4; i += 256) nx[i] = sNext[i];
  for (int i = tid; i < N * 4; i += 256) nx[N * 4 + i] = st[N * 4 + i];
  __syncthreads();

  // ---- phase 3: LiDAR rescan from NEXT positions + lidar mask ------------
  const float TWO_PI = 6.283185307179586f;
  for (int item = tid; item < N * R; item += 256) {
    const int n = item / R;
    const int rr = item % R;
    const float x1 = sNext[n * 4], y1 = sNext[n * 4 + 1];
    const float th = -3.14159265358979f + TWO_PI * rr / R;
    const float x2 = x1 + __cosf(th) * comm;
    const float y2 = y1 + __sinf(th) * comm;
    float alpha = 1e6f;
    bool inside = false;
    for (int k = 0; k < K; ++k) {
      const float* q = sObs + k * 8;
      const float cx = 0.5f * (q[0] + q[4]), cy = 0.5f * (q[1] + q[5]);
      const float ux = q[0] - q[2], uy = q[1] - q[3];
      const float vx = q[2] - q[4], vy = q[3] - q[5];
      const float w2 = ux * ux + uy * uy, h2 = vx * vx + vy * vy;
      const float rx = x1 - cx, ry = y1 - cy;
      const float du = fabsf(rx * ux + ry * uy) / sqrtf(w2) - 0.5f * sqrtf(w2);
      const float dv = fabsf(rx * vx + ry * vy) / sqrtf(h2) - 0.5f * sqrtf(h2);
      inside = inside || (du < 0.f && dv < 0.f);
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        const float x3 = q[e * 2], y3 = q[e * 2 + 1];
        const int ep = (e + 3) & 3;
        const float x4 = q[ep * 2], y4 = q[ep * 2 + 1];
        float det = (x1 - x2) * (y4 - y3) - (y1 - y2) * (x4 - x3);
        const float sgn = det >= 0.f ? 1.f : -1.f;
        det = sgn * fminf(fmaxf(fabsf(det), 1e-7f), 1e7f);
        const float aa = ((y4 - y3) * (x1 - x3) - (x4 - x3) * (y1 - y3)) / det;
        const float bt = (-(y1 - y2) * (x1 - x3) + (x1 - x2) * (y1 - y3)) / det;
        if (aa >= 0.f && aa <= 1.f && bt >= 0.f && bt <= 1.f) alpha = fminf(alpha, aa);
      }
    }
    if (inside) alpha = 0.f;
    float* hit = nx + (long)(2 * N + n * R + rr) * 4;
    hit[0] = x1 + (x2 - x1) * alpha;
    hit[1] = y1 + (y2 - y1) * alpha;
    hit[2] = 0.f;
    hit[3] = 0.f;
    // lidar mask: |next_pos - hit| < comm - 0.1 ; dist = alpha * comm
    mask[((long)b * N + n) * D + N + 1 + rr] = (alpha * comm) < (comm - 1e-1f);
  }

  // ---- phase 4: agent-agent + goal mask on NEXT positions ----------------
  for (int item = tid; item < N * N; item += 256) {
    const int i = item / N, j = item % N;
    const float dx = sNext[i * 4] - sNext[j * 4];
    const float dy = sNext[i * 4 + 1] - sNext[j * 4 + 1];
    mask[((long)b * N + i) * D + j] = (i != j) && (dx * dx + dy * dy < comm * comm);
  }
  for (int i = tid; i < N; i += 256) mask[((long)b * N + i) * D + N] = true;

  // ---- reduce reward / cost ---------------------------------------------
  r_part = wave_reduce_sum(r_part);
  c_part = wave_reduce_sum(c_part);
  const int w = tid >> 6;
  if ((tid & 63) == 0) {
    red[w][0] = r_part;
    red[w][1] = c_part;
  }
  __syncthreads();
  if (tid == 0) {
    reward[b] = -(red[0][0] + red[1][0] + red[2][0] + red[3][0]) / N;
    cost[b] = (red[0][1] + red[1][1] + red[2][1] + red[3][1]) / N;
  }
}
