// K5-K8: fused environment steps.
//
// One workgroup per env world; replaces ~60 eager torch kernels per rollout
// step with one launch.
//
// DYN=0 DoubleIntegrator (reference env/double_integrator.py):
//   u_ref (clipped-error LQR, :332-338)        reward = -mean|a - u_ref|^2
//   clip_action + euler + clip_state (:128-143) cost (:183-198)
// DYN=1 DubinsCar (reference env/dubins_car.py):
//   u_ref (PID heading/speed, :328-379)         reward as above
//   clip_action(+-3) + stop-at-goal freeze (:483-487)
//   xdot = [v cos th, v sin th, 20*omega, a] euler + v clip (:104-122)
// Shared phases: cost on the CURRENT state, LiDAR re-scan from the NEXT
// positions (2D rectangles), edge-slot mask (aa dist / goal / lidar).
// Outputs: next_states (B,V,S), mask (B,N,D), reward (B,), cost (B,).
#include "common.h"

__device__ __forceinline__ float pos_mod_2pi(float x) {
  const float TWO_PI = 6.283185307179586f;
  return x - floorf(x / TWO_PI) * TWO_PI;  // python % semantics (in [0, 2pi))
}

template <int DYN>
__launch_bounds__(256) __global__
void env_step2d_kernel(const float* __restrict__ states,   // (B, V, 4)
                       const float* __restrict__ action,   // (B, N, 2)
                       const float* __restrict__ points,   // (B, K, 4, 2)
                       const float* __restrict__ Kmat,     // (2, 4) LQR gain (DI)
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
    if (DYN == 0) {
      // u_ref (reference double_integrator.py:332-338)
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
    } else {
      // DubinsCar: state (x, y, theta, v); action (omega, acc)
      const float PI = 3.14159265358979f;
      const float k_omega = 1.f, k_v = 2.3f, k_a = 2.5f;
      const float pdx = a[0] - gl[0], pdy = a[1] - gl[1];
      const float dist = sqrtf(pdx * pdx + pdy * pdy);
      // u_ref (reference dubins_car.py:328-379)
      const float theta_t = pos_mod_2pi(atan2f(-pdy, -pdx));
      const float theta = pos_mod_2pi(a[2]);
      const float theta_diff = theta_t - theta;
      const float ct = __cosf(a[2]), st_ = __sinf(a[2]);
      const float inner = (-pdx * ct - pdy * st_) / (dist + 1e-4f);
      const float theta_between = acosf(fminf(fmaxf(inner, -1.f), 1.f));
      const bool fwd = (theta_diff < PI) && (theta_diff >= 0.f);
      const bool bwd = (theta_diff > -PI) && (theta_diff <= 0.f);
      const bool le_pi = theta <= PI;
      float omega = le_pi ? (fwd ? k_omega * theta_between : -k_omega * theta_between)
                          : (bwd ? -k_omega * theta_between : k_omega * theta_between);
      omega = fminf(fmaxf(omega, -5.f), 5.f);
      const float nrm = sqrtf(1e-6f + pdx * pdx + pdy * pdy);
      const float coef = nrm > comm ? comm / nrm : 1.f;
      const float a_ref = -k_a * a[3] + k_v * sqrtf(coef * pdx * coef * pdx +
                                                    coef * pdy * coef * pdy);
      // clip_action +-3 (action_lim)
      const float aw = fminf(fmaxf(action[((long)b * N + i) * 2], -3.f), 3.f);
      const float aa = fminf(fmaxf(action[((long)b * N + i) * 2 + 1], -3.f), 3.f);
      r_part += (aw - omega) * (aw - omega) + (aa - a_ref) * (aa - a_ref);
      // stop-at-goal freeze (:483-487) + euler + v clip (:104-122)
      const float stop = dist < car_r * 0.5f ? 0.f : 1.f;
      sNext[i * 4 + 0] = a[0] + __cosf(a[2]) * a[3] * dt * stop;
      sNext[i * 4 + 1] = a[1] + st_ * a[3] * dt * stop;
      sNext[i * 4 + 2] = a[2] + aw * 20.f * dt * stop;
      sNext[i * 4 + 3] = fminf(fmaxf(a[3] + aa * dt * stop, -vmax), vmax);
    }
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

template __global__ void env_step2d_kernel<0>(const float*, const float*, const float*,
                                              const float*, float*, bool*, float*, float*,
                                              int, int, int, float, float, float, float, float);
template __global__ void env_step2d_kernel<1>(const float*, const float*, const float*,
                                              const float*, float*, bool*, float*, float*,
                                              int, int, int, float, float, float, float, float);

// ---------------------------------------------------------------------------
// LinearDrone (3D, spheres) fused step part A (reference linear_drone.py):
//   u_ref = clip(clip_err(goal - x) @ K^T) (inherited DI :332-338 form, 3x6 K)
//   clip_action (+-1) + euler xdot = A x + B u (:122-134) + vel clip (+-0.5)
//   reward = -mean|a - u_ref|^2; cost = collisions + sphere inside (:136-150)
//   agent-agent + goal mask on NEXT positions
// Part B (lidar rows + lidar mask) is raytrace_sphere_topk with graph output.
// ---------------------------------------------------------------------------
__launch_bounds__(256) __global__
void drone3d_step_kernel(const float* __restrict__ states,   // (B, V, 6)
                         const float* __restrict__ action,   // (B, N, 3)
                         const float* __restrict__ centers,  // (B, K, 3)
                         const float* __restrict__ radii,    // (B, K)
                         const float* __restrict__ Kmat,     // (3, 6)
                         const float* __restrict__ Arow,     // (6, 6) A matrix
                         float* __restrict__ next_states,    // (B, V, 6)
                         bool* __restrict__ mask,            // (B, N, D)
                         float* __restrict__ reward,         // (B,)
                         float* __restrict__ cost,           // (B,)
                         int N, int K, int R, float dt, float bgain, float comm,
                         float drone_r, float vmax) {
  extern __shared__ float smem[];
  float* sNext = smem;             // [N][6]
  float* sCur = sNext + N * 6;     // [N][3] current pos
  float* sSph = sCur + N * 3;      // [K][4]
  __shared__ float red[4][2];
  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  const int V = 2 * N + N * R;
  const int D = N + 1 + R;
  const float* st = states + (long)b * V * 6;
  float* nx = next_states + (long)b * V * 6;

  for (int i = tid; i < K; i += 256) {
    sSph[i * 4 + 0] = centers[((long)b * K + i) * 3 + 0];
    sSph[i * 4 + 1] = centers[((long)b * K + i) * 3 + 1];
    sSph[i * 4 + 2] = centers[((long)b * K + i) * 3 + 2];
    sSph[i * 4 + 3] = radii[(long)b * K + i];
  }

  float r_part = 0.f, c_part = 0.f;
  for (int i = tid; i < N; i += 256) {
    const float* a = st + (long)i * 6;
    const float* gl = st + (long)(N + i) * 6;
    sCur[i * 3 + 0] = a[0];
    sCur[i * 3 + 1] = a[1];
    sCur[i * 3 + 2] = a[2];
    // u_ref: clipped-error LQR over the 6-dim error
    float err[6], nrm = 0.f;
#pragma unroll
    for (int s = 0; s < 6; ++s) {
      err[s] = gl[s] - a[s];
      nrm += err[s] * err[s];
    }
    nrm = fmaxf(sqrtf(nrm), 1e-9f);
#pragma unroll
    for (int s = 0; s < 6; ++s) {
      const float emax = fabsf(err[s] / nrm * comm);
      err[s] = fminf(fmaxf(err[s], -emax), emax);
    }
    float ac[3];
#pragma unroll
    for (int u = 0; u < 3; ++u) {
      float acc = 0.f;
#pragma unroll
      for (int s = 0; s < 6; ++s) acc += Kmat[u * 6 + s] * err[s];
      const float uref = fminf(fmaxf(acc, -1.f), 1.f);
      const float av = fminf(fmaxf(action[((long)b * N + i) * 3 + u], -1.f), 1.f);
      ac[u] = av;
      r_part += (av - uref) * (av - uref);
    }
    // euler: x + (A x + B u) dt, then clip velocities
    float xd[6];
#pragma unroll
    for (int s = 0; s < 6; ++s) {
      float acc = 0.f;
#pragma unroll
      for (int t = 0; t < 6; ++t) acc += Arow[s * 6 + t] * a[t];
      if (s >= 3) acc += bgain * ac[s - 3];
      xd[s] = acc;
    }
#pragma unroll
    for (int s = 0; s < 3; ++s) sNext[i * 6 + s] = a[s] + xd[s] * dt;
#pragma unroll
    for (int s = 3; s < 6; ++s)
      sNext[i * 6 + s] = fminf(fmaxf(a[s] + xd[s] * dt, -vmax), vmax);
  }
  __syncthreads();

  // cost on CURRENT state
  for (int i = tid; i < N; i += 256) {
    const float xi = sCur[i * 3], yi = sCur[i * 3 + 1], zi = sCur[i * 3 + 2];
    bool coll = false;
    for (int j = 0; j < N; ++j) {
      if (j == i) continue;
      const float dx = xi - sCur[j * 3], dy = yi - sCur[j * 3 + 1],
                  dz = zi - sCur[j * 3 + 2];
      coll = coll || (dx * dx + dy * dy + dz * dz < 4.f * drone_r * drone_r);
    }
    bool inside = false;
    for (int k = 0; k < K; ++k) {
      const float dx = xi - sSph[k * 4], dy = yi - sSph[k * 4 + 1],
                  dz = zi - sSph[k * 4 + 2];
      const float rr = sSph[k * 4 + 3] + drone_r;
      inside = inside || (dx * dx + dy * dy + dz * dz < rr * rr);
    }
    c_part += (coll ? 1.f : 0.f) + (inside ? 1.f : 0.f);
  }

  for (int i = tid; i < N * 6; i += 256) nx[i] = sNext[i];
  for (int i = tid; i < N * 6; i += 256) nx[N * 6 + i] = st[N * 6 + i];
  __syncthreads();

  // aa + goal mask on NEXT positions (lidar mask written by part B)
  for (int item = tid; item < N * N; item += 256) {
    const int i = item / N, j = item % N;
    const float dx = sNext[i * 6] - sNext[j * 6];
    const float dy = sNext[i * 6 + 1] - sNext[j * 6 + 1];
    const float dz = sNext[i * 6 + 2] - sNext[j * 6 + 2];
    mask[((long)b * N + i) * D + j] =
        (i != j) && (dx * dx + dy * dy + dz * dz < comm * comm);
  }
  for (int i = tid; i < N; i += 256) mask[((long)b * N + i) * D + N] = true;

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
