// K1: batched 2D LiDAR raytrace against rotated-rectangle sets.
// One workgroup per env (obstacle corners staged in LDS once), threads cover
// the N*R beam grid. Mirrors reference obstacle.py:65-96 (4-edge 2x2 solves)
// + env/utils.py:110-124 (inside -> alpha=0, no hit -> alpha=1e6).
#include "common.h"

__launch_bounds__(256) __global__
void raytrace_rect_kernel(const float* __restrict__ pos,     // (B, N, 2)
                          const float* __restrict__ points,  // (B, K, 4, 2)
                          float* __restrict__ hits,          // (B, N, R, 2)
                          int N, int K, int R, float range) {
  extern __shared__ float sp[];  // [K][4][2]
  const int b = blockIdx.x;
  for (int i = threadIdx.x; i < K * 8; i += 256) sp[i] = points[(long)b * K * 8 + i];
  __syncthreads();

  const float TWO_PI = 6.283185307179586f;
  for (int item = threadIdx.x; item < N * R; item += 256) {
    const int n = item / R;
    const int r = item % R;
    const float x1 = pos[((long)b * N + n) * 2];
    const float y1 = pos[((long)b * N + n) * 2 + 1];
    const float th = -3.14159265358979f + TWO_PI * r / R;
    const float x2 = x1 + __cosf(th) * range;
    const float y2 = y1 + __sinf(th) * range;

    float alpha = 1e6f;
    bool inside = false;
    for (int k = 0; k < K; ++k) {
      const float* q = sp + k * 8;
      // inside test (r=0): both body-frame offsets negative. Recover the
      // body frame from corners: c = (q0+q2)/2, axis u = (q0-q1)/w.
      const float cx = 0.5f * (q[0] + q[4]);
      const float cy = 0.5f * (q[1] + q[5]);
      const float ux = q[0] - q[2], uy = q[1] - q[3];       // width axis * w
      const float vx = q[2] - q[4], vy = q[3] - q[5];       // height axis * h
      const float w2 = ux * ux + uy * uy, h2 = vx * vx + vy * vy;
      const float rx = x1 - cx, ry = y1 - cy;
      const float du = fabsf(rx * ux + ry * uy) / sqrtf(w2) - 0.5f * sqrtf(w2);
      const float dv = fabsf(rx * vx + ry * vy) / sqrtf(h2) - 0.5f * sqrtf(h2);
      inside = inside || (du < 0.f && dv < 0.f);
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        const float x3 = q[e * 2], y3 = q[e * 2 + 1];
        const int ep = (e + 3) & 3;  // previous corner (reference edge order)
        const float x4 = q[ep * 2], y4 = q[ep * 2 + 1];
        float det = (x1 - x2) * (y4 - y3) - (y1 - y2) * (x4 - x3);
        const float sgn = det >= 0.f ? 1.f : -1.f;
        det = sgn * fminf(fmaxf(fabsf(det), 1e-7f), 1e7f);
        const float a = ((y4 - y3) * (x1 - x3) - (x4 - x3) * (y1 - y3)) / det;
        const float bt = (-(y1 - y2) * (x1 - x3) + (x1 - x2) * (y1 - y3)) / det;
        if (a >= 0.f && a <= 1.f && bt >= 0.f && bt <= 1.f) alpha = fminf(alpha, a);
      }
    }
    if (inside) alpha = 0.f;
    hits[(((long)b * N + n) * R + r) * 2] = x1 + (x2 - x1) * alpha;
    hits[(((long)b * N + n) * R + r) * 2 + 1] = y1 + (y2 - y1) * alpha;
  }
}

// K1 (3D): LiDAR theta x phi fan + poles vs sphere set, fused with the
// stable top-k closest-hit selection (reference env/utils.py:49-79 3D grid,
// obstacle.py:237-270 quadratic solve, utils.py:127-131 argsort top-k).
// One workgroup per (env, agent); spheres + per-beam alphas staged in LDS.
// Beam count R = (n_beams/2)*n_beams + 2 (<= ~2k), selection by packed
// (alpha_bits, beam_idx) u64 min-reduction — ties resolve to the lowest
// beam index, matching jnp.argsort's stable order.
// When states_out != nullptr the selected hits are written as graph node
// rows instead: states_out[b, 2N + n*topk + t, 0:3] = hit, rows padded with
// zeros to S floats, and mask_out[b, n, N+1+t] = hit_dist < range - 0.1
// (the lidar edge mask of build_mask) — this is part B of the fused
// LinearDrone step.
__launch_bounds__(256) __global__
void raytrace_sphere_topk_kernel(const float* __restrict__ pos,      // (B, N, 3)
                                 const float* __restrict__ centers,  // (B, K, 3)
                                 const float* __restrict__ radii,    // (B, K)
                                 float* __restrict__ hits,           // (B, N, T, 3)
                                 float* __restrict__ states_out,     // (B, V, S) or null
                                 bool* __restrict__ mask_out,        // (B, N, D) or null
                                 int S, int D,
                                 int N, int K, int n_beams, int topk,
                                 float range) {
  extern __shared__ float smem[];
  float* sSph = smem;                  // [K][4]: cx, cy, cz, r
  float* sAlpha = sSph + K * 4;        // [R]
  const int nt = n_beams / 2;
  const int R = nt * n_beams + 2;
  unsigned long long* sKey =
      (unsigned long long*)(sAlpha + ((R + 1) & ~1));  // [256] reduce scratch
  const int bn = blockIdx.x;
  const int b = bn / N, n = bn % N;
  const int tid = threadIdx.x;

  for (int i = tid; i < K; i += 256) {
    sSph[i * 4 + 0] = centers[((long)b * K + i) * 3 + 0];
    sSph[i * 4 + 1] = centers[((long)b * K + i) * 3 + 1];
    sSph[i * 4 + 2] = centers[((long)b * K + i) * 3 + 2];
    sSph[i * 4 + 3] = radii[(long)b * K + i];
  }
  __syncthreads();

  const float ox = pos[((long)b * N + n) * 3 + 0];
  const float oy = pos[((long)b * N + n) * 3 + 1];
  const float oz = pos[((long)b * N + n) * 3 + 2];

  // origin-inside test (get_lidar zeroes all alphas for inside origins)
  bool inside = false;
  for (int k = 0; k < K; ++k) {
    const float dx = ox - sSph[k * 4], dy = oy - sSph[k * 4 + 1],
                dz = oz - sSph[k * 4 + 2];
    inside = inside || (dx * dx + dy * dy + dz * dz <
                        sSph[k * 4 + 3] * sSph[k * 4 + 3]);
  }

  const float PI = 3.14159265358979f;
  for (int r = tid; r < R; r += 256) {
    // beam direction (matches env/utils.beam_dirs_3d linspaces exactly)
    float dx, dy, dz;
    if (r >= nt * n_beams) {
      dx = 0.f; dy = 0.f; dz = (r == nt * n_beams) ? 1.f : -1.f;
    } else {
      const int it = r / n_beams, ip = r % n_beams;
      const float t0 = -PI / 2 + 2 * PI / n_beams;
      const float t1 = PI / 2 - 2 * PI / n_beams;
      const float th = (nt > 1) ? t0 + (t1 - t0) * it / (nt - 1) : t0;
      const float ph = -PI + 2 * PI * ip / n_beams;
      dx = __cosf(th) * __cosf(ph);
      dy = __cosf(th) * __sinf(ph);
      dz = __sinf(th);
    }
    const float ex = dx * range, ey = dy * range, ez = dz * range;
    float alpha = 1e6f;
    const float a2 = ex * ex + ey * ey + ez * ez;  // = range^2
    for (int k = 0; k < K; ++k) {
      const float sx = ox - sSph[k * 4], sy = oy - sSph[k * 4 + 1],
                  sz = oz - sSph[k * 4 + 2];
      const float rad = sSph[k * 4 + 3];
      const float bq = 2.f * (sx * ex + sy * ey + sz * ez);
      const float cq = sx * sx + sy * sy + sz * sz - rad * rad;
      const float disc = bq * bq - 4.f * a2 * cq;
      if (disc >= 0.f) {
        const float sq = sqrtf(disc);
        const float r1 = (-bq - sq) / (2.f * a2);
        const float r2 = (-bq + sq) / (2.f * a2);
        if (r1 >= 0.f && r1 <= 1.f) alpha = fminf(alpha, r1);
        if (r2 >= 0.f && r2 <= 1.f) alpha = fminf(alpha, r2);
      }
    }
    if (inside) alpha = 0.f;
    sAlpha[r] = alpha;
  }
  __syncthreads();

  // stable top-k smallest: k rounds of u64 (alpha_bits<<32 | idx) min-reduce.
  // alphas are >= 0 so their IEEE bits are monotone as unsigned.
  float* out = (hits != nullptr) ? hits + ((long)b * N + n) * (long)topk * 3 : nullptr;
  for (int sel = 0; sel < topk; ++sel) {
    unsigned long long best = ~0ULL;
    for (int r = tid; r < R; r += 256) {
      const float a = sAlpha[r];
      if (a < 1e30f) {  // not yet consumed
        const unsigned long long key =
            ((unsigned long long)__float_as_uint(a) << 32) | (unsigned)r;
        best = key < best ? key : best;
      }
    }
    sKey[tid] = best;
    __syncthreads();
    for (int s = 128; s > 0; s >>= 1) {
      if (tid < s) sKey[tid] = sKey[tid + s] < sKey[tid] ? sKey[tid + s] : sKey[tid];
      __syncthreads();
    }
    const unsigned long long win = sKey[0];
    const int widx = (int)(win & 0xffffffffu);
    if (tid == 0) {
      const float a = __uint_as_float((unsigned)(win >> 32));
      // recompute the winning beam's direction
      float dx, dy, dz;
      if (widx >= nt * n_beams) {
        dx = 0.f; dy = 0.f; dz = (widx == nt * n_beams) ? 1.f : -1.f;
      } else {
        const int it = widx / n_beams, ip = widx % n_beams;
        const float t0 = -PI / 2 + 2 * PI / n_beams;
        const float t1 = PI / 2 - 2 * PI / n_beams;
        const float th = (nt > 1) ? t0 + (t1 - t0) * it / (nt - 1) : t0;
        const float ph = -PI + 2 * PI * ip / n_beams;
        dx = __cosf(th) * __cosf(ph);
        dy = __cosf(th) * __sinf(ph);
        dz = __sinf(th);
      }
      if (states_out != nullptr) {
        const int V = 2 * N + N * topk;
        float* row = states_out + ((long)b * V + 2 * N + n * topk + sel) * S;
        row[0] = ox + dx * range * a;
        row[1] = oy + dy * range * a;
        row[2] = oz + dz * range * a;
        for (int s = 3; s < S; ++s) row[s] = 0.f;
        mask_out[((long)b * N + n) * D + N + 1 + sel] = (a * range) < (range - 1e-1f);
      } else {
        out[sel * 3 + 0] = ox + dx * range * a;
        out[sel * 3 + 1] = oy + dy * range * a;
        out[sel * 3 + 2] = oz + dz * range * a;
      }
      sAlpha[widx] = 1e31f;  // consume
    }
    __syncthreads();
  }
}
