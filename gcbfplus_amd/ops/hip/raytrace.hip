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
