// K11: batched dense QP solver, one 64-lane workgroup per problem.
//
//   min 1/2 x^T H x + g^T x   s.t.  C x <= b,  l <= x <= u
//
// Same algorithm as the torch oracle (gcbfplus_amd/ops/qp.py): Ruiz
// equilibration + cost scaling, ADMM with over-relaxation on A=[C; I], then
// an f64 active-set penalty polish. Fixed iteration count (JaxProxQP-style
// batched semantics, reference gcbf_plus.py:341-346) — every QP does the
// same work, no divergence across workgroups.
//
// Sizes: nv <= MAXNV variables, k <= MAXK inequality rows (m_c = k + nv).
// The per-iteration KKT solve uses an explicit K^{-1} = L^{-T} L^{-1}
// (computed once per QP) so each ADMM iteration is two lane-parallel
// matvecs; the f64 polish redoes an exact Cholesky solve on the active set.
#include "common.h"

template <int MAXNV, int MAXK>
__launch_bounds__(64) __global__
void proxqp_kernel(const float* __restrict__ Hg, const float* __restrict__ gg,
                   const float* __restrict__ Cg, const float* __restrict__ bg,
                   const float* __restrict__ lg, const float* __restrict__ ug,
                   float* __restrict__ xg, int M, int nv, int k, int iters,
                   float rho, float sigma, float alpha) {
  constexpr int MAXC = MAXNV + MAXK;
  const int q = blockIdx.x;
  if (q >= M) return;
  const int lane = threadIdx.x;

  __shared__ float Hs[MAXNV][MAXNV];    // scaled H
  __shared__ float As[MAXC][MAXNV];     // scaled [C; I]
  __shared__ float Km[MAXNV][MAXNV];    // K, then L (Cholesky factor)
  __shared__ float Linv[MAXNV][MAXNV];  // L^{-1}
  __shared__ float Kinv[MAXNV][MAXNV];
  __shared__ float gs[MAXNV], Dv[MAXNV], Ev[MAXC], lo[MAXC], hi[MAXC];
  __shared__ float xv[MAXNV], zv[MAXC], yv[MAXC], rhs[MAXNV], xt[MAXNV], zt[MAXC];
  __shared__ float red[1];
  __shared__ double Kp[MAXNV][MAXNV];   // f64 polish system
  __shared__ double xp[MAXNV], rp[MAXNV];

  const int mc = k + nv;
  const float INF = 3.0e38f;

  // ---- load & init -------------------------------------------------------
  for (int i = lane; i < nv * nv; i += 64) Hs[i / nv][i % nv] = Hg[(long)q * nv * nv + i];
  for (int i = lane; i < k * nv; i += 64) As[i / nv][i % nv] = Cg[(long)q * k * nv + i];
  for (int i = lane; i < nv * nv; i += 64) As[k + i / nv][i % nv] = (i / nv == i % nv) ? 1.f : 0.f;
  for (int i = lane; i < nv; i += 64) {
    gs[i] = gg[(long)q * nv + i];
    Dv[i] = 1.f;
    xv[i] = 0.f;
  }
  for (int i = lane; i < mc; i += 64) {
    lo[i] = (i < k) ? -INF : lg[(long)q * nv + (i - k)];
    hi[i] = (i < k) ? bg[(long)q * k + i] : ug[(long)q * nv + (i - k)];
    Ev[i] = 1.f;
    zv[i] = 0.f;
    yv[i] = 0.f;
  }
  __syncthreads();

  // ---- Ruiz equilibration (5 sweeps) -------------------------------------
  for (int sweep = 0; sweep < 5; ++sweep) {
    // column scales dn[i] = rsqrt(max(colinf(Hs), colinf(As)))
    for (int i = lane; i < nv; i += 64) {
      float cm = 1e-8f;
      for (int r = 0; r < nv; ++r) cm = fmaxf(cm, fabsf(Hs[r][i]));
      for (int r = 0; r < mc; ++r) cm = fmaxf(cm, fabsf(As[r][i]));
      rhs[i] = rsqrtf(cm);  // reuse rhs as dn scratch
    }
    for (int i = lane; i < mc; i += 64) {
      float rm = 1e-8f;
      for (int c = 0; c < nv; ++c) rm = fmaxf(rm, fabsf(As[i][c]));
      zt[i] = rsqrtf(rm);  // reuse zt as de scratch
    }
    __syncthreads();
    for (int i = lane; i < nv * nv; i += 64)
      Hs[i / nv][i % nv] *= rhs[i / nv] * rhs[i % nv];
    for (int i = lane; i < mc * nv; i += 64)
      As[i / nv][i % nv] *= zt[i / nv] * rhs[i % nv];
    for (int i = lane; i < nv; i += 64) Dv[i] *= rhs[i];
    for (int i = lane; i < mc; i += 64) Ev[i] *= zt[i];
    __syncthreads();
  }
  // cost scaling c = 1/max(1, mean colinf(Hs), ||gs*D||inf)
  {
    float cmax = 0.f, gmax = 0.f;
    for (int i = lane; i < nv; i += 64) {
      float cm = 0.f;
      for (int r = 0; r < nv; ++r) cm = fmaxf(cm, fabsf(Hs[r][i]));
      cmax += cm;
      gmax = fmaxf(gmax, fabsf(gs[i] * Dv[i]));
    }
    cmax = wave_reduce_sum(cmax) / nv;
    gmax = wave_reduce_max(gmax);
    const float cc = 1.f / fmaxf(1.f, fmaxf(cmax, gmax));
    __syncthreads();
    for (int i = lane; i < nv * nv; i += 64) Hs[i / nv][i % nv] *= cc;
    for (int i = lane; i < nv; i += 64) gs[i] = gs[i] * Dv[i] * cc;
    for (int i = lane; i < mc; i += 64) {
      lo[i] = (lo[i] <= -INF) ? -INF : lo[i] * Ev[i];
      hi[i] = (hi[i] >= INF) ? INF : hi[i] * Ev[i];
    }
  }
  __syncthreads();

  // ---- K = Hs + sigma I + rho A^T A; Cholesky; L^{-1}; K^{-1} ------------
  for (int i = lane; i < nv * nv; i += 64) {
    const int r = i / nv, c = i % nv;
    float a = Hs[r][c] + (r == c ? sigma : 0.f);
    for (int t = 0; t < mc; ++t) a += rho * As[t][r] * As[t][c];
    Km[r][c] = a;
  }
  __syncthreads();
  // in-place lower Cholesky (sequential over columns, lanes over rows)
  for (int j = 0; j < nv; ++j) {
    if (lane == 0) {
      float d = Km[j][j];
      for (int t = 0; t < j; ++t) d -= Km[j][t] * Km[j][t];
      Km[j][j] = sqrtf(fmaxf(d, 1e-12f));
    }
    __syncthreads();
    const float dj = Km[j][j];
    for (int i = j + 1 + lane; i < nv; i += 64) {
      float a = Km[i][j];
      for (int t = 0; t < j; ++t) a -= Km[i][t] * Km[j][t];
      Km[i][j] = a / dj;
    }
    __syncthreads();
  }
  // L^{-1}: lane c solves column c by forward substitution
  for (int c = lane; c < nv; c += 64) {
    for (int i = 0; i < nv; ++i) {
      float a = (i == c) ? 1.f : 0.f;
      for (int t = c; t < i; ++t) a -= Km[i][t] * Linv[t][c];
      Linv[i][c] = (i >= c) ? a / Km[i][i] : 0.f;
    }
  }
  __syncthreads();
  // Kinv[i][j] = sum_r Linv[r][i] * Linv[r][j]
  for (int i = lane; i < nv * nv; i += 64) {
    const int r = i / nv, c = i % nv;
    float a = 0.f;
    const int start = (r > c) ? r : c;
    for (int t = start; t < nv; ++t) a += Linv[t][r] * Linv[t][c];
    Kinv[r][c] = a;
  }
  __syncthreads();

  // ---- ADMM loop ---------------------------------------------------------
  for (int it = 0; it < iters; ++it) {
    for (int i = lane; i < nv; i += 64) {
      float a = sigma * xv[i] - gs[i];
      for (int r = 0; r < mc; ++r) a += As[r][i] * (rho * zv[r] - yv[r]);
      rhs[i] = a;
    }
    __syncthreads();
    for (int i = lane; i < nv; i += 64) {
      float a = 0.f;
      for (int j = 0; j < nv; ++j) a += Kinv[i][j] * rhs[j];
      xt[i] = a;
    }
    __syncthreads();
    for (int r = lane; r < mc; r += 64) {
      float a = 0.f;
      for (int j = 0; j < nv; ++j) a += As[r][j] * xt[j];
      zt[r] = a;
    }
    __syncthreads();
    for (int i = lane; i < nv; i += 64) xv[i] = alpha * xt[i] + (1.f - alpha) * xv[i];
    for (int r = lane; r < mc; r += 64) {
      const float zrel = alpha * zt[r] + (1.f - alpha) * zv[r];
      const float znew = fminf(fmaxf(zrel + yv[r] / rho, lo[r]), hi[r]);
      yv[r] += rho * (zrel - znew);
      zv[r] = znew;
    }
    __syncthreads();
  }
  // unscale x
  for (int i = lane; i < nv; i += 64) xv[i] *= Dv[i];
  __syncthreads();

  // ---- f64 active-set polish --------------------------------------------
  // active r: |y| above threshold OR unscaled Ax at a bound (tol 1e-4)
  float ymax = 0.f;
  for (int r = lane; r < mc; r += 64) ymax = fmaxf(ymax, fabsf(yv[r]));
  ymax = wave_reduce_max(ymax);
  // compute unscaled A x and activity; store bound target in zt, flag in Ev
  for (int r = lane; r < mc; r += 64) {
    // unscaled row r of A: [C; I] raw from global
    float ax = 0.f;
    if (r < k) {
      for (int j = 0; j < nv; ++j) ax += Cg[((long)q * k + r) * nv + j] * xv[j];
    } else {
      ax = xv[r - k];
    }
    const float lor = (r < k) ? -INF : lg[(long)q * nv + (r - k)];
    const float hir = (r < k) ? bg[(long)q * k + r] : ug[(long)q * nv + (r - k)];
    const bool at_lo = ax <= lor + 1e-4f;
    const bool at_hi = ax >= hir - 1e-4f;
    const bool act = at_lo || at_hi || (fabsf(yv[r]) > 1e-6f * ymax);
    float vb = at_hi ? hir : lor;
    if (!(act && fabsf(vb) < INF)) vb = ax;
    zt[r] = vb;          // bound target
    Ev[r] = act ? 1.f : 0.f;  // active flag (Ev re-used)
  }
  __syncthreads();

  const double mu = 1e8;
  for (int i = lane; i < nv * nv; i += 64) {
    const int r = i / nv, c = i % nv;
    double a = (double)Hg[(long)q * nv * nv + i];
    for (int t = 0; t < mc; ++t) {
      if (Ev[t] == 0.f) continue;
      double ar, ac;
      if (t < k) {
        ar = Cg[((long)q * k + t) * nv + r];
        ac = Cg[((long)q * k + t) * nv + c];
      } else {
        ar = (t - k == r) ? 1.0 : 0.0;
        ac = (t - k == c) ? 1.0 : 0.0;
      }
      a += mu * ar * ac;
    }
    Kp[r][c] = a;
  }
  for (int i = lane; i < nv; i += 64) {
    double a = -(double)gg[(long)q * nv + i];
    for (int t = 0; t < mc; ++t) {
      if (Ev[t] == 0.f) continue;
      double ar = (t < k) ? (double)Cg[((long)q * k + t) * nv + i] : ((t - k == i) ? 1.0 : 0.0);
      a += mu * ar * (double)zt[t];
    }
    rp[i] = a;
  }
  __syncthreads();
  // f64 Cholesky + solve (in Kp)
  for (int j = 0; j < nv; ++j) {
    if (lane == 0) {
      double d = Kp[j][j];
      for (int t = 0; t < j; ++t) d -= Kp[j][t] * Kp[j][t];
      Kp[j][j] = sqrt(fmax(d, 1e-18));
    }
    __syncthreads();
    const double dj = Kp[j][j];
    for (int i = j + 1 + lane; i < nv; i += 64) {
      double a = Kp[i][j];
      for (int t = 0; t < j; ++t) a -= Kp[i][t] * Kp[j][t];
      Kp[i][j] = a / dj;
    }
    __syncthreads();
  }
  if (lane == 0) {
    for (int i = 0; i < nv; ++i) {
      double a = rp[i];
      for (int t = 0; t < i; ++t) a -= Kp[i][t] * xp[t];
      xp[i] = a / Kp[i][i];
    }
    for (int i = nv - 1; i >= 0; --i) {
      double a = xp[i];
      for (int t = i + 1; t < nv; ++t) a -= Kp[t][i] * xp[t];
      xp[i] = a / Kp[i][i];
    }
  }
  __syncthreads();

  // score both candidates on the ORIGINAL problem; take polish if no worse
  // (lane 0 computes scalars; nv,mc tiny)
  if (lane == 0) {
    float obj_a = 0.f, obj_p = 0.f, viol_a = 0.f, viol_p = 0.f;
    for (int i = 0; i < nv; ++i) {
      float ha = 0.f, hp = 0.f;
      for (int j = 0; j < nv; ++j) {
        const float hij = Hg[((long)q * nv + i) * nv + j];
        ha += hij * xv[j];
        hp += hij * (float)xp[j];
      }
      const float gi = gg[(long)q * nv + i];
      obj_a += 0.5f * xv[i] * ha + gi * xv[i];
      obj_p += 0.5f * (float)xp[i] * hp + gi * (float)xp[i];
    }
    for (int r = 0; r < mc; ++r) {
      float axa = 0.f, axp = 0.f;
      if (r < k) {
        for (int j = 0; j < nv; ++j) {
          const float c = Cg[((long)q * k + r) * nv + j];
          axa += c * xv[j];
          axp += c * (float)xp[j];
        }
      } else {
        axa = xv[r - k];
        axp = (float)xp[r - k];
      }
      const float lor = (r < k) ? -INF : lg[(long)q * nv + (r - k)];
      const float hir = (r < k) ? bg[(long)q * k + r] : ug[(long)q * nv + (r - k)];
      viol_a += fmaxf(axa - hir, 0.f) + fmaxf(lor - axa, 0.f);
      viol_p += fmaxf(axp - hir, 0.f) + fmaxf(lor - axp, 0.f);
    }
    red[0] = (viol_p <= viol_a + 1e-5f && obj_p <= obj_a + 1e-6f) ? 1.f : 0.f;
  }
  __syncthreads();
  const bool take_p = red[0] > 0.5f;
  for (int i = lane; i < nv; i += 64)
    xg[(long)q * nv + i] = take_p ? (float)xp[i] : xv[i];
}

template __global__ void proxqp_kernel<32, 16>(const float*, const float*, const float*, const float*, const float*, const float*, float*, int, int, int, int, float, float, float);
template __global__ void proxqp_kernel<64, 32>(const float*, const float*, const float*, const float*, const float*, const float*, float*, int, int, int, int, float, float, float);
