// K11: batched dense QP solver, one 64-lane workgroup per problem.
//
//   min 1/2 x^T H x + g^T x   s.t.  C x <= b,  l <= x <= u
//
// Mirrors the torch oracle (gcbfplus_amd/ops/qp.py) step for step: Ruiz
// equilibration + cost scaling, ADMM with over-relaxation and OSQP-style
// adaptive rho (re-factor every 25 iters), then an iterated f64 active-set
// penalty polish with multiplier-sign release. Fixed iteration count
// (JaxProxQP-style batched semantics, reference gcbf_plus.py:341-346).
//
// The per-iteration KKT solve uses an explicit K^{-1} = L^{-T} L^{-1}
// (recomputed on each rho change) so each ADMM iteration is two
// lane-parallel matvecs.
#include "common.h"

template <int MAXNV, int MAXK>
__launch_bounds__(64) __global__
void proxqp_kernel(const float* __restrict__ Hg, const float* __restrict__ gg,
                   const float* __restrict__ Cg, const float* __restrict__ bg,
                   const float* __restrict__ lg, const float* __restrict__ ug,
                   float* __restrict__ xg, int M, int nv, int k, int iters,
                   float rho0, float sigma, float alpha) {
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
  __shared__ float xbest[MAXNV], red[2];
  __shared__ unsigned char act[MAXC], rel[MAXC];
  __shared__ double Kp[MAXNV][MAXNV];   // f64 polish system
  __shared__ double xp[MAXNV], rp[MAXNV], xc[MAXNV], vb[MAXC];

  const int mc = k + nv;
  const float INF = 3.0e38f;
  float rho = rho0;

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
    for (int i = lane; i < nv; i += 64) {
      float cm = 1e-8f;
      for (int r = 0; r < nv; ++r) cm = fmaxf(cm, fabsf(Hs[r][i]));
      for (int r = 0; r < mc; ++r) cm = fmaxf(cm, fabsf(As[r][i]));
      rhs[i] = rsqrtf(cm);  // dn scratch
    }
    for (int i = lane; i < mc; i += 64) {
      float rm = 1e-8f;
      for (int c = 0; c < nv; ++c) rm = fmaxf(rm, fabsf(As[i][c]));
      zt[i] = rsqrtf(rm);  // de scratch
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
  {  // cost scaling c = 1/max(1, mean colinf(Hs), ||g*D||inf)
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

  // ---- factorization: K = Hs + sigma I + rho A^T A -> Kinv ---------------
  auto factor = [&]() {
    for (int i = lane; i < nv * nv; i += 64) {
      const int r = i / nv, c = i % nv;
      float a = Hs[r][c] + (r == c ? sigma : 0.f);
      for (int t = 0; t < mc; ++t) a += rho * As[t][r] * As[t][c];
      Km[r][c] = a;
    }
    __syncthreads();
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
    for (int c = lane; c < nv; c += 64) {
      for (int i = 0; i < nv; ++i) {
        float a = (i == c) ? 1.f : 0.f;
        for (int t = c; t < i; ++t) a -= Km[i][t] * Linv[t][c];
        Linv[i][c] = (i >= c) ? a / Km[i][i] : 0.f;
      }
    }
    __syncthreads();
    for (int i = lane; i < nv * nv; i += 64) {
      const int r = i / nv, c = i % nv;
      float a = 0.f;
      const int start = (r > c) ? r : c;
      for (int t = start; t < nv; ++t) a += Linv[t][r] * Linv[t][c];
      Kinv[r][c] = a;
    }
    __syncthreads();
  };
  factor();

  // ---- ADMM loop with adaptive rho ---------------------------------------
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

    if ((it + 1) % 25 == 0 && it + 1 < iters) {
      // residual-ratio rho update (OSQP)
      float pri = 0.f, pri_den = 0.f, dua = 0.f, dua_den = 0.f;
      for (int r = lane; r < mc; r += 64) {
        float ax = 0.f;
        for (int j = 0; j < nv; ++j) ax += As[r][j] * xv[j];
        pri = fmaxf(pri, fabsf(ax - zv[r]));
        pri_den = fmaxf(pri_den, fmaxf(fabsf(ax), fabsf(zv[r])));
      }
      for (int i = lane; i < nv; i += 64) {
        float hx = 0.f, aty = 0.f;
        for (int j = 0; j < nv; ++j) hx += Hs[i][j] * xv[j];
        for (int r = 0; r < mc; ++r) aty += As[r][i] * yv[r];
        dua = fmaxf(dua, fabsf(hx + gs[i] + aty));
        dua_den = fmaxf(dua_den, fmaxf(fabsf(hx), fmaxf(fabsf(aty), fabsf(gs[i]))));
      }
      pri = wave_reduce_max(pri) / fmaxf(wave_reduce_max(pri_den), 1e-8f);
      dua = wave_reduce_max(dua) / fmaxf(wave_reduce_max(dua_den), 1e-8f);
      const float ratio = sqrtf(fmaxf(pri, 1e-10f) / fmaxf(dua, 1e-10f));
      rho = fminf(fmaxf(rho * ratio, 1e-6f), 1e6f);
      __syncthreads();
      factor();
    }
  }
  for (int i = lane; i < nv; i += 64) {
    xv[i] *= Dv[i];          // unscale
    xbest[i] = xv[i];
  }
  __syncthreads();

  // ---- merit of the ADMM iterate -----------------------------------------
  // score(x) = obj(x) + 1e6 * sum(violations) on the ORIGINAL problem
  auto score_f32 = [&](const float* xx) -> float {
    float s = 0.f;
    if (lane == 0) {
      for (int i = 0; i < nv; ++i) {
        float hx = 0.f;
        for (int j = 0; j < nv; ++j) hx += Hg[((long)q * nv + i) * nv + j] * xx[j];
        s += 0.5f * xx[i] * hx + gg[(long)q * nv + i] * xx[i];
      }
      for (int r = 0; r < mc; ++r) {
        float ax = 0.f;
        if (r < k) {
          for (int j = 0; j < nv; ++j) ax += Cg[((long)q * k + r) * nv + j] * xx[j];
        } else {
          ax = xx[r - k];
        }
        const float lor = (r < k) ? -INF : lg[(long)q * nv + (r - k)];
        const float hir = (r < k) ? bg[(long)q * k + r] : ug[(long)q * nv + (r - k)];
        s += 1e6f * (fmaxf(ax - hir, 0.f) + fmaxf(lor - ax, 0.f));
      }
    }
    return s;  // only lane 0's value is meaningful
  };
  float best_score = score_f32(xv);

  // ---- iterated f64 active-set polish with release -----------------------
  float ymax = 0.f;
  for (int r = lane; r < mc; r += 64) ymax = fmaxf(ymax, fabsf(yv[r]));
  ymax = wave_reduce_max(ymax);
  for (int r = lane; r < mc; r += 64) rel[r] = 0;
  for (int i = lane; i < nv; i += 64) xc[i] = xv[i];
  __syncthreads();

  for (int pass = 0; pass < 4; ++pass) {
    // activity from the current candidate xc (f64)
    for (int r = lane; r < mc; r += 64) {
      double ax = 0.0;
      if (r < k) {
        for (int j = 0; j < nv; ++j) ax += (double)Cg[((long)q * k + r) * nv + j] * xc[j];
      } else {
        ax = xc[r - k];
      }
      const float lor = (r < k) ? -INF : lg[(long)q * nv + (r - k)];
      const float hir = (r < k) ? bg[(long)q * k + r] : ug[(long)q * nv + (r - k)];
      const bool at_lo = ax <= (double)lor + 1e-4;
      const bool at_hi = ax >= (double)hir - 1e-4;
      bool a = (at_lo || at_hi) && !rel[r];
      if (pass == 0 && fabsf(yv[r]) > 1e-6f * ymax) a = true;
      double v = at_hi ? (double)hir : (double)lor;
      if (!(a && fabs(v) < (double)INF)) v = ax;
      vb[r] = v;
      act[r] = a ? (at_hi ? 2 : 1) : 0;  // 2 = pinned hi, 1 = pinned lo
    }
    __syncthreads();

    const double mu = 1e8;
    for (int i = lane; i < nv * nv; i += 64) {
      const int r = i / nv, c = i % nv;
      double a = (double)Hg[(long)q * nv * nv + i];
      for (int t = 0; t < mc; ++t) {
        if (!act[t]) continue;
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
        if (!act[t]) continue;
        double ar = (t < k) ? (double)Cg[((long)q * k + t) * nv + i] : ((t - k == i) ? 1.0 : 0.0);
        a += mu * ar * vb[t];
      }
      rp[i] = a;
    }
    __syncthreads();
    for (int j = 0; j < nv; ++j) {  // f64 Cholesky
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
    // multiplier-sign release: lambda = mu * (A xp_raw - vb); wrong sign
    // means the pin fights KKT -> drop it next pass
    for (int r = lane; r < mc; r += 64) {
      if (!act[r]) continue;
      double ax = 0.0;
      if (r < k) {
        for (int j = 0; j < nv; ++j) ax += (double)Cg[((long)q * k + r) * nv + j] * xp[j];
      } else {
        ax = xp[r - k];
      }
      const double resid = ax - vb[r];
      if ((act[r] == 2 && resid < -1e-12) || (act[r] == 1 && resid > 1e-12)) rel[r] = 1;
    }
    // project onto the box (penalty leaves g/mu bias on box-active vars)
    for (int i = lane; i < nv; i += 64) {
      const double bl = lg[(long)q * nv + i];
      const double bu = (double)fminf(ug[(long)q * nv + i], INF);
      xp[i] = fmin(fmax(xp[i], bl), bu);
    }
    __syncthreads();
    // score and keep the best candidate
    if (lane == 0) {
      float xx[MAXNV];
      for (int i = 0; i < nv; ++i) xx[i] = (float)xp[i];
      const float sp = score_f32(xx);
      red[0] = (sp <= best_score) ? 1.f : 0.f;
      if (sp <= best_score) best_score = sp;
    }
    __syncthreads();
    if (red[0] > 0.5f)
      for (int i = lane; i < nv; i += 64) xbest[i] = (float)xp[i];
    for (int i = lane; i < nv; i += 64) xc[i] = xp[i];
    __syncthreads();
  }

  for (int i = lane; i < nv; i += 64) xg[(long)q * nv + i] = xbest[i];
}

template __global__ void proxqp_kernel<32, 16>(const float*, const float*, const float*, const float*, const float*, const float*, float*, int, int, int, int, float, float, float);
template __global__ void proxqp_kernel<64, 32>(const float*, const float*, const float*, const float*, const float*, const float*, float*, int, int, int, int, float, float, float);
