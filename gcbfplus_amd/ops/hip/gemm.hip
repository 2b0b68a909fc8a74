// MFMA GEMM kernels for the GNN MLP stack.
//
// The networks are stacks of Dense layers with K, N in {16..384} and M up to
// ~3e5 rows (batch * agents * edge-slots), bf16 inputs / f32 accumulate.
// Kernel family (launcher picks by shape, bindings.hip):
//   gemm_bias_act_kernel      : Y = act(X @ W + b), 128x64 tile, BK=32;
//                               B staged once per block in a fragment-blocked
//                               LDS image, A staged per K-tile. Template
//                               <ACT, BT, ACTIN>: BT consumes W^T through a
//                               transposed B-stage (backward dX without a
//                               transpose copy); ACTIN folds the upstream
//                               activation backward into the A-stage.
//   gemm_bias_act_glds_kernel : BK=64 double-buffered direct-to-LDS A staging
//                               (global_load_lds) for M%128==0, K%64==0.
//   gemm_bias_act_sm_kernel   : 32x64 tile so mid-size M still fills 256 CUs.
//   gemm_bias_act_bn128_kernel: BN=128 experiment (GCBF_GEMM_BN128 env).
//   dot_bias_act_kernel       : N == 1 gate/head outputs, thread-per-row.
//   gemv_bias_act_kernel      : 2 <= N <= 16 outputs, wave-per-row, f32 out.
//   gemm_tn_partial + reduce  : dW = X^T @ dZ, deterministic split-M partials
//                               + fixed-order reduction (no atomics), with
//                               fused db and optional upstream-act fold.
#include "common.h"


// ---------------------------------------------------------------------------
// B-operand staging into the k-blocked LDS image sB[[K/8][BN=64][8]].
// BT=false: W is (K, N) row-major, stage cols n0..n0+63.
// BT=true:  W is the ORIGINAL forward weight (N_out, K) row-major and the
//           GEMM consumes W^T — stage rows n0..n0+63 transposed, so the
//           backward dX = dZ @ W^T needs NO materialized transpose copy.
// ---------------------------------------------------------------------------
template <bool BT>
__device__ __forceinline__ void stage_B_image(const bf16_t* __restrict__ W, bf16_t* sB,
                                              int N, int K, int n0, int tid) {
  constexpr int BN = 64;
  if constexpr (!BT) {
    for (int c = tid; c < K * 8; c += 256) {
      const int k = c >> 3;
      const int co = (c & 7) * 8;
      bf16_t v[8];
      if (n0 + co + 7 < N) {
        *(bf16x8*)v = *(const bf16x8*)(W + (long)k * N + n0 + co);
      } else {
#pragma unroll
        for (int i = 0; i < 8; ++i)
          v[i] = (n0 + co + i < N) ? W[(long)k * N + n0 + co + i] : (bf16_t)0.f;
      }
#pragma unroll
      for (int i = 0; i < 8; ++i) sB[((k >> 3) * BN + (co + i)) * 8 + (k & 7)] = v[i];
    }
  } else {
    const int kch = K >> 3;  // K % 8 == 0 (launcher enforces)
    for (int c = tid; c < BN * kch; c += 256) {
      const int r = c / kch;
      const int kc = (c % kch) * 8;
      bf16_t v[8];
      if (n0 + r < N) {
        *(bf16x8*)v = *(const bf16x8*)(W + (long)(n0 + r) * K + kc);
      } else {
#pragma unroll
        for (int i = 0; i < 8; ++i) v[i] = (bf16_t)0.f;
      }
#pragma unroll
      for (int i = 0; i < 8; ++i)
        sB[(((kc + i) >> 3) * BN + r) * 8 + ((kc + i) & 7)] = v[i];
    }
  }
}

// ---------------------------------------------------------------------------
// Y[M,N] = act(X[M,K] @ W[K,N] + bias[N]);  K % 32 == 0 (launcher pads).
// grid: (ceil(M/128), ceil(N/64)); block: 256 threads (4 waves, 2x2).
// LDS: B-block [K/8][64][8] (k-blocked so B-fragments are single b128 reads)
//      + A-tile [128][40] (pad 32->40 kills b128 bank conflicts).
// ---------------------------------------------------------------------------
template <int ACT, bool BT, int ACTIN>
__launch_bounds__(256) __global__
void gemm_bias_act_kernel(const bf16_t* __restrict__ X, const bf16_t* __restrict__ W,
                          const float* __restrict__ bias, bf16_t* __restrict__ Y,
                          const bf16_t* __restrict__ Xact, int M, int N, int K) {
  constexpr int BM = 128, BN = 64, BK = 32, APAD = 40;
  extern __shared__ char smem[];
  bf16_t* sB = (bf16_t*)smem;                    // [K/8][BN][8]
  bf16_t* sA = (bf16_t*)(smem + (K / 8) * BN * 8 * sizeof(bf16_t));  // [BM][APAD]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const int wm = w >> 1, wn = w & 1;  // wave tile: 64(M) x 32(N)
  const int m0 = blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;

  // ---- preload W block n0..n0+63 into k-blocked LDS image ----------------
  // chunk c -> k = c>>3, col8 = (c&7)*8 ; threads cover K*8 chunks.
  stage_B_image<BT>(W, sB, N, K, n0, tid);

  f32x4 acc[4][2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int arow = tid >> 1;            // A-stage: 2 threads per row
  const int acol = (tid & 1) * 16;      // each stages 16 bf16 = 32 B
  for (int k0 = 0; k0 < K; k0 += BK) {
    __syncthreads();
    // stage A[m0..+128][k0..+32]
    bf16_t av[16];
    const long arow_g = (long)(m0 + arow);
    if (arow_g < M) {
      *(bf16x8*)av = *(const bf16x8*)(X + arow_g * K + k0 + acol);
      *(bf16x8*)(av + 8) = *(const bf16x8*)(X + arow_g * K + k0 + acol + 8);
      if constexpr (ACTIN != 0) {
        // X operand is dY of an activated layer: fold dZ = dY * act'(Y)
        bf16_t yv[16];
        *(bf16x8*)yv = *(const bf16x8*)(Xact + arow_g * K + k0 + acol);
        *(bf16x8*)(yv + 8) = *(const bf16x8*)(Xact + arow_g * K + k0 + acol + 8);
#pragma unroll
        for (int i = 0; i < 16; ++i)
          av[i] = (bf16_t)((float)av[i] * act_grad_from_out((float)yv[i], ACTIN));
      }
    } else {
#pragma unroll
      for (int i = 0; i < 16; ++i) av[i] = (bf16_t)0.f;
    }
#pragma unroll
    for (int i = 0; i < 16; i += 8) *(bf16x8*)(sA + arow * APAD + acol + i) = *(bf16x8*)(av + i);
    __syncthreads();

    // fragments + MFMA
    bf16x8 afr[4];
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
      afr[mf] = *(const bf16x8*)(sA + (wm * 64 + mf * 16 + (lane & 15)) * APAD + (lane >> 4) * 8);
    bf16x8 bfr[2];
#pragma unroll
    for (int nf = 0; nf < 2; ++nf)
      bfr[nf] = *(const bf16x8*)(sB + (((k0 >> 3) + (lane >> 4)) * BN + wn * 32 + nf * 16 + (lane & 15)) * 8);
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
#pragma unroll
      for (int nf = 0; nf < 2; ++nf)
        acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afr[mf], bfr[nf], acc[mf][nf], 0, 0, 0);
  }

  // ---- epilogue: bias + activation + bf16 store --------------------------
#pragma unroll
  for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 2; ++nf) {
      const int col = n0 + wn * 32 + nf * 16 + (lane & 15);
      if (col >= N) continue;
      const float bv = bias[col];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long row = m0 + wm * 64 + mf * 16 + (lane >> 4) * 4 + r;
        if (row < M) Y[row * N + col] = (bf16_t)apply_act(acc[mf][nf][r] + bv, ACT);
      }
    }
  }
}


// ---------------------------------------------------------------------------
// BN=128 variant: halves the A re-read traffic for N=256 layers.
// grid: (ceil(M/128), ceil(N/128)); 4 waves 2x2, wave tile 64x64.
// ---------------------------------------------------------------------------
template <int ACT>
__launch_bounds__(256) __global__
void gemm_bias_act_bn128_kernel(const bf16_t* __restrict__ X, const bf16_t* __restrict__ W,
                                const float* __restrict__ bias, bf16_t* __restrict__ Y,
                                int M, int N, int K) {
  constexpr int BM = 128, BN = 128, BK = 32, APAD = 40;
  extern __shared__ char smem[];
  bf16_t* sB = (bf16_t*)smem;                                        // [K/8][BN][8]
  bf16_t* sA = (bf16_t*)(smem + (K / 8) * BN * 8 * sizeof(bf16_t));  // [BM][APAD]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const int wm = w >> 1, wn = w & 1;  // wave tile: 64(M) x 64(N)
  const int m0 = blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;

  for (int c = tid; c < K * 16; c += 256) {
    const int k = c >> 4;
    const int co = (c & 15) * 8;
    bf16_t v[8];
    if (n0 + co + 7 < N) {
      *(bf16x8*)v = *(const bf16x8*)(W + (long)k * N + n0 + co);
    } else {
#pragma unroll
      for (int i = 0; i < 8; ++i)
        v[i] = (n0 + co + i < N) ? W[(long)k * N + n0 + co + i] : (bf16_t)0.f;
    }
#pragma unroll
    for (int i = 0; i < 8; ++i) sB[((k >> 3) * BN + (co + i)) * 8 + (k & 7)] = v[i];
  }

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int arow = tid >> 1;
  const int acol = (tid & 1) * 16;
  for (int k0 = 0; k0 < K; k0 += BK) {
    __syncthreads();
    bf16_t av[16];
    const long arow_g = (long)(m0 + arow);
    if (arow_g < M) {
      *(bf16x8*)av = *(const bf16x8*)(X + arow_g * K + k0 + acol);
      *(bf16x8*)(av + 8) = *(const bf16x8*)(X + arow_g * K + k0 + acol + 8);
    } else {
#pragma unroll
      for (int i = 0; i < 16; ++i) av[i] = (bf16_t)0.f;
    }
#pragma unroll
    for (int i = 0; i < 16; i += 8) *(bf16x8*)(sA + arow * APAD + acol + i) = *(bf16x8*)(av + i);
    __syncthreads();

    bf16x8 afr[4];
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
      afr[mf] = *(const bf16x8*)(sA + (wm * 64 + mf * 16 + (lane & 15)) * APAD + (lane >> 4) * 8);
    bf16x8 bfr[4];
#pragma unroll
    for (int nf = 0; nf < 4; ++nf)
      bfr[nf] = *(const bf16x8*)(sB + (((k0 >> 3) + (lane >> 4)) * BN + wn * 64 + nf * 16 + (lane & 15)) * 8);
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
#pragma unroll
      for (int nf = 0; nf < 4; ++nf)
        acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afr[mf], bfr[nf], acc[mf][nf], 0, 0, 0);
  }

#pragma unroll
  for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      const int col = n0 + wn * 64 + nf * 16 + (lane & 15);
      if (col >= N) continue;
      const float bv = bias[col];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long row = m0 + wm * 64 + mf * 16 + (lane >> 4) * 4 + r;
        if (row < M) Y[row * N + col] = (bf16_t)apply_act(acc[mf][nf][r] + bv, ACT);
      }
    }
  }
}

template __global__ void gemm_bias_act_bn128_kernel<0>(const bf16_t*, const bf16_t*, const float*, bf16_t*, int, int, int);
template __global__ void gemm_bias_act_bn128_kernel<1>(const bf16_t*, const bf16_t*, const float*, bf16_t*, int, int, int);
template __global__ void gemm_bias_act_bn128_kernel<2>(const bf16_t*, const bf16_t*, const float*, bf16_t*, int, int, int);

// ---------------------------------------------------------------------------
// Small-M variant: BM=32 x BN=64 tile (4 waves as 2x2, wave 16x32) so
// mid-size rows (update/head layers, M ~ 2k) still fill all 256 CUs.
// ---------------------------------------------------------------------------
template <int ACT, bool BT, int ACTIN>
__launch_bounds__(256) __global__
void gemm_bias_act_sm_kernel(const bf16_t* __restrict__ X, const bf16_t* __restrict__ W,
                             const float* __restrict__ bias, bf16_t* __restrict__ Y,
                             const bf16_t* __restrict__ Xact, int M, int N, int K) {
  constexpr int BM = 32, BN = 64, BK = 32, APAD = 40;
  extern __shared__ char smem[];
  bf16_t* sB = (bf16_t*)smem;                                        // [K/8][BN][8]
  bf16_t* sA = (bf16_t*)(smem + (K / 8) * BN * 8 * sizeof(bf16_t));  // [BM][APAD]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const int wm = w >> 1, wn = w & 1;  // wave tile: 16(M) x 32(N)
  const int m0 = blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;

  stage_B_image<BT>(W, sB, N, K, n0, tid);

  f32x4 acc[1][2];
  acc[0][0] = f32x4{0.f, 0.f, 0.f, 0.f};
  acc[0][1] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int arow = tid >> 3;           // 32 rows x 8 threads per row
  const int acol = (tid & 7) * 4;      // 4 bf16 (8 B) per thread
  for (int k0 = 0; k0 < K; k0 += BK) {
    __syncthreads();
    bf16_t av[4];
    const long arow_g = (long)(m0 + arow);
    if (arow_g < M) {
      *(uint2*)av = *(const uint2*)(X + arow_g * K + k0 + acol);
      if constexpr (ACTIN != 0) {
        bf16_t yv[4];
        *(uint2*)yv = *(const uint2*)(Xact + arow_g * K + k0 + acol);
#pragma unroll
        for (int i = 0; i < 4; ++i)
          av[i] = (bf16_t)((float)av[i] * act_grad_from_out((float)yv[i], ACTIN));
      }
    } else {
#pragma unroll
      for (int i = 0; i < 4; ++i) av[i] = (bf16_t)0.f;
    }
    *(uint2*)(sA + arow * APAD + acol) = *(uint2*)av;
    __syncthreads();

    bf16x8 afr = *(const bf16x8*)(sA + (wm * 16 + (lane & 15)) * APAD + (lane >> 4) * 8);
    bf16x8 bfr[2];
#pragma unroll
    for (int nf = 0; nf < 2; ++nf)
      bfr[nf] = *(const bf16x8*)(sB + (((k0 >> 3) + (lane >> 4)) * BN + wn * 32 + nf * 16 + (lane & 15)) * 8);
#pragma unroll
    for (int nf = 0; nf < 2; ++nf)
      acc[0][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afr, bfr[nf], acc[0][nf], 0, 0, 0);
  }

#pragma unroll
  for (int nf = 0; nf < 2; ++nf) {
    const int col = n0 + wn * 32 + nf * 16 + (lane & 15);
    if (col >= N) continue;
    const float bv = bias[col];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const long row = m0 + wm * 16 + (lane >> 4) * 4 + r;
      if (row < M) Y[row * N + col] = (bf16_t)apply_act(acc[0][nf][r] + bv, ACT);
    }
  }
}

template __global__ void gemm_bias_act_sm_kernel<0, false, 0>(const bf16_t*, const bf16_t*, const float*, bf16_t*, const bf16_t*, int, int, int);
template __global__ void gemm_bias_act_sm_kernel<1, false, 0>(const bf16_t*, const bf16_t*, const float*, bf16_t*, const bf16_t*, int, int, int);
template __global__ void gemm_bias_act_sm_kernel<2, false, 0>(const bf16_t*, const bf16_t*, const float*, bf16_t*, const bf16_t*, int, int, int);
template __global__ void gemm_bias_act_sm_kernel<0, true, 0>(const bf16_t*, const bf16_t*, const float*, bf16_t*, const bf16_t*, int, int, int);
template __global__ void gemm_bias_act_sm_kernel<0, true, 1>(const bf16_t*, const bf16_t*, const float*, bf16_t*, const bf16_t*, int, int, int);
template __global__ void gemm_bias_act_sm_kernel<0, true, 2>(const bf16_t*, const bf16_t*, const float*, bf16_t*, const bf16_t*, int, int, int);

// ---------------------------------------------------------------------------
// N == 1 path (attention gate, CBF head): ONE THREAD per output row — the
// wave-per-row gemv dispatches M/4 tiny workgroups and is dispatch-bound at
// M ~ 3e5; here 256 rows share a workgroup and W (<= 384 values, uniform
// across the wave) comes from the scalar cache.
// ---------------------------------------------------------------------------
template <int ACT>
__launch_bounds__(256) __global__
void dot_bias_act_kernel(const bf16_t* __restrict__ X, const bf16_t* __restrict__ W,
                         const float* __restrict__ bias, float* __restrict__ Y,
                         long M, int K) {
  const long m = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (m >= M) return;
  float a0 = 0.f, a1 = 0.f;
  const int kv = ((K & 7) == 0) ? (K / 16) * 16 : 0;  // b128 path needs 16-B
  int k = 0;                                          // aligned rows (K%8==0)
  for (; k < kv; k += 16) {  // two b128 loads in flight per iteration
    const bf16x8 x0 = *(const bf16x8*)(X + m * K + k);
    const bf16x8 x1 = *(const bf16x8*)(X + m * K + k + 8);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      a0 += (float)x0[i] * (float)W[k + i];
      a1 += (float)x1[i] * (float)W[k + 8 + i];
    }
  }
  for (; k < K; ++k) a0 += (float)X[m * K + k] * (float)W[k];
  Y[m] = apply_act(a0 + a1 + bias[0], ACT);
}

template __global__ void dot_bias_act_kernel<0>(const bf16_t*, const bf16_t*, const float*, float*, long, int);
template __global__ void dot_bias_act_kernel<1>(const bf16_t*, const bf16_t*, const float*, float*, long, int);
template __global__ void dot_bias_act_kernel<2>(const bf16_t*, const bf16_t*, const float*, float*, long, int);

// ---------------------------------------------------------------------------
// Small-N path (N <= 16): one wave per output row, W cached in LDS.
// grid: ceil(M/4); block 256 (4 waves).
// ---------------------------------------------------------------------------
template <int ACT, typename OutT>
__launch_bounds__(256) __global__
void gemv_bias_act_kernel(const bf16_t* __restrict__ X, const bf16_t* __restrict__ W,
                          const float* __restrict__ bias, OutT* __restrict__ Y,
                          int M, int N, int K) {
  extern __shared__ char smem[];
  bf16_t* sW = (bf16_t*)smem;  // [K][N]
  for (int i = threadIdx.x; i < K * N; i += 256) sW[i] = W[i];
  __syncthreads();

  const int lane = threadIdx.x & 63;
  const long m = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
  if (m >= M) return;
  float p[16];
#pragma unroll
  for (int n = 0; n < 16; ++n) p[n] = 0.f;
  const int kvec = ((K & 7) == 0) ? K : 0;  // b128 rows need K % 8 == 0
  for (int k8 = lane * 8; k8 < kvec; k8 += WAVE * 8) {  // b128 row loads
    const bf16x8 xv = *(const bf16x8*)(X + m * K + k8);
#pragma unroll
    for (int i = 0; i < 8; ++i)
      for (int n = 0; n < N; ++n) p[n] += (float)xv[i] * (float)sW[(k8 + i) * N + n];
  }
  for (int k = kvec + lane; k < K; k += WAVE) {  // ragged tail
    const float xv = (float)X[m * K + k];
    for (int n = 0; n < N; ++n) p[n] += xv * (float)sW[k * N + n];
  }
  for (int n = 0; n < N; ++n) {
    const float s = wave_reduce_sum(p[n]);
    if (lane == 0) Y[m * N + n] = (OutT)apply_act(s + bias[n], ACT);
  }
}

// ---------------------------------------------------------------------------
// act_bwd: dZ = dY * act'(Y), elementwise bf16.
// ---------------------------------------------------------------------------
__global__ void act_bwd_kernel(const bf16_t* __restrict__ dY, const bf16_t* __restrict__ Y,
                               bf16_t* __restrict__ dZ, long n, int act) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) dZ[i] = (bf16_t)((float)dY[i] * act_grad_from_out((float)Y[i], act));
}

// ---------------------------------------------------------------------------
// dW = X^T @ dZ  (deterministic split-M).  Stage 1: partials[s][K][N].
// grid: (ceil(K/64), ceil(N/64), S); block 256 (4 waves 2x2, wave 32x32).
// A-operand = X^T staged transposed in LDS (scalar transpose writes),
// B-operand = dZ staged k-blocked like the forward GEMM.
// ---------------------------------------------------------------------------
template <int ACT>
__launch_bounds__(256) __global__
void gemm_tn_partial_kernel(const bf16_t* __restrict__ X, const bf16_t* __restrict__ dZ,
                            const bf16_t* __restrict__ Yact,
                            const bool* __restrict__ rowgate,  // null or (M,):
                            // rows with gate=false contribute NOTHING to
                            // dW/db (the dX path elsewhere stays ungated) —
                            // implements per-sample stop-gradient of the
                            // parameters (GCBF+ unlabeled h_dot rows)
                            float* __restrict__ partial, float* __restrict__ db_partial,
                            int M, int N, int K, int S, int remap) {
  // 64x64 output tile, BMR=64 reduction steps with register-prefetch
  // staging (load tile t+1 into registers while tile t computes).
  constexpr int BKDIM = 64, BN = 64, BMR = 64, TPAD = 72;  // pad >= BMR + 8
  __shared__ bf16_t sXT[BKDIM][TPAD];   // [k][m] transposed X tile
  __shared__ bf16_t sB[BMR / 8][BN][8]; // dZ tile, m-blocked
  __shared__ float sDb[4][BN];          // db tree reduce (k0==0 blocks)
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const int wk = w >> 1, wn = w & 1;  // wave tile 32(K) x 32(N)
  // XCD-aware placement (1-D launch, S % 8 == 0): all gk*gn blocks of one
  // split-M slab s get ids with id%8 == s%8 -> same XCD (dispatch places
  // block b on XCD b%8), so the X and dZ slab reads of the 2nd..24th block
  // hit that XCD's L2 instead of re-reading HBM from 6 different XCDs.
  int kb, nb, s;
  if (remap) {
    const int gk = (K + BKDIM - 1) / BKDIM, gn = (N + BN - 1) / BN;
    const int per = gk * gn;
    const int id = blockIdx.x;
    const int rest = id >> 3;
    s = (id & 7) + 8 * (rest / per);
    const int j = rest % per;
    kb = j / gn;
    nb = j % gn;
  } else {
    kb = blockIdx.x;
    nb = blockIdx.y;
    s = blockIdx.z;
  }
  const int k0 = kb * BKDIM;
  const int n0 = nb * BN;

  const long m_per = ((long)M + S - 1) / S;
  const long ms = (long)s * m_per;
  const long me = (ms + m_per < (long)M) ? ms + m_per : (long)M;

  f32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};
  float db_acc = 0.f;
  const int db_c = tid & 63, db_q = tid >> 6;

  // per-thread staging assignments (2 chunks each for X and dZ)
  //   X: chunk c -> row m0 + (c>>3), k-chunk (c&7)*8 (64 rows x 8 chunks)
  //   dZ: chunk c -> row m0 + (c>>3), n-chunk (c&7)*8
  bf16_t xv[2][8], zv[2][8];

  auto load_tile = [&](long m0) {
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      const int c = tid + h * 256;
      const long mr = m0 + (c >> 3);
      const int kc = (c & 7) * 8;
      // b128 loads need 16-B alignment: row base mr*K is aligned only when
      // K % 8 == 0 (misaligned b128 at a segment end page-faults)
      if (mr < me && k0 + kc + 7 < K && (K & 7) == 0) {
        *(bf16x8*)xv[h] = *(const bf16x8*)(X + mr * K + k0 + kc);
      } else {
#pragma unroll
        for (int i = 0; i < 8; ++i)
          xv[h][i] = (mr < me && k0 + kc + i < K) ? X[mr * K + k0 + kc + i] : (bf16_t)0.f;
      }
      const bool gated = rowgate != nullptr && mr < me && !rowgate[mr];
      if (gated) {
#pragma unroll
        for (int i = 0; i < 8; ++i) zv[h][i] = (bf16_t)0.f;
      } else if (mr < me && n0 + kc + 7 < N && (N & 7) == 0) {
        *(bf16x8*)zv[h] = *(const bf16x8*)(dZ + mr * N + n0 + kc);
        if constexpr (ACT != 0) {
          // dZ operand is dY here: fold dZ = dY * act'(Y) into the stage
          bf16x8 yv = *(const bf16x8*)(Yact + mr * N + n0 + kc);
#pragma unroll
          for (int i = 0; i < 8; ++i)
            zv[h][i] = (bf16_t)((float)zv[h][i] * act_grad_from_out((float)yv[i], ACT));
        }
      } else {
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          const bool ok = mr < me && n0 + kc + i < N;
          float z = ok ? (float)dZ[mr * N + n0 + kc + i] : 0.f;
          if constexpr (ACT != 0)
            if (ok) z *= act_grad_from_out((float)Yact[mr * N + n0 + kc + i], ACT);
          zv[h][i] = (bf16_t)z;
        }
      }
    }
  };

  auto write_tile = [&]() {
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      const int c = tid + h * 256;
      const int mr = c >> 3;
      const int kc = (c & 7) * 8;
#pragma unroll
      for (int i = 0; i < 8; ++i) sXT[kc + i][mr] = xv[h][i];
#pragma unroll
      for (int i = 0; i < 8; ++i) sB[mr >> 3][kc + i][mr & 7] = zv[h][i];
    }
  };

  load_tile(ms);
  for (long m0 = ms; m0 < me; m0 += BMR) {
    __syncthreads();
    write_tile();
    __syncthreads();
    if (m0 + BMR < me) load_tile(m0 + BMR);  // overlap with the MFMA below

    if (kb == 0) {
#pragma unroll
      for (int q = 0; q < 2; ++q) {
        const bf16x8 v = *(const bf16x8*)(&sB[db_q * 2 + q][db_c][0]);
#pragma unroll
        for (int i = 0; i < 8; ++i) db_acc += (float)v[i];
      }
    }
#pragma unroll
    for (int mm = 0; mm < 2; ++mm) {  // two 32-deep reduction steps
      bf16x8 afr[2], bfr[2];
#pragma unroll
      for (int kf = 0; kf < 2; ++kf)
        afr[kf] = *(const bf16x8*)(&sXT[wk * 32 + kf * 16 + (lane & 15)][mm * 32 + (lane >> 4) * 8]);
#pragma unroll
      for (int nf = 0; nf < 2; ++nf)
        bfr[nf] = *(const bf16x8*)(&sB[mm * 4 + (lane >> 4)][wn * 32 + nf * 16 + (lane & 15)][0]);
#pragma unroll
      for (int kf = 0; kf < 2; ++kf)
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
          acc[kf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afr[kf], bfr[nf], acc[kf][nf], 0, 0, 0);
    }
  }

  if (kb == 0) {
    // db_acc covers rows {db_q*16..+16} interleaved (q pairs): tree-reduce
    sDb[db_q][db_c] = db_acc;
    __syncthreads();
    if (db_q == 0 && n0 + db_c < N)
      db_partial[(long)s * N + n0 + db_c] =
          (sDb[0][db_c] + sDb[1][db_c]) + (sDb[2][db_c] + sDb[3][db_c]);
  }
  float* out = partial + (long)s * K * N;
#pragma unroll
  for (int kf = 0; kf < 2; ++kf)
#pragma unroll
    for (int nf = 0; nf < 2; ++nf) {
      const int col = n0 + wn * 32 + nf * 16 + (lane & 15);
      if (col >= N) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int krow = k0 + wk * 32 + kf * 16 + (lane >> 4) * 4 + r;
        if (krow < K) out[(long)krow * N + col] = acc[kf][nf][r];
      }
    }
}

template __global__ void gemm_tn_partial_kernel<0>(const bf16_t*, const bf16_t*, const bf16_t*, const bool*, float*, float*, int, int, int, int, int);
template __global__ void gemm_tn_partial_kernel<1>(const bf16_t*, const bf16_t*, const bf16_t*, const bool*, float*, float*, int, int, int, int, int);
template __global__ void gemm_tn_partial_kernel<2>(const bf16_t*, const bf16_t*, const bf16_t*, const bool*, float*, float*, int, int, int, int, int);

// ---------------------------------------------------------------------------
// dW computed TRANSPOSED: dW^T[n,k] = sum_m dZ[m,n] X[m,k]. With i=n, j=k,
// p=m BOTH MFMA operands are m-blocked LDS images ([m/8][col][8]) staged
// with single b128 vector writes — no scalar transpose stores at all (the
// 64x64 kernel above spends its time on 8-way scalar LDS writes for the
// transposed X stage). Global loads are column-wise b16 but lane-coalesced
// (adjacent lanes take adjacent columns). The 16x16 C fragments are written
// back transposed so the partial keeps the (S, K, N) layout.
// Block 64(n) x 64(k), 4 waves 2x2 of 32x32, BMR=64 with register prefetch.
// ---------------------------------------------------------------------------
template <int ACT>
__launch_bounds__(256) __global__
void gemm_tn_partial3_kernel(const bf16_t* __restrict__ X, const bf16_t* __restrict__ dZ,
                             const bf16_t* __restrict__ Yact,
                             const bool* __restrict__ rowgate,
                             float* __restrict__ partial, float* __restrict__ db_partial,
                             int M, int N, int K, int S, int remap) {
  constexpr int BNR = 64, BKD = 64, BMR = 64;
  __shared__ bf16_t sZ[BMR / 8][BNR][8];  // dZ image (A-operand)
  __shared__ bf16_t sX[BMR / 8][BKD][8];  // X image (B-operand)
  __shared__ float sDb[4][BNR];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const int wi = w >> 1, wj = w & 1;  // wave tile 32(n) x 32(k)
  int kb, nb, s;
  if (remap) {
    const int gk = (K + BKD - 1) / BKD, gn = (N + BNR - 1) / BNR;
    const int per = gk * gn;
    const int id = blockIdx.x;
    const int rest = id >> 3;
    s = (id & 7) + 8 * (rest / per);
    const int j = rest % per;
    kb = j / gn;
    nb = j % gn;
  } else {
    kb = blockIdx.x;
    nb = blockIdx.y;
    s = blockIdx.z;
  }
  const int k0 = kb * BKD;
  const int n0 = nb * BNR;

  const long m_per = ((long)M + S - 1) / S;
  const long ms = (long)s * m_per;
  const long me = (ms + m_per < (long)M) ? ms + m_per : (long)M;

  f32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};
  float db_acc = 0.f;
  const int db_c = tid & 63, db_q = tid >> 6;

  // staging, 2 chunks per thread each:
  //   X: column-wise (col = c & 63, m-block = c >> 6): 8 lane-coalesced b16
  //      global loads -> ONE b128 LDS store (X has no act fold).
  //   dZ: row-wise like the fwd kernels (row = c >> 3, 8 cols = (c&7)*8):
  //      b128 dZ (+ b128 Yact for the act fold) -> 8 scalar LDS stores.
  //      Column-wise dZ staging would read Yact as 8 extra b16 loads per
  //      chunk — measured as a net step regression.
  bf16_t xv[2][8], zv[2][8];

  auto load_tile = [&](long m0) {
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      const int c = tid + h * 256;
      const int col = c & 63;
      const long mr0 = m0 + (c >> 6) * 8;
      const bool kin = k0 + col < K;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const long mr = mr0 + i;
        xv[h][i] = (mr < me && kin) ? X[mr * K + k0 + col] : (bf16_t)0.f;
      }
      const long zr = m0 + (c >> 3);
      const int nc = (c & 7) * 8;
      const bool zgated = rowgate != nullptr && zr < me && !rowgate[zr];
      if (zgated) {
#pragma unroll
        for (int i = 0; i < 8; ++i) zv[h][i] = (bf16_t)0.f;
      } else if (zr < me && n0 + nc + 7 < N && (N & 7) == 0) {
        *(bf16x8*)zv[h] = *(const bf16x8*)(dZ + zr * N + n0 + nc);
        if constexpr (ACT != 0) {
          bf16x8 yv = *(const bf16x8*)(Yact + zr * N + n0 + nc);
#pragma unroll
          for (int i = 0; i < 8; ++i)
            zv[h][i] = (bf16_t)((float)zv[h][i] * act_grad_from_out((float)yv[i], ACT));
        }
      } else {
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          const bool ok = zr < me && n0 + nc + i < N;
          float z = ok ? (float)dZ[zr * N + n0 + nc + i] : 0.f;
          if constexpr (ACT != 0)
            if (ok) z *= act_grad_from_out((float)Yact[zr * N + n0 + nc + i], ACT);
          zv[h][i] = (bf16_t)z;
        }
      }
    }
  };

  auto write_tile = [&]() {
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      const int c = tid + h * 256;
      const int col = c & 63;
      const int mb = c >> 6;
      *(bf16x8*)(&sX[mb][col][0]) = *(bf16x8*)xv[h];
      const int zr = c >> 3;
      const int nc = (c & 7) * 8;
#pragma unroll
      for (int i = 0; i < 8; ++i) sZ[zr >> 3][nc + i][zr & 7] = zv[h][i];
    }
  };

  load_tile(ms);
  for (long m0 = ms; m0 < me; m0 += BMR) {
    __syncthreads();
    write_tile();
    __syncthreads();
    if (m0 + BMR < me) load_tile(m0 + BMR);  // overlap with the MFMA below

    if (kb == 0) {
#pragma unroll
      for (int q = 0; q < 2; ++q) {
        const bf16x8 v = *(const bf16x8*)(&sZ[db_q * 2 + q][db_c][0]);
#pragma unroll
        for (int i = 0; i < 8; ++i) db_acc += (float)v[i];
      }
    }
#pragma unroll
    for (int mm = 0; mm < 2; ++mm) {  // two 32-deep reduction substeps
      bf16x8 afr[2], bfr[2];
#pragma unroll
      for (int nf = 0; nf < 2; ++nf)
        afr[nf] = *(const bf16x8*)(&sZ[mm * 4 + (lane >> 4)][wi * 32 + nf * 16 + (lane & 15)][0]);
#pragma unroll
      for (int kf = 0; kf < 2; ++kf)
        bfr[kf] = *(const bf16x8*)(&sX[mm * 4 + (lane >> 4)][wj * 32 + kf * 16 + (lane & 15)][0]);
#pragma unroll
      for (int nf = 0; nf < 2; ++nf)
#pragma unroll
        for (int kf = 0; kf < 2; ++kf)
          acc[nf][kf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afr[nf], bfr[kf], acc[nf][kf], 0, 0, 0);
    }
  }

  if (kb == 0) {
    sDb[db_q][db_c] = db_acc;
    __syncthreads();
    if (db_q == 0 && n0 + db_c < N)
      db_partial[(long)s * N + n0 + db_c] =
          (sDb[0][db_c] + sDb[1][db_c]) + (sDb[2][db_c] + sDb[3][db_c]);
  }

  // C frag: lane holds C[row = n-frag][col = k-frag]; write transposed into
  // partial[s][k][n] (4-float runs are contiguous in n? no: row=n varies by
  // r -> stride 1 writes along n for fixed k = per-lane column)
  float* out = partial + (long)s * K * N;
#pragma unroll
  for (int nf = 0; nf < 2; ++nf) {
#pragma unroll
    for (int kf = 0; kf < 2; ++kf) {
      const int krow = k0 + wj * 32 + kf * 16 + (lane & 15);
      if (krow >= K) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int ncol = n0 + wi * 32 + nf * 16 + (lane >> 4) * 4 + r;
        if (ncol < N) out[(long)krow * N + ncol] = acc[nf][kf][r];
      }
    }
  }
}

template __global__ void gemm_tn_partial3_kernel<0>(const bf16_t*, const bf16_t*, const bf16_t*, const bool*, float*, float*, int, int, int, int, int);
template __global__ void gemm_tn_partial3_kernel<1>(const bf16_t*, const bf16_t*, const bf16_t*, const bool*, float*, float*, int, int, int, int, int);
template __global__ void gemm_tn_partial3_kernel<2>(const bf16_t*, const bf16_t*, const bf16_t*, const bool*, float*, float*, int, int, int, int, int);

// ---------------------------------------------------------------------------
// dW^T orientation at 128x128 block / 64x64 wave tile: same vector-only
// staging as kernel3, 4x fragment reuse (0.5 LDS loads per MFMA).
// Requires K >= 128 and N >= 128 (guards handle ragged edges).
// ---------------------------------------------------------------------------
template <int ACT>
__launch_bounds__(256) __global__
void gemm_tn_partial4_kernel(const bf16_t* __restrict__ X, const bf16_t* __restrict__ dZ,
                             const bf16_t* __restrict__ Yact,
                             float* __restrict__ partial, float* __restrict__ db_partial,
                             int M, int N, int K, int S, int remap) {
  constexpr int BNR = 128, BKD = 128, BMR = 64;
  __shared__ bf16_t sZ[BMR / 8][BNR][8];  // 16 KB
  __shared__ bf16_t sX[BMR / 8][BKD][8];  // 16 KB
  __shared__ float sDb[2][BNR];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const int wi = w >> 1, wj = w & 1;  // wave tile 64(n) x 64(k)
  int kb, nb, s;
  if (remap) {
    const int gk = (K + BKD - 1) / BKD, gn = (N + BNR - 1) / BNR;
    const int per = gk * gn;
    const int id = blockIdx.x;
    const int rest = id >> 3;
    s = (id & 7) + 8 * (rest / per);
    const int j = rest % per;
    kb = j / gn;
    nb = j % gn;
  } else {
    kb = blockIdx.x;
    nb = blockIdx.y;
    s = blockIdx.z;
  }
  const int k0 = kb * BKD;
  const int n0 = nb * BNR;

  const long m_per = ((long)M + S - 1) / S;
  const long ms = (long)s * m_per;
  const long me = (ms + m_per < (long)M) ? ms + m_per : (long)M;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};
  float db_acc = 0.f;
  const int db_c = tid & 127, db_h = tid >> 7;

  // staging: 1024 col-chunks each (128 cols x 8 m-blocks); 4 per thread
  bf16_t xv[4][8], zv[4][8];

  auto load_tile = [&](long m0) {
#pragma unroll
    for (int h = 0; h < 4; ++h) {
      const int c = tid + h * 256;
      const int col = c & 127;
      const long mr0 = m0 + (c >> 7) * 8;
      const bool kin = k0 + col < K;
      const bool nin = n0 + col < N;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const long mr = mr0 + i;
        xv[h][i] = (mr < me && kin) ? X[mr * K + k0 + col] : (bf16_t)0.f;
        float z = (mr < me && nin) ? (float)dZ[mr * N + n0 + col] : 0.f;
        if constexpr (ACT != 0)
          if (mr < me && nin) z *= act_grad_from_out((float)Yact[mr * N + n0 + col], ACT);
        zv[h][i] = (bf16_t)z;
      }
    }
  };

  auto write_tile = [&]() {
#pragma unroll
    for (int h = 0; h < 4; ++h) {
      const int c = tid + h * 256;
      const int col = c & 127;
      const int mb = c >> 7;
      *(bf16x8*)(&sX[mb][col][0]) = *(bf16x8*)xv[h];
      *(bf16x8*)(&sZ[mb][col][0]) = *(bf16x8*)zv[h];
    }
  };

  load_tile(ms);
  for (long m0 = ms; m0 < me; m0 += BMR) {
    __syncthreads();
    write_tile();
    __syncthreads();
    if (m0 + BMR < me) load_tile(m0 + BMR);

    if (kb == 0) {
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        const bf16x8 v = *(const bf16x8*)(&sZ[db_h * 4 + q][db_c][0]);
#pragma unroll
        for (int i = 0; i < 8; ++i) db_acc += (float)v[i];
      }
    }
#pragma unroll
    for (int mm = 0; mm < 2; ++mm) {
      bf16x8 afr[4], bfr[4];
#pragma unroll
      for (int nf = 0; nf < 4; ++nf)
        afr[nf] = *(const bf16x8*)(&sZ[mm * 4 + (lane >> 4)][wi * 64 + nf * 16 + (lane & 15)][0]);
#pragma unroll
      for (int kf = 0; kf < 4; ++kf)
        bfr[kf] = *(const bf16x8*)(&sX[mm * 4 + (lane >> 4)][wj * 64 + kf * 16 + (lane & 15)][0]);
#pragma unroll
      for (int nf = 0; nf < 4; ++nf)
#pragma unroll
        for (int kf = 0; kf < 4; ++kf)
          acc[nf][kf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afr[nf], bfr[kf], acc[nf][kf], 0, 0, 0);
    }
  }

  if (kb == 0) {
    sDb[db_h][db_c] = db_acc;
    __syncthreads();
    if (db_h == 0 && n0 + db_c < N)
      db_partial[(long)s * N + n0 + db_c] = sDb[0][db_c] + sDb[1][db_c];
  }

  float* out = partial + (long)s * K * N;
#pragma unroll
  for (int nf = 0; nf < 4; ++nf) {
#pragma unroll
    for (int kf = 0; kf < 4; ++kf) {
      const int krow = k0 + wj * 64 + kf * 16 + (lane & 15);
      if (krow >= K) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int ncol = n0 + wi * 64 + nf * 16 + (lane >> 4) * 4 + r;
        if (ncol < N) out[(long)krow * N + ncol] = acc[nf][kf][r];
      }
    }
  }
}

template __global__ void gemm_tn_partial4_kernel<0>(const bf16_t*, const bf16_t*, const bf16_t*, float*, float*, int, int, int, int, int);
template __global__ void gemm_tn_partial4_kernel<1>(const bf16_t*, const bf16_t*, const bf16_t*, float*, float*, int, int, int, int, int);
template __global__ void gemm_tn_partial4_kernel<2>(const bf16_t*, const bf16_t*, const bf16_t*, float*, float*, int, int, int, int, int);

// ---------------------------------------------------------------------------
// dW = X^T dZ, 128x128 output tile / 64x64 wave tile (K >= 128, N >= 128).
// The 64x64 kernel above is LDS-load bound (1 b128 fragment read per MFMA,
// measured 121 TF); the 64x64 WAVE tile reuses each fragment 4x -> 0.5
// loads/MFMA. Same split-M partial + fused-db scheme, same XCD-aware 1-D
// placement decode.
// ---------------------------------------------------------------------------
template <int ACT>
__launch_bounds__(256) __global__
void gemm_tn_partial2_kernel(const bf16_t* __restrict__ X, const bf16_t* __restrict__ dZ,
                             const bf16_t* __restrict__ Yact,
                             float* __restrict__ partial, float* __restrict__ db_partial,
                             int M, int N, int K, int S, int remap) {
  constexpr int BK2 = 128, BN2 = 128, BMR = 64, TPAD = 72;
  __shared__ bf16_t sXT[BK2][TPAD];      // [k][m] transposed X tile (18.4 KB)
  __shared__ bf16_t sB[BMR / 8][BN2][8]; // dZ tile, m-blocked (16.4 KB)
  __shared__ float sDb[2][BN2];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const int wk = w >> 1, wn = w & 1;  // wave tile 64(K) x 64(N)
  int kb, nb, s;
  if (remap) {
    const int gk = (K + BK2 - 1) / BK2, gn = (N + BN2 - 1) / BN2;
    const int per = gk * gn;
    const int id = blockIdx.x;
    const int rest = id >> 3;
    s = (id & 7) + 8 * (rest / per);
    const int j = rest % per;
    kb = j / gn;
    nb = j % gn;
  } else {
    kb = blockIdx.x;
    nb = blockIdx.y;
    s = blockIdx.z;
  }
  const int k0 = kb * BK2;
  const int n0 = nb * BN2;

  const long m_per = ((long)M + S - 1) / S;
  const long ms = (long)s * m_per;
  const long me = (ms + m_per < (long)M) ? ms + m_per : (long)M;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};
  float db_acc = 0.f;
  const int db_c = tid & 127, db_h = tid >> 7;

  // staging: 1024 8-elem chunks each for X and dZ -> 4 chunks per thread
  //   chunk c: local row mr = c >> 4 (64 rows), 8-col group (c & 15) * 8
  bf16_t xv[4][8], zv[4][8];

  auto load_tile = [&](long m0) {
#pragma unroll
    for (int h = 0; h < 4; ++h) {
      const int c = tid + h * 256;
      const long mr = m0 + (c >> 4);
      const int kc = (c & 15) * 8;
      if (mr < me && k0 + kc + 7 < K && (K & 7) == 0) {
        *(bf16x8*)xv[h] = *(const bf16x8*)(X + mr * K + k0 + kc);
      } else {
#pragma unroll
        for (int i = 0; i < 8; ++i)
          xv[h][i] = (mr < me && k0 + kc + i < K) ? X[mr * K + k0 + kc + i] : (bf16_t)0.f;
      }
      if (mr < me && n0 + kc + 7 < N && (N & 7) == 0) {
        *(bf16x8*)zv[h] = *(const bf16x8*)(dZ + mr * N + n0 + kc);
        if constexpr (ACT != 0) {
          bf16x8 yv = *(const bf16x8*)(Yact + mr * N + n0 + kc);
#pragma unroll
          for (int i = 0; i < 8; ++i)
            zv[h][i] = (bf16_t)((float)zv[h][i] * act_grad_from_out((float)yv[i], ACT));
        }
      } else {
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          const bool ok = mr < me && n0 + kc + i < N;
          float z = ok ? (float)dZ[mr * N + n0 + kc + i] : 0.f;
          if constexpr (ACT != 0)
            if (ok) z *= act_grad_from_out((float)Yact[mr * N + n0 + kc + i], ACT);
          zv[h][i] = (bf16_t)z;
        }
      }
    }
  };

  auto write_tile = [&]() {
#pragma unroll
    for (int h = 0; h < 4; ++h) {
      const int c = tid + h * 256;
      const int mr = c >> 4;
      const int kc = (c & 15) * 8;
#pragma unroll
      for (int i = 0; i < 8; ++i) sXT[kc + i][mr] = xv[h][i];
#pragma unroll
      for (int i = 0; i < 8; ++i) sB[mr >> 3][kc + i][mr & 7] = zv[h][i];
    }
  };

  load_tile(ms);
  for (long m0 = ms; m0 < me; m0 += BMR) {
    __syncthreads();
    write_tile();
    __syncthreads();
    if (m0 + BMR < me) load_tile(m0 + BMR);  // overlap with the MFMA below

    if (kb == 0) {
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        const bf16x8 v = *(const bf16x8*)(&sB[db_h * 4 + q][db_c][0]);
#pragma unroll
        for (int i = 0; i < 8; ++i) db_acc += (float)v[i];
      }
    }
#pragma unroll
    for (int mm = 0; mm < 2; ++mm) {  // two 32-deep reduction substeps
      bf16x8 afr[4], bfr[4];
#pragma unroll
      for (int kf = 0; kf < 4; ++kf)
        afr[kf] = *(const bf16x8*)(&sXT[wk * 64 + kf * 16 + (lane & 15)][mm * 32 + (lane >> 4) * 8]);
#pragma unroll
      for (int nf = 0; nf < 4; ++nf)
        bfr[nf] = *(const bf16x8*)(&sB[mm * 4 + (lane >> 4)][wn * 64 + nf * 16 + (lane & 15)][0]);
#pragma unroll
      for (int kf = 0; kf < 4; ++kf)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
          acc[kf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afr[kf], bfr[nf], acc[kf][nf], 0, 0, 0);
    }
  }

  if (kb == 0) {
    sDb[db_h][db_c] = db_acc;
    __syncthreads();
    if (db_h == 0 && n0 + db_c < N)
      db_partial[(long)s * N + n0 + db_c] = sDb[0][db_c] + sDb[1][db_c];
  }

  float* out = partial + (long)s * K * N;
#pragma unroll
  for (int kf = 0; kf < 4; ++kf) {
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      const int col = n0 + wn * 64 + nf * 16 + (lane & 15);
      if (col >= N) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int krow = k0 + wk * 64 + kf * 16 + (lane >> 4) * 4 + r;
        if (krow < K) out[(long)krow * N + col] = acc[kf][nf][r];
      }
    }
  }
}

template __global__ void gemm_tn_partial2_kernel<0>(const bf16_t*, const bf16_t*, const bf16_t*, float*, float*, int, int, int, int, int);
template __global__ void gemm_tn_partial2_kernel<1>(const bf16_t*, const bf16_t*, const bf16_t*, float*, float*, int, int, int, int, int);
template __global__ void gemm_tn_partial2_kernel<2>(const bf16_t*, const bf16_t*, const bf16_t*, float*, float*, int, int, int, int, int);

// (dw_partial (S,K,N), db_partial (S,N)) -> (dW, db) in ONE launch
// (fixed-order sums: deterministic; 4 accumulators hide add latency).
// acc != 0 accumulates (+=) into dW/db — used to write straight into the
// optimizer's flat-gradient views, skipping autograd's AccumulateGrad adds.
__global__ void reduce_dw_db_kernel(const float* __restrict__ pw, const float* __restrict__ pb,
                                    float* __restrict__ dW, float* __restrict__ db,
                                    float* __restrict__ db2,
                                    long KN, int N, int S, int acc) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const bool is_db = i >= KN;
  if (i >= KN + N) return;
  const float* src = is_db ? pb : pw;
  const long stride = is_db ? N : KN;
  const long off = is_db ? i - KN : i;
  float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
  int s = 0;
  for (; s + 4 <= S; s += 4) {
    a0 += src[(long)s * stride + off];
    a1 += src[(long)(s + 1) * stride + off];
    a2 += src[(long)(s + 2) * stride + off];
    a3 += src[(long)(s + 3) * stride + off];
  }
  for (; s < S; ++s) a0 += src[(long)s * stride + off];
  const float r = (a0 + a1) + (a2 + a3);
  float* dst = is_db ? db + off : dW + off;
  *dst = acc ? (*dst + r) : r;
  // optional second db accumulation target (one-hot bias fold: the same db
  // flows into both bias.grad and kernel.grad[oh_row])
  if (is_db && db2 != nullptr) db2[off] += r;
}

// ---------------------------------------------------------------------------
// host-visible launchers (called from bindings.cpp)
// ---------------------------------------------------------------------------
template __global__ void gemm_bias_act_kernel<0, false, 0>(const bf16_t*, const bf16_t*, const float*, bf16_t*, const bf16_t*, int, int, int);
template __global__ void gemm_bias_act_kernel<1, false, 0>(const bf16_t*, const bf16_t*, const float*, bf16_t*, const bf16_t*, int, int, int);
template __global__ void gemm_bias_act_kernel<2, false, 0>(const bf16_t*, const bf16_t*, const float*, bf16_t*, const bf16_t*, int, int, int);
template __global__ void gemm_bias_act_kernel<0, true, 0>(const bf16_t*, const bf16_t*, const float*, bf16_t*, const bf16_t*, int, int, int);
template __global__ void gemm_bias_act_kernel<0, true, 1>(const bf16_t*, const bf16_t*, const float*, bf16_t*, const bf16_t*, int, int, int);
template __global__ void gemm_bias_act_kernel<0, true, 2>(const bf16_t*, const bf16_t*, const float*, bf16_t*, const bf16_t*, int, int, int);
template __global__ void gemv_bias_act_kernel<0, bf16_t>(const bf16_t*, const bf16_t*, const float*, bf16_t*, int, int, int);
template __global__ void gemv_bias_act_kernel<1, bf16_t>(const bf16_t*, const bf16_t*, const float*, bf16_t*, int, int, int);
template __global__ void gemv_bias_act_kernel<2, bf16_t>(const bf16_t*, const bf16_t*, const float*, bf16_t*, int, int, int);
template __global__ void gemv_bias_act_kernel<0, float>(const bf16_t*, const bf16_t*, const float*, float*, int, int, int);
template __global__ void gemv_bias_act_kernel<1, float>(const bf16_t*, const bf16_t*, const float*, float*, int, int, int);
template __global__ void gemv_bias_act_kernel<2, float>(const bf16_t*, const bf16_t*, const float*, float*, int, int, int);

// ---------------------------------------------------------------------------
// glds-pipelined main GEMM: BK=64, double-buffered A staged by
// global_load_lds (16 B per lane), XOR-swizzled LDS image so the b128
// fragment reads stay bank-conflict-free (granule g' = g ^ ((row>>1)&7) on
// the [128][64] bf16 tile). Requires M % 128 == 0 and K % 64 == 0 (glds has
// no per-lane predication); the launcher falls back otherwise.
// ---------------------------------------------------------------------------
template <int ACT, bool BT>
__launch_bounds__(256) __global__
void gemm_bias_act_glds_kernel(const bf16_t* __restrict__ X, const bf16_t* __restrict__ W,
                               const float* __restrict__ bias, bf16_t* __restrict__ Y,
                               int M, int N, int K) {
  constexpr int BM = 128, BN = 64, BK = 64;
  extern __shared__ char smem[];
  bf16_t* sB = (bf16_t*)smem;                                    // [K/8][BN][8]
  bf16_t* sA = (bf16_t*)(smem + (K / 8) * BN * 8 * sizeof(bf16_t));  // [2][128][64]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const int wm = w >> 1, wn = w & 1;  // wave tile: 64(M) x 32(N)
  const int m0 = blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;

  // B preload (same image as the base kernel)
  stage_B_image<BT>(W, sB, N, K, n0, tid);

  // glds stage of one A tile into buffer `buf`: 16 wave-chunks of 1 KiB
  auto stage_A = [&](int k0, int buf) {
    bf16_t* dstbase = sA + buf * 128 * 64;
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int idx = (it * 4 + w) * 64 + lane;  // 16-byte granule index
      const int row = idx >> 3;
      const int g = idx & 7;
      const int gsrc = g ^ ((row >> 1) & 7);     // pre-swizzled source
      const bf16_t* src = X + (long)(m0 + row) * K + k0 + gsrc * 8;
      __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) uint32_t*)src,
                                       (__attribute__((address_space(3))) uint32_t*)(dstbase + (long)(it * 4 + w) * 512),
                                       16, 0, 0);
    }
  };

  f32x4 acc[4][2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  stage_A(0, 0);
  const int ntiles = K / BK;
  for (int t = 0; t < ntiles; ++t) {
    __syncthreads();  // drains the in-flight glds for buffer t&1
    if (t + 1 < ntiles) stage_A((t + 1) * BK, (t + 1) & 1);
    const bf16_t* buf = sA + (t & 1) * 128 * 64;
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {  // two 16x16x32 K-steps per tile
      bf16x8 afr[4];
#pragma unroll
      for (int mf = 0; mf < 4; ++mf) {
        const int row = wm * 64 + mf * 16 + (lane & 15);
        const int g = kk * 4 + (lane >> 4);
        const int gs = g ^ ((row >> 1) & 7);
        afr[mf] = *(const bf16x8*)(buf + (long)row * 64 + gs * 8);
      }
      bf16x8 bfr[2];
      const int kb = t * BK + kk * 32;
#pragma unroll
      for (int nf = 0; nf < 2; ++nf)
        bfr[nf] = *(const bf16x8*)(sB + (((kb >> 3) + (lane >> 4)) * BN + wn * 32 + nf * 16 + (lane & 15)) * 8);
#pragma unroll
      for (int mf = 0; mf < 4; ++mf)
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afr[mf], bfr[nf], acc[mf][nf], 0, 0, 0);
    }
  }

#pragma unroll
  for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 2; ++nf) {
      const int col = n0 + wn * 32 + nf * 16 + (lane & 15);
      if (col >= N) continue;
      const float bv = bias[col];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long row = m0 + wm * 64 + mf * 16 + (lane >> 4) * 4 + r;
        Y[row * N + col] = (bf16_t)apply_act(acc[mf][nf][r] + bv, ACT);
      }
    }
  }
}

template __global__ void gemm_bias_act_glds_kernel<0, false>(const bf16_t*, const bf16_t*, const float*, bf16_t*, int, int, int);
template __global__ void gemm_bias_act_glds_kernel<1, false>(const bf16_t*, const bf16_t*, const float*, bf16_t*, int, int, int);
template __global__ void gemm_bias_act_glds_kernel<2, false>(const bf16_t*, const bf16_t*, const float*, bf16_t*, int, int, int);
template __global__ void gemm_bias_act_glds_kernel<0, true>(const bf16_t*, const bf16_t*, const float*, bf16_t*, int, int, int);
