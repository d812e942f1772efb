// Linear fwd/bwd MFMA GEMM kernels, fp32 path, skinny-N
// aware (LeNet fc1: [M,320]x[320,50], fc2: [M,50]x[50,10]).
//
//   fwd : y[m][co] = x[m][ci] . w[co][ci] + b[co]      (optional ReLU epilogue)
//   dx  : dx[m][ci] = dyr[m][co] . w[co][ci]           (dyr = relu-masked dy)
//   dw  : dw[co][ci] = sum_m dyr[m][co] x[m][ci]  (+ db = colsum(dyr))
//
// fwd/dx stage both operand chunks through LDS transposed so every global
// access group is contiguous; dw needs no LDS at all (both gathers are
// row-major coalesced), split-M partial slabs folded deterministically.
#include "common.h"

// ---------------------------------------------------------------------------
// Forward: D[m][co]; block = 4 waves x 16 m-rows = BM 64, BN 64 co columns.
// ---------------------------------------------------------------------------
template <bool HAS_BIAS, bool RELU, int BN = 64>
__global__ __launch_bounds__(256) void linear_fwd_f32_k(
    const float* __restrict__ x, const float* __restrict__ w,
    const float* __restrict__ bias, float* __restrict__ y,
    int M, int CI, int CO) {
  // BN=16 variant: skinny-CO shapes (LeNet fc1 CO=50, M=8192) give only
  // 128 blocks at BN=64 — half the CUs idle; 16-wide co tiles restore the
  // grid (x re-reads stay L2-hot)
  constexpr int BM = 64, BK = 32;
  const int m0 = blockIdx.x * BM;
  const int co0 = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lm = l & 15;
  const int lk = l >> 4;

  __shared__ float lds_x[BK][BM + 1];  // [ci][m]
  __shared__ float lds_w[BK][BN + 1];  // [ci][co]

  f32x4 acc[BN / 16] = {};

  for (int k0 = 0; k0 < CI; k0 += BK) {
    __syncthreads();
    // stage x[m0:m0+64][k0:k0+32] -> lds_x[ci][m]; reads coalesced over ci
    for (int idx = tid; idx < BM * BK; idx += 256) {
      int mloc = idx >> 5, ci = idx & 31;  // ci fastest
      float v = 0.f;
      if (m0 + mloc < M && k0 + ci < CI) v = x[(m0 + mloc) * CI + k0 + ci];
      lds_x[ci][mloc] = v;
    }
    // stage w[co0:co0+64][k0:k0+32] -> lds_w[ci][co]
    for (int idx = tid; idx < BN * BK; idx += 256) {
      int coloc = idx >> 5, ci = idx & 31;
      float v = 0.f;
      if (co0 + coloc < CO && k0 + ci < CI) v = w[(co0 + coloc) * CI + k0 + ci];
      lds_w[ci][coloc] = v;
    }
    __syncthreads();

    int klim = min(BK, CI - k0);
    for (int kk = 0; kk < klim; kk += 4) {
      bool kin = lk < (klim - kk);
      float a = kin ? lds_x[kk + lk][wid * 16 + lm] : 0.f;  // A[row=m][k=ci]
#pragma unroll
      for (int ni = 0; ni < BN / 16; ++ni) {
        float b = kin ? lds_w[kk + lk][ni * 16 + lm] : 0.f;  // B[k][col=co]
        acc[ni] = mfma16x16x4(a, b, acc[ni]);
      }
    }
  }

  // D: row=(lk*4+reg) -> m-local, col=lm -> co; stores contiguous over co
#pragma unroll
  for (int ni = 0; ni < BN / 16; ++ni) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int m = m0 + wid * 16 + lk * 4 + reg;
      int co = co0 + ni * 16 + lm;
      if (m < M && co < CO) {
        float v = acc[ni][reg];
        if (HAS_BIAS) v += bias[co];
        if (RELU) v = fmaxf(v, 0.f);
        y[m * CO + co] = v;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// dx: D[m][ci] = dyr[m][co] . w[co][ci]; dyr staged (mask applied), w read
// straight from global (row-major over ci => coalesced, L2-resident).
// ---------------------------------------------------------------------------
template <bool RELU>
__global__ __launch_bounds__(256) void linear_dx_f32_k(
    const float* __restrict__ dy, const float* __restrict__ w,
    const float* __restrict__ yact, float* __restrict__ dx,
    int M, int CI, int CO) {
  constexpr int BM = 64, BN = 64, BK = 16;
  const int m0 = blockIdx.x * BM;
  const int ci0 = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lm = l & 15;
  const int lk = l >> 4;

  __shared__ float lds_dy[BK][BM + 1];  // [co][m], relu-masked

  f32x4 acc[BN / 16] = {};

  for (int k0 = 0; k0 < CO; k0 += BK) {
    __syncthreads();
    for (int idx = tid; idx < BM * BK; idx += 256) {
      int mloc = idx >> 4, co = idx & 15;
      float v = 0.f;
      int m = m0 + mloc;
      if (m < M && k0 + co < CO) {
        v = dy[m * CO + k0 + co];
        if (RELU && yact[m * CO + k0 + co] <= 0.f) v = 0.f;
      }
      lds_dy[co][mloc] = v;
    }
    __syncthreads();

    int klim = min(BK, CO - k0);
    for (int kk = 0; kk < klim; kk += 4) {
      bool kin = lk < (klim - kk);
      float a = kin ? lds_dy[kk + lk][wid * 16 + lm] : 0.f;  // A[row=m][k=co]
#pragma unroll
      for (int ni = 0; ni < BN / 16; ++ni) {
        int ci = ci0 + ni * 16 + lm;
        float b = 0.f;
        if (kin && ci < CI) b = w[(k0 + kk + lk) * CI + ci];  // B[k=co][col=ci]
        acc[ni] = mfma16x16x4(a, b, acc[ni]);
      }
    }
  }

#pragma unroll
  for (int ni = 0; ni < BN / 16; ++ni) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int m = m0 + wid * 16 + lk * 4 + reg;
      int ci = ci0 + ni * 16 + lm;
      if (m < M && ci < CI) dx[m * CI + ci] = acc[ni][reg];
    }
  }
}

// ---------------------------------------------------------------------------
// dw (+db): D[co][ci]; A = dyr^T gather (16 consecutive co), B = x gather
// (16 consecutive ci) — both coalesced row-major; reduce over an M split into
// per-split slabs (deterministic fold). Waves split the ci tile.
// ---------------------------------------------------------------------------
template <bool RELU, bool WANT_BIAS>
__global__ __launch_bounds__(256) void linear_dw_f32_k(
    const float* __restrict__ dy, const float* __restrict__ x,
    const float* __restrict__ yact, float* __restrict__ dw,
    float* __restrict__ db, int M, int CI, int CO, int m_per_split) {
  const int co0 = blockIdx.x * 16;
  const int ci0 = blockIdx.y * 64;  // 4 waves x 16 ci
  const int ms = blockIdx.z;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lm = l & 15;
  const int lk = l >> 4;

  const int ci = ci0 + wid * 16 + lm;
  const int co = co0 + lm;
  f32x4 acc = {};
  float db_part = 0.f;

  int m_begin = ms * m_per_split;
  int m_end = min(M, m_begin + m_per_split);

  for (int mm = m_begin; mm < m_end; mm += 4) {
    int m = mm + lk;
    bool mok = m < m_end;
    int msel = mok ? m : m_begin;
    float a = 0.f;  // A[row=co][k=m] = dyr[m][co]
    if (mok && co < CO) {
      a = dy[msel * CO + co];
      if (RELU && yact[msel * CO + co] <= 0.f) a = 0.f;
    }
    float b = 0.f;  // B[k=m][col=ci] = x[m][ci]
    if (mok && ci < CI) b = x[msel * CI + ci];
    acc = mfma16x16x4(a, b, acc);
    if (WANT_BIAS) db_part += a;
  }

  // deterministic partial slabs (dw points at workspace [ms][CO*CI]);
  // reduce_splits_k folds the split axis in fixed order
  {
    float* slab = dw + (long)ms * ((long)CO * CI);
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int coo = co0 + lk * 4 + reg;
      int cii = ci0 + wid * 16 + lm;
      if (coo < CO && cii < CI) slab[(long)coo * CI + cii] = acc[reg];
    }
  }
  if (WANT_BIAS && blockIdx.y == 0 && wid == 0) {
    db_part += __shfl_down(db_part, 16);
    db_part += __shfl_down(db_part, 32);
    if (lk == 0 && co0 + lm < CO) db[(long)ms * CO + co0 + lm] = db_part;
  }
}
