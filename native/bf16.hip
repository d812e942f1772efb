// bf16 NHWC engine — the MI355X-first hot path (BASELINE.json config 4).
//
// Layout contract:
//  - activations: NHWC bf16, channels padded to a multiple of 8 ("Cp") so an
//    MFMA A-fragment (8 contiguous k) is ONE 16-byte global load;
//  - GEMM k order for conv: k = kh*(KW*Cp) + kw*Cp + ci  — contiguous in x
//    within each kh row because Cp % 8 == 0;
//  - master weights fp32 canonical NCHW (checkpoint-compatible); converted to
//    bf16 + repacked during LDS staging (weights are tiny and L2-resident);
//  - accumulation fp32 (MFMA C/D); weight grads fp32 canonical.
//
// MFMA: v_mfma_f32_16x16x32_bf16 (guide §3):
//   A: lane l holds A[l&15][(l>>4)*8 + j], j=0..7 (8 contiguous k, 4 VGPRs)
//   B: lane l holds B[(l>>4)*8 + j][l&15]
//   C/D: col = l&15, row = (l>>4)*4 + reg  (f32x4)
// (verified on-device by mfma_probe_16x16x32, tests/test_kernels_gpu.py)
#include "common.h"
#include <hip/hip_bf16.h>

using bf16 = __bf16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using bf16x4 = __attribute__((ext_vector_type(4))) __bf16;

DEV f32x4 mfma16x16x32bf(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

DEV bf16 to_bf16(float f) { return (bf16)f; }
DEV float to_f32(bf16 h) { return (float)h; }

// zero-filled bf16x8
DEV bf16x8 bf8_zero() {
  bf16x8 z = {};
  return z;
}

// Hardware-bounds-checked 16-B gather (guide T8): a dead lane's offset is
// pushed past num_records and the load returns 0 IN HARDWARE — no exec
// predication, so hipcc can COUNT the load in its s_waitcnt bookkeeping.
// (An if-guarded gather is uncountable: the compiler wraps it in
// s_and_saveexec and drains vmcnt(0) at every consumer — measured as the
// dominant stall of every gather/staging loop this engine has.)
DEV buf_rsrc_t make_bf16_rsrc(const bf16* p, long n_elems) {
  // tensors >= 2 GB fall back to predicated loads (u32 offset limit);
  // callers gate on rsrc_ok()
  return __builtin_amdgcn_make_buffer_rsrc((void*)p, (short)0,
                                           (int)(n_elems * 2), 0x00020000);
}
DEV bool rsrc_fits(long n_elems) { return n_elems * 2 < (1LL << 31); }
DEV bf16x8 buf_load8(buf_rsrc_t r, long elem_off, bool ok) {
  int boff = ok ? (int)(elem_off * 2) : 0x7ffffff0;
  f32x4 t = __builtin_amdgcn_raw_buffer_load_b128(r, boff, 0, 0);
  return __builtin_bit_cast(bf16x8, t);
}
using f32x2_v = __attribute__((ext_vector_type(2))) float;
DEV bf16x4 buf_load4(buf_rsrc_t r, long elem_off, bool ok) {
  int boff = ok ? (int)(elem_off * 2) : 0x7ffffff0;
  f32x2_v t = __builtin_amdgcn_raw_buffer_load_b64(r, boff, 0, 0);
  return __builtin_bit_cast(bf16x4, t);
}
DEV buf_rsrc_t make_u8_rsrc(const unsigned char* p, long n_bytes) {
  return __builtin_amdgcn_make_buffer_rsrc((void*)p, (short)0, (int)n_bytes,
                                           0x00020000);
}
DEV unsigned long long buf_load_u64(buf_rsrc_t r, long byte_off, bool ok) {
  int boff = ok ? (int)byte_off : 0x7ffffff0;
  f32x2_v t = __builtin_amdgcn_raw_buffer_load_b64(r, boff, 0, 0);
  return __builtin_bit_cast(unsigned long long, t);
}

// ---------------------------------------------------------------------------
// Mapping probe: one wave computes D = A(16x32) @ B(32x16) via the assumed
// fragment mapping; the GPU test compares against torch.matmul.
// ---------------------------------------------------------------------------
__global__ void mfma_probe_16x16x32_k(const float* __restrict__ A,
                                      const float* __restrict__ B,
                                      float* __restrict__ D) {
  int l = threadIdx.x;
  if (l >= 64) return;
  int lm = l & 15, lk = l >> 4;
  bf16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = to_bf16(A[lm * 32 + lk * 8 + j]);      // A[row][k]
    b[j] = to_bf16(B[(lk * 8 + j) * 16 + lm]);    // B[k][col]
  }
  f32x4 c = {};
  c = mfma16x16x32bf(a, b, c);
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) D[(lk * 4 + reg) * 16 + lm] = c[reg];
}

// ---------------------------------------------------------------------------
// Conv2d forward, NHWC bf16.
//   D[m][co]: A = x patches (ONE predicated 16B load per lane per k-step),
//   B = weights repacked fp32->bf16 into LDS per BK=32 chunk.
// Block = 256 threads = 4 waves, each wave MW m-frags => BM = 4*MW*16 rows;
// grid.y covers 16-wide co tiles. Stores NHWC (co fastest), bf16.
// Bias fp32; optional fused ReLU (ResNet); co >= CO stores 0 (keeps channel
// padding clean for the next layer's A loads).
// ---------------------------------------------------------------------------
template <bool HAS_BIAS, bool RELU, bool BOUNDS, int MW, int NW>
__global__ __launch_bounds__(256) void conv2d_fwd_bf16_k(
    const bf16* __restrict__ x, const float* __restrict__ w,
    const float* __restrict__ bias, bf16* __restrict__ y,
    int CIp, int CI, int H, int W, int COp, int CO, int KH, int KW,
    int OH, int OW, int stride, int pad, int M, int K) {
  constexpr int BK = 32;
  const int BM = 4 * MW * 16;
  const int m0 = blockIdx.x * BM;
  const int co0 = blockIdx.y * (NW * 16);
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lm = l & 15;
  const int lk = l >> 4;

  // lds_w[co][k-chunk]: row pad 8 => conflict-light b128 reads
  __shared__ bf16 lds_w[NW * 16][BK + 8];
  __shared__ int lds_koff[BK / 8];
  __shared__ int lds_khw[BK / 8];

  const int KWC = KW * CIp;
  const int KHKW = KH * KW;

  int xbase[MW];
  int ih0[MW], iw0[MW];
  bool mval[MW];
#pragma unroll
  for (int mi = 0; mi < MW; ++mi) {
    int m = m0 + (wid * MW + mi) * 16 + lm;
    mval[mi] = m < M;
    int mm = mval[mi] ? m : 0;
    int n = mm / (OH * OW), r = mm % (OH * OW);
    int oh = r / OW, ow = r % OW;
    ih0[mi] = oh * stride - pad;
    iw0[mi] = ow * stride - pad;
    xbase[mi] = ((n * H + ih0[mi]) * W + iw0[mi]) * CIp;
  }

  f32x4 acc[MW][NW] = {};

  for (int k0 = 0; k0 < K; k0 += BK) {
    __syncthreads();
    for (int idx = tid; idx < NW * 16 * BK; idx += 256) {
      int co = idx >> 5;
      int kk = idx & 31;
      int k = k0 + kk;
      float v = 0.f;
      if (k < K && co0 + co < CO) {
        int kh = k / KWC, r = k % KWC;
        int kw = r / CIp, ci = r % CIp;
        if (ci < CI)
          v = w[((co0 + co) * CI + ci) * KHKW + kh * KW + kw];
      }
      lds_w[co][kk] = to_bf16(v);
    }
    if (tid < BK / 8) {
      int k = k0 + tid * 8;
      int kcl = k < K ? k : 0;
      int kh = kcl / KWC, r = kcl % KWC;
      int kw = r / CIp, ci0 = r % CIp;
      lds_koff[tid] = (kh * W + kw) * CIp + ci0;
      lds_khw[tid] = (kh << 16) | kw;
    }
    __syncthreads();

    int kgroups = min(BK, K - k0) / 8;
    for (int kg = 0; kg < kgroups; kg += 4) {
      bool gin = lk < (kgroups - kg);
      int gsel = gin ? kg + lk : 0;
      int koff = lds_koff[gsel];
      int khw = lds_khw[gsel];
      int kh = khw >> 16, kw = khw & 0xffff;
      bf16x8 bw[NW];
#pragma unroll
      for (int ni = 0; ni < NW; ++ni)
        bw[ni] = gin ? *reinterpret_cast<const bf16x8*>(&lds_w[ni * 16 + lm][gsel * 8])
                     : bf8_zero();
#pragma unroll
      for (int mi = 0; mi < MW; ++mi) {
        bool ok = mval[mi] & gin;
        if (BOUNDS) {
          int ih = ih0[mi] + kh, iw = iw0[mi] + kw;
          ok = ok && ih >= 0 && ih < H && iw >= 0 && iw < W;
        }
        bf16x8 a = ok ? *reinterpret_cast<const bf16x8*>(&x[xbase[mi] + koff])
                      : bf8_zero();
#pragma unroll
        for (int ni = 0; ni < NW; ++ni)
          acc[mi][ni] = mfma16x16x32bf(a, bw[ni], acc[mi][ni]);
      }
    }
  }

#pragma unroll
  for (int mi = 0; mi < MW; ++mi) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int m = m0 + (wid * MW + mi) * 16 + lk * 4 + reg;
      if (m >= M) continue;
      int n = m / (OH * OW), r = m % (OH * OW);
      long yb = (long)(n * (OH * OW) + r) * COp;
#pragma unroll
      for (int ni = 0; ni < NW; ++ni) {
        int co = co0 + ni * 16 + lm;
        if (co >= COp) continue;
        float v = (co < CO) ? acc[mi][ni][reg] : 0.f;
        if (HAS_BIAS && co < CO) v += bias[co];
        if (RELU) v = fmaxf(v, 0.f);
        y[yb + co] = to_bf16(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Conv2d backward-input (dgrad), NHWC bf16.
//   D[m'=(n,ih,iw)][ci]: A = dy patches (k' = kh*(KW*COp)+kw*COp+co, so an
//   A-frag is 8 contiguous co of one (kh,kw) tap — ONE predicated 16B load),
//   B = weights staged fp32->bf16 into LDS as [ci][k'] rows.
//   oh = (ih + pad - kh)/stride when divisible & in range, else 0.
// ---------------------------------------------------------------------------
template <bool S1, int MW, int NW>
__global__ __launch_bounds__(256) void conv2d_dgrad_bf16_k(
    const bf16* __restrict__ dy, const float* __restrict__ w,
    bf16* __restrict__ dx,
    int CIp, int CI, int H, int W, int COp, int CO, int KH, int KW,
    int OH, int OW, int stride, int pad, int Mp, int Kp) {
  constexpr int BK = 32;
  const int BM = 4 * MW * 16;
  const int m0 = blockIdx.x * BM;
  const int ci0 = blockIdx.y * (NW * 16);
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lm = l & 15;
  const int lk = l >> 4;

  __shared__ bf16 lds_w[NW * 16][BK + 8];
  __shared__ int lds_kdec[BK / 8];

  const int KWC = KW * COp;
  const int KHKW = KH * KW;

  int ih[MW], iw[MW], nn[MW];
  bool mval[MW];
#pragma unroll
  for (int mi = 0; mi < MW; ++mi) {
    int m = m0 + (wid * MW + mi) * 16 + lm;
    mval[mi] = m < Mp;
    int mm = mval[mi] ? m : 0;
    nn[mi] = mm / (H * W);
    int r = mm % (H * W);
    ih[mi] = r / W;
    iw[mi] = r % W;
  }

  f32x4 acc[MW][NW] = {};

  for (int k0 = 0; k0 < Kp; k0 += BK) {
    __syncthreads();
    for (int idx = tid; idx < NW * 16 * BK; idx += 256) {
      int ci = idx >> 5;
      int kk = idx & 31;
      int k = k0 + kk;
      float v = 0.f;
      if (k < Kp && ci0 + ci < CI) {
        int kh = k / KWC, r = k % KWC;
        int kw = r / COp, co = r % COp;
        if (co < CO)
          v = w[((co * CI) + ci0 + ci) * KHKW + kh * KW + kw];
      }
      lds_w[ci][kk] = to_bf16(v);
    }
    if (tid < BK / 8) {
      int k = k0 + tid * 8;
      int kcl = k < Kp ? k : 0;
      int kh = kcl / KWC, r = kcl % KWC;
      int kw = r / COp, co0g = r % COp;
      lds_kdec[tid] = (kh << 20) | (kw << 10) | co0g;
    }
    __syncthreads();

    int kgroups = min(BK, Kp - k0) / 8;
    for (int kg = 0; kg < kgroups; kg += 4) {
      bool gin = lk < (kgroups - kg);
      int gsel = gin ? kg + lk : 0;
      int kdec = lds_kdec[gsel];
      int kh = kdec >> 20, kw = (kdec >> 10) & 0x3ff, co0g = kdec & 0x3ff;
      bf16x8 bw[NW];
#pragma unroll
      for (int ni = 0; ni < NW; ++ni)
        bw[ni] = gin ? *reinterpret_cast<const bf16x8*>(&lds_w[ni * 16 + lm][gsel * 8])
                     : bf8_zero();
#pragma unroll
      for (int mi = 0; mi < MW; ++mi) {
        bf16x8 a = bf8_zero();
        if (mval[mi] & gin) {
          int tH = ih[mi] + pad - kh;
          int tW = iw[mi] + pad - kw;
          if (S1) {
            if (tH >= 0 && tH < OH && tW >= 0 && tW < OW)
              a = *reinterpret_cast<const bf16x8*>(
                  &dy[((long)(nn[mi] * OH + tH) * OW + tW) * COp + co0g]);
          } else {
            if (tH >= 0 && tW >= 0 && tH % stride == 0 && tW % stride == 0) {
              int oh = tH / stride, ow = tW / stride;
              if (oh < OH && ow < OW)
                a = *reinterpret_cast<const bf16x8*>(
                    &dy[((long)(nn[mi] * OH + oh) * OW + ow) * COp + co0g]);
            }
          }
        }
#pragma unroll
        for (int ni = 0; ni < NW; ++ni)
          acc[mi][ni] = mfma16x16x32bf(a, bw[ni], acc[mi][ni]);
      }
    }
  }

#pragma unroll
  for (int mi = 0; mi < MW; ++mi) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int m = m0 + (wid * MW + mi) * 16 + lk * 4 + reg;
      if (m >= Mp) continue;
      long xb = (long)m * CIp;
#pragma unroll
      for (int ni = 0; ni < NW; ++ni) {
        int ci = ci0 + ni * 16 + lm;
        if (ci >= CIp) continue;
        dx[xb + ci] = to_bf16(ci < CI ? acc[mi][ni][reg] : 0.f);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Small-K gather conv (fwd and dgrad), NHWC bf16 — whole-weight LDS residency.
// For LeNet-class shapes (K <= ~768, one or two 16-col tiles) the generic
// gather engines above are dominated by their per-BK=32 chunk structure:
// at K=400 that is 13 weight restages (each with a full k->(kh,kw,ci)
// integer-division decode per element) and 26 __syncthreads per block for
// only ~4 MFMAs of work per barrier pair — measured 3.2-3.5% MFMA-busy
// (profiles/lenet_bf16_pmc_counters.txt).  Here the ENTIRE repacked weight
// image [NW*16][K] plus the per-8k decode tables stage ONCE (one decode per
// element total), then the k loop runs with NO barriers and NO LDS writes:
// per k-group each lane issues one predicated 16-B x gather (L1-resident at
// these image sizes: a block's rows span ~2 images = ~9 KB) and MW*NW MFMAs.
// DGRAD=true consumes dy as A with the transposed/flipped weight decode of
// conv2d_dgrad_bf16_k (stride-1 only; s2 stays on the parity engines).
// ---------------------------------------------------------------------------
template <bool DGRAD, bool HAS_BIAS, bool RELU, bool BOUNDS, int MW, int NW>
__global__ __launch_bounds__(256) void conv_smallk_bf16_k(
    const bf16* __restrict__ x, const float* __restrict__ w,
    const float* __restrict__ bias, bf16* __restrict__ y,
    int CIp, int CI, int H, int W, int COp, int CO, int KH, int KW,
    int OH, int OW, int pad, int M, int K) {
  // dynamic LDS: [NW*16][K+8] bf16 weight image + K/8-entry decode tables
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int Krow = K + 8;                       // K is a multiple of 8
  bf16* lds_w = reinterpret_cast<bf16*>(smem);
  int* lds_dec = reinterpret_cast<int*>(smem + ((NW * 16 * Krow * 2 + 15) & ~15));
  int* lds_off = lds_dec + (K >> 3);            // fwd only: x offset per group

  const int BM = 4 * MW * 16;
  const int m0 = blockIdx.x * BM;
  const int n0 = blockIdx.y * (NW * 16);        // co tile (fwd) / ci tile (dgrad)
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lm = l & 15;
  const int lk = l >> 4;

  const int KWC = KW * (DGRAD ? COp : CIp);
  const int KHKW = KH * KW;

  // ---- one-time stage: whole weight image + decode tables ----------------
  for (int idx = tid; idx < NW * 16 * K; idx += 256) {
    int row = idx / K;                          // co (fwd) / ci (dgrad)
    int k = idx % K;
    int kh = k / KWC, r = k % KWC;
    float v = 0.f;
    if (DGRAD) {
      int kw = r / COp, co = r % COp;
      if (co < CO && n0 + row < CI)
        v = w[((co * CI) + n0 + row) * KHKW + kh * KW + kw];
    } else {
      int kw = r / CIp, ci = r % CIp;
      if (ci < CI && n0 + row < CO)
        v = w[((n0 + row) * CI + ci) * KHKW + kh * KW + kw];
    }
    lds_w[row * Krow + k] = to_bf16(v);
  }
  for (int g = tid; g < (K >> 3); g += 256) {
    int k = g * 8;
    int kh = k / KWC, r = k % KWC;
    if (DGRAD) {
      int kw = r / COp, c0 = r % COp;
      lds_dec[g] = (kh << 20) | (kw << 10) | c0;
    } else {
      int kw = r / CIp, ci0 = r % CIp;
      lds_dec[g] = (kh << 16) | kw;
      lds_off[g] = (kh * W + kw) * CIp + ci0;
    }
  }
  __syncthreads();

  // ---- per-lane m decode --------------------------------------------------
  int xbase[MW], ih0[MW], iw0[MW], nn[MW];
  bool mval[MW];
#pragma unroll
  for (int mi = 0; mi < MW; ++mi) {
    int m = m0 + (wid * MW + mi) * 16 + lm;
    mval[mi] = m < M;
    int mm = mval[mi] ? m : 0;
    if (DGRAD) {
      nn[mi] = mm / (H * W);
      int r = mm % (H * W);
      ih0[mi] = r / W;                          // ih
      iw0[mi] = r % W;                          // iw
    } else {
      int n = mm / (OH * OW), r = mm % (OH * OW);
      int oh = r / OW, ow = r % OW;
      ih0[mi] = oh - pad;
      iw0[mi] = ow - pad;
      xbase[mi] = ((n * H + ih0[mi]) * W + iw0[mi]) * CIp;
    }
  }

  f32x4 acc[MW][NW] = {};

  // ---- barrier-free k loop, A prefetched one group ahead ------------------
  // Per k-step each lane's A-gather feeds its MFMAs directly; without
  // prefetch the wave's critical path is ngroups/4 dependent L1/L2 round
  // trips. Loading group kg+4 before the MFMAs of group kg lets the next
  // load's latency hide under the current step's matrix work.
  const int ngroups = K >> 3;
  // Gathers are hardware-bounds-checked buffer loads (guide T8): an
  // exec-predicated gather is uncountable for hipcc's s_waitcnt
  // bookkeeping, so every MFMA group drained vmcnt(0) — 34 full drains per
  // loop body in the .s — serializing the one-group-ahead prefetch this
  // loop is built around (guide §5 ".s-level traps" (c)). A raw buffer
  // load needs NO predication: a dead lane's voffset is pushed past
  // num_records and the hardware returns 0.
  // x extent in bytes (host gates the smallK route to < 2 GB tensors):
  // fwd M = N*OH*OW -> images N = M/(OH*OW); dgrad x==dy with M = N*H*W
  const long n_imgs = DGRAD ? (long)M / (H * W) : (long)M / (OH * OW);
  const long xbytes =
      (DGRAD ? n_imgs * OH * OW * COp : n_imgs * H * W * CIp) * 2;
  auto xrsrc = __builtin_amdgcn_make_buffer_rsrc((void*)x, (short)0,
                                                 (int)xbytes, 0x00020000);
  auto gather_a = [&](int kg, bf16x8 (&a)[MW]) {
    bool gin = lk < (ngroups - kg);
    int gsel = gin ? kg + lk : 0;
    int kdec = lds_dec[gsel];
    if (DGRAD) {
      int kh = kdec >> 20, kw = (kdec >> 10) & 0x3ff, c0g = kdec & 0x3ff;
#pragma unroll
      for (int mi = 0; mi < MW; ++mi) {
        int tH = ih0[mi] + pad - kh;
        int tW = iw0[mi] + pad - kw;
        bool ok = mval[mi] & gin && tH >= 0 && tH < OH && tW >= 0 && tW < OW;
        long off = ((long)(nn[mi] * OH + tH) * OW + tW) * COp + c0g;
        int boff = ok ? (int)(off * 2) : 0x7ffffff0;
        f32x4 t = __builtin_amdgcn_raw_buffer_load_b128(xrsrc, boff, 0, 0);
        a[mi] = __builtin_bit_cast(bf16x8, t);
      }
    } else {
      int koff = lds_off[gsel];
      int kh = kdec >> 16, kw = kdec & 0xffff;
#pragma unroll
      for (int mi = 0; mi < MW; ++mi) {
        bool ok = mval[mi] & gin;
        if (BOUNDS) {
          int ih = ih0[mi] + kh, iw = iw0[mi] + kw;
          ok = ok && ih >= 0 && ih < H && iw >= 0 && iw < W;
        }
        int boff = ok ? (int)(((long)xbase[mi] + koff) * 2) : 0x7ffffff0;
        f32x4 t = __builtin_amdgcn_raw_buffer_load_b128(xrsrc, boff, 0, 0);
        a[mi] = __builtin_bit_cast(bf16x8, t);
      }
    }
  };

  // B fragments prefetch one group ahead too (unconditional reads at the
  // clamped gsel=0 row, zero-selected after — keeps the LDS reads off the
  // MFMA block's critical wait)
  auto load_bw = [&](int kg, bf16x8 (&bw)[NW]) {
    bool gin = lk < (ngroups - kg);
    int gsel = gin ? kg + lk : 0;
#pragma unroll
    for (int ni = 0; ni < NW; ++ni) {
      bf16x8 t = *reinterpret_cast<const bf16x8*>(
          &lds_w[(ni * 16 + lm) * Krow + gsel * 8]);
      bw[ni] = gin ? t : bf8_zero();
    }
  };

  bf16x8 a_cur[MW], bw_cur[NW];
  gather_a(0, a_cur);
  load_bw(0, bw_cur);
  for (int kg = 0; kg < ngroups; kg += 4) {
    bf16x8 a_nxt[MW], bw_nxt[NW];
    if (kg + 4 < ngroups) {
      gather_a(kg + 4, a_nxt);
      load_bw(kg + 4, bw_nxt);
    } else {
#pragma unroll
      for (int mi = 0; mi < MW; ++mi) a_nxt[mi] = bf8_zero();
#pragma unroll
      for (int ni = 0; ni < NW; ++ni) bw_nxt[ni] = bf8_zero();
    }
#pragma unroll
    for (int mi = 0; mi < MW; ++mi)
#pragma unroll
      for (int ni = 0; ni < NW; ++ni)
        acc[mi][ni] = mfma16x16x32bf(a_cur[mi], bw_cur[ni], acc[mi][ni]);
#pragma unroll
    for (int mi = 0; mi < MW; ++mi) a_cur[mi] = a_nxt[mi];
#pragma unroll
    for (int ni = 0; ni < NW; ++ni) bw_cur[ni] = bw_nxt[ni];
  }

  // ---- epilogue -----------------------------------------------------------
  const int NC = DGRAD ? CIp : COp;             // padded col extent
  const int NCv = DGRAD ? CI : CO;              // valid cols
#pragma unroll
  for (int mi = 0; mi < MW; ++mi) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int m = m0 + (wid * MW + mi) * 16 + lk * 4 + reg;
      if (m >= M) continue;
      long yb = (long)m * NC;
#pragma unroll
      for (int ni = 0; ni < NW; ++ni) {
        int c = n0 + ni * 16 + lm;
        if (c >= NC) continue;
        float v = (c < NCv) ? acc[mi][ni][reg] : 0.f;
        if (HAS_BIAS && c < NCv) v += bias[c];
        if (RELU) v = fmaxf(v, 0.f);
        y[yb + c] = to_bf16(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Conv2d backward-weight (wgrad) + bias, NHWC bf16 -> fp32 canonical dw/db.
// Reduction over m. Waves split the k-tile (k-tile = 4*KT*16); the dy chunk
// and the im2col(x) chunk are staged NON-transposed as [m][16] sub-images
// (vectorized bf16x8 global reads -> ds_write_b128), and MFMA fragments come
// out via gfx950's hardware transpose read ds_read_tr16_b64 (guide T10) —
// no scalar LDS transpose pass. Each block writes one partial slab; the
// deterministic reduce_splits_k folds the split axis.
// Requires K >= ~128 for full wave utilization (LeNet: K=200/400; ResNet
// 3x3: >=576). KT = per-wave 16-row k-frags (2 => k-tile 128).
// ---------------------------------------------------------------------------
using lds_bf4_ptr = __attribute__((address_space(3))) bf16x4*;

DEV bf16x8 tr_read_8m(const bf16* lds_img, int l) {
  // image: [m][16] bf16 row-major. Returns this lane's 8 m-elements at
  // column (l&15), rows (l>>4)*8 .. +7, via two 4-row transpose reads.
  // ds_read_b64_tr_b16 semantics (verified by tr16_probe on-device): with
  // the NATURAL per-lane addressing addr_l = base + lane_in_group*8B, a
  // 16-lane group collectively reads 64 contiguous bf16 (4 rows x 16) and
  // lane l receives column (l&15), element j = row j of the 4-row block.
  int lm = l & 15, lk = l >> 4;
  const bf16* p0 = lds_img + lk * 128 + lm * 4;
  bf16x4 r0 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16((lds_bf4_ptr)p0);
  bf16x4 r1 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16((lds_bf4_ptr)(p0 + 64));
  bf16x8 a;
#pragma unroll
  for (int j = 0; j < 4; ++j) { a[j] = r0[j]; a[4 + j] = r1[j]; }
  return a;
}

template <bool BOUNDS, bool WANT_BIAS, int KT, int NW, int MC = 64>
__global__ __launch_bounds__(256) void conv2d_wgrad_bf16_k(
    const bf16* __restrict__ dy, const bf16* __restrict__ x,
    float* __restrict__ dw_ws, float* __restrict__ db_ws,
    int CIp, int CI, int H, int W, int COp, int CO, int KH, int KW,
    int OH, int OW, int stride, int pad, int M, int K, int Kcan,
    int m_per_split) {
  // MC = m per chunk (shared by all 4 waves)
  constexpr int NKF = 4 * KT;     // 16-row k-frag images per block
  const int k0 = blockIdx.x * (NKF * 16);
  const int co0 = blockIdx.y * (NW * 16);
  const int ms = blockIdx.z;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lm = l & 15;
  const int lk = l >> 4;

  __shared__ bf16 lds_x[NKF][MC][16];       // [kf][m][16k] sub-images
  __shared__ bf16 lds_dy[NW][MC][16];       // [ni][m][16co]
  __shared__ int lds_mbase[MC];
  __shared__ int lds_mhw[MC];
  // per-8k decode tables in LDS: per-thread int[NKF*2] arrays are
  // runtime-indexed in the staging loop -> scratch spills (guide rule 20)
  __shared__ int lds_goff[NKF * 2];
  __shared__ int lds_gkhw[NKF * 2];

  const int KWC = KW * CIp;
  const int OHOW = OH * OW;

  if (tid < NKF * 2) {
    int g = tid;
    int k = k0 + g * 8;
    int kcl = k < K ? k : 0;
    int kh = kcl / KWC, kr = kcl % KWC;
    int kw = kr / CIp, ci0 = kr % CIp;
    lds_goff[g] = (kh * W + kw) * CIp + ci0;
    lds_gkhw[g] = (kh << 16) | kw;
  }

  f32x4 acc[KT][NW] = {};
  float db_part[NW] = {};

  int m_begin = ms * m_per_split;
  int m_end = min(M, m_begin + m_per_split);

  const long x_elems = (long)(M / OHOW) * H * W * CIp;   // N*H*W*CIp
  const long dy_elems = (long)M * COp;
  const bool use_buf = rsrc_fits(x_elems) && rsrc_fits(dy_elems);
  buf_rsrc_t xr = make_bf16_rsrc(x, x_elems);
  buf_rsrc_t dyr = make_bf16_rsrc(dy, dy_elems);

  for (int mc = m_begin; mc < m_end; mc += MC) {
    __syncthreads();
    if (tid < MC) {
      int m = mc + tid;
      int mm = m < m_end ? m : 0;
      int n = mm / OHOW, r = mm % OHOW;
      int oh = r / OW, ow = r % OW;
      int ih0 = oh * stride - pad, iw0 = ow * stride - pad;
      lds_mbase[tid] = (n * H + ih0) * W * CIp + iw0 * CIp;
      lds_mhw[tid] = ((ih0 + 16384) << 16) | (iw0 + 16384);
    }
    __syncthreads();
    for (int idx = tid; idx < NKF * MC * 2; idx += 256) {
      int half = idx & 1;
      int rest = idx >> 1;
      int mloc = rest % MC;
      int kf = rest / MC;
      int g = kf * 2 + half;
      int m = mc + mloc;
      int k = k0 + g * 8;
      bool ok = m < m_end && k < K;
      if (BOUNDS && ok) {
        int hw = lds_mhw[mloc];
        int khw = lds_gkhw[g];
        int ihh = (hw >> 16) - 16384 + (khw >> 16);
        int iww = (hw & 0xffff) - 16384 + (khw & 0xffff);
        ok = ihh >= 0 && ihh < H && iww >= 0 && iww < W;
      }
      long off = (long)lds_mbase[mloc] + lds_goff[g];
      bf16x8 v;
      if (use_buf) {
        v = buf_load8(xr, off, ok);   // countable: no vmcnt(0) per stage
      } else {
        v = bf8_zero();
        if (ok) v = *reinterpret_cast<const bf16x8*>(&x[off]);
      }
      *reinterpret_cast<bf16x8*>(&lds_x[kf][mloc][half * 8]) = v;
    }
    for (int idx = tid; idx < NW * MC * 2; idx += 256) {
      int half = idx & 1;
      int rest = idx >> 1;
      int mloc = rest % MC;
      int ni = rest / MC;
      int m = mc + mloc;
      int cb = co0 + ni * 16 + half * 8;
      bool ok = m < m_end && cb < COp;
      bf16x8 v;
      if (use_buf) {
        v = buf_load8(dyr, (long)m * COp + cb, ok);
      } else {
        v = bf8_zero();
        if (ok) v = *reinterpret_cast<const bf16x8*>(&dy[(long)m * COp + cb]);
      }
      *reinterpret_cast<bf16x8*>(&lds_dy[ni][mloc][half * 8]) = v;
    }
    __syncthreads();

#pragma unroll
    for (int mm = 0; mm < MC; mm += 32) {
      bf16x8 b[NW];
#pragma unroll
      for (int ni = 0; ni < NW; ++ni) {
        b[ni] = tr_read_8m(&lds_dy[ni][mm][0], l);
        if (WANT_BIAS && wid == 0) {
#pragma unroll
          for (int j = 0; j < 8; ++j) db_part[ni] += to_f32(b[ni][j]);
        }
      }
      // all KT A tr-reads issue before the MFMA block (wait collapse)
      bf16x8 av[KT];
#pragma unroll
      for (int kf = 0; kf < KT; ++kf)
        av[kf] = tr_read_8m(&lds_x[wid * KT + kf][mm][0], l);
#pragma unroll
      for (int kf = 0; kf < KT; ++kf)
#pragma unroll
        for (int ni = 0; ni < NW; ++ni)
          acc[kf][ni] = mfma16x16x32bf(av[kf], b[ni], acc[kf][ni]);
    }
  }

  {
    float* slab = dw_ws + (long)ms * (CO * (long)Kcan);
#pragma unroll
    for (int kf = 0; kf < KT; ++kf) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        int kk = k0 + (wid * KT + kf) * 16 + lk * 4 + reg;
        if (kk >= K) continue;
        int kh = kk / KWC, r = kk % KWC;
        int kw = r / CIp, ci = r % CIp;
        if (ci >= CI) continue;
        long koffc = ((long)ci) * (KH * KW) + kh * KW + kw;
#pragma unroll
        for (int ni = 0; ni < NW; ++ni) {
          int co = co0 + ni * 16 + lm;
          if (co < CO) slab[(long)co * Kcan + koffc] = acc[kf][ni][reg];
        }
      }
    }
  }
  if (WANT_BIAS && blockIdx.x == 0 && wid == 0) {
#pragma unroll
    for (int ni = 0; ni < NW; ++ni) {
      float v = db_part[ni];
      v += __shfl_down(v, 16);
      v += __shfl_down(v, 32);
      int co = co0 + ni * 16 + lm;
      if (lk == 0 && co < CO) db_ws[(long)ms * CO + co] = v;
    }
  }
}

// k-tile for the bf16 wgrad above (element rows covered per block)
constexpr int WGRAD_BF16_KTILE = 4 * 2 * 16;  // KT=2 -> 128

// ---------------------------------------------------------------------------
// Region-stationary wgrad for the s1 3x3 pad-1 shapes (ResNet residual
// convs). The generic engine above re-gathers the im2col form of x from
// global per k-tile (~9x the raw bytes, x2 per extra co-tile: ~925 MB/call
// on layer2, L2-thrashed across splits — measured 7% MFMA-busy). Here a
// block owns [64 co x 32 ci x all 9 taps] and marches m in 256-pixel
// chunks:
//   - the x ROWS for a chunk stage once (raw bytes, not im2col): region
//     [rows+halo][W+2 zero-padded cols][32 ci], vec8-aligned since the
//     ci stride is 64 B;
//   - dy [256][64] stages once per chunk and each wave hoists its 8
//     B-fragments into VGPRs immediately (dy is TAP-INVARIANT — the
//     insight that makes the 9-tap reuse free);
//   - per tap, [256][16]x2 A-images build LDS->LDS from the region with
//     per-m row predicates (vec8 reads at ci-stride, vec8 writes), into
//     the same 32 KB the dy planes used (double-buffered across taps);
//   - fragments then come out via the same tr16 recipe as the generic
//     engine: 144 MFMAs per wave-chunk against 152 transpose reads.
// Global traffic per call: x once per co-tile + dy once per ci-tile
// (~310 MB on layer2 vs ~925) and the MFMA:gather ratio flips the kernel
// from L2-latency-bound to matrix-core-bound.
// ---------------------------------------------------------------------------
// S=2 extension (the layer2/3/4 block-0 3x3 s2 convs): requires
// H==2*OH, W==2*OW so the input global row for output row g is the LINEAR
// 2g+kh-1 — the same contiguous-row staging then works with a 2x-wide
// region and the tap predicate S*oy+kh in [0, S*OH).
template <int MC, int S = 1, bool AFF = false>
__global__ __launch_bounds__(256) void conv2d_wgrad_rs_bf16_k(
    const bf16* __restrict__ dy, const bf16* __restrict__ x,
    float* __restrict__ dw_ws, float* __restrict__ db_ws, int CIp, int CI,
    int COp, int CO, int OH, int OW, int M, int Kcan, int m_per_split,
    const float* __restrict__ aff = nullptr) {
  // AFF: x is the raw conv output whose BN(+ReLU) apply is fused into the
  // region staging (same transform as the fused forward — the conv's true
  // input y_bn is never materialized, so wgrad reproduces it on load)
  const int cic0 = blockIdx.x * 32;       // ci tile
  const int co0 = blockIdx.y * 64;        // co tile (4 waves x 16)
  const int ms = blockIdx.z;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lm = l & 15;
  const int Wi = S * OW;   // input width (host guarantees W == S*OW)
  const int W2 = Wi + 2;
  // region capacity: worst-case rows*(Wi+2) over the routed OW range
  constexpr int PX = (S == 2) ? (MC <= 64 ? 528 : 640)
                             : (MC <= 64 ? 296 : MC <= 128 ? 352 : 464);

  __shared__ bf16 lds_reg[PX * 32];           // [px][32ci]
  // dy planes then A-image slots; +16-elem stagger per plane so the build's
  // paired-plane vec8 writes land on different bank columns
  __shared__ bf16 lds_img_raw[4 * MC * 16 + 3 * 16];
  __shared__ int lds_mm[MC];                  // packed (px_index<<10 | oy+1)
  auto plane = [&](int p) { return &lds_img_raw[p * (MC * 16 + 16)]; };
  constexpr int MSTEPS = MC / 32;

  const int m_begin = ms * m_per_split;
  const int m_end = min(M, m_begin + m_per_split);

  f32x4 acc[2][9] = {};   // [ci-frag][tap]
  float db_part = 0.f;    // bias grad: colsum of this wave's dy fragments

  const long x_elems = (long)(M / OW) * S * Wi * CIp;  // N*H*W*CIp
  const long dy_elems = (long)M * COp;
  const bool use_buf = rsrc_fits(x_elems) && rsrc_fits(dy_elems);
  buf_rsrc_t xr = make_bf16_rsrc(x, x_elems);
  buf_rsrc_t dyr = make_bf16_rsrc(dy, dy_elems);

  for (int mc = m_begin; mc < m_end; mc += MC) {
    const int g0 = mc / OW;               // first OUTPUT global row of chunk
    const int mlast = min(mc + MC - 1, m_end - 1);
    const int rows = S * (mlast / OW - g0) + 3;  // input rows incl. +-1 halo
    __syncthreads();
    // per-m map: region pixel index + output row (for tap predicates)
    if (tid < MC) {
      int m = mc + tid;
      int mm = m < m_end ? m : mc;
      int g = mm / OW, ox = mm % OW;
      int px = (S * (g - g0) + 1) * W2 + S * ox + 1;
      int oy = g % OH;
      lds_mm[tid] = (px << 10) | (m < m_end ? oy + 1 : 0);  // 0 => dead m
    }
    // x region rows [g0-1, g0+rows-2], zero-padded cols and out-of-image rows
    for (int idx = tid; idx < rows * W2 * 4; idx += 256) {
      int g8 = idx & 3;
      int p = idx >> 2;
      int c = p % W2;
      int r = p / W2;
      int gi = S * g0 - 1 + r;             // input global row
      int iw = c - 1;
      bool ok = gi >= 0 && gi < S * (M / OW) && iw >= 0 && iw < Wi;
      long off = ((long)gi * Wi + iw) * CIp + cic0 + g8 * 8;
      bf16x8 v;
      if (use_buf) {
        v = buf_load8(xr, off, ok);   // countable: no vmcnt(0) per stage
      } else {
        v = bf8_zero();
        if (ok) v = *reinterpret_cast<const bf16x8*>(&x[off]);
      }
      if (AFF && ok) {
        const float* sc = &aff[cic0 + g8 * 8];
        const float* sh = &aff[CIp + cic0 + g8 * 8];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          v[j] = to_bf16(fmaxf(to_f32(v[j]) * sc[j] + sh[j], 0.f));
      }
      *reinterpret_cast<bf16x8*>(&lds_reg[(p << 5) + g8 * 8]) = v;
    }
    // dy planes [ni][256][16co]
    for (int idx = tid; idx < 4 * MC * 2; idx += 256) {
      int half = idx & 1;
      int rest = idx >> 1;
      int mloc = rest % MC;
      int ni = rest / MC;
      int m = mc + mloc;
      int cb = co0 + ni * 16 + half * 8;
      bool ok = m < m_end && cb < COp;
      bf16x8 v;
      if (use_buf) {
        v = buf_load8(dyr, (long)m * COp + cb, ok);
      } else {
        v = bf8_zero();
        if (ok) v = *reinterpret_cast<const bf16x8*>(&dy[(long)m * COp + cb]);
      }
      *reinterpret_cast<bf16x8*>(&plane(ni)[mloc * 16 + half * 8]) = v;
    }
    __syncthreads();

    // hoist this wave's 8 dy B-fragments before the planes are reused
    bf16x8 bfr[MSTEPS];
#pragma unroll
    for (int s = 0; s < MSTEPS; ++s)
      bfr[s] = tr_read_8m(plane(wid) + s * 32 * 16, l);
    if (db_ws != nullptr && blockIdx.x == 0) {
#pragma unroll
      for (int s = 0; s < MSTEPS; ++s)
#pragma unroll
        for (int j = 0; j < 8; ++j) db_part += to_f32(bfr[s][j]);
    }
    __syncthreads();

    // A-image build for one tap: planes (buf*2, buf*2+1) = ci halves
    auto build = [&](int tap, int buf) {
      int kh = tap / 3 - 1, kw = tap % 3 - 1;  // tap offsets (pad 1)
      for (int idx = tid; idx < MC * 4; idx += 256) {
        int g8 = idx & 3;
        int mloc = idx >> 2;
        int mm = lds_mm[mloc];
        int oy = (mm & 1023) - 1;
        bf16x8 v = bf8_zero();
        if (oy >= 0 && S * oy + kh >= 0 && S * oy + kh < S * OH) {
          int px = (mm >> 10) + kh * W2 + kw;
          v = *reinterpret_cast<const bf16x8*>(&lds_reg[(px << 5) + g8 * 8]);
        }
        *reinterpret_cast<bf16x8*>(
            &plane(buf * 2 + (g8 >> 1))[mloc * 16 + (g8 & 1) * 8]) = v;
      }
    };

    build(0, 0);
    __syncthreads();
#pragma unroll
    for (int tap = 0; tap < 9; ++tap) {
      int buf = tap & 1;
      if (tap + 1 < 9) build(tap + 1, buf ^ 1);  // overlap with compute
      // paired tr-reads BEFORE their MFMAs (and alternating accumulators):
      // the per-f read-then-consume order exposed a full lgkmcnt drain per
      // MFMA and chained same-acc MFMAs (dep latency 40 > issue 17)
#pragma unroll
      for (int s = 0; s < MSTEPS; ++s) {
        bf16x8 a0 = tr_read_8m(plane(buf * 2 + 0) + s * 32 * 16, l);
        bf16x8 a1 = tr_read_8m(plane(buf * 2 + 1) + s * 32 * 16, l);
        acc[0][tap] = mfma16x16x32bf(a0, bfr[s], acc[0][tap]);
        acc[1][tap] = mfma16x16x32bf(a1, bfr[s], acc[1][tap]);
      }
      __syncthreads();
    }
  }

  // partial slab: D row (lk*4+reg) = ci-local, col (lm) = co
  float* slab = dw_ws + (long)ms * (CO * (long)Kcan);
  const int lk = l >> 4;
#pragma unroll
  for (int f = 0; f < 2; ++f) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int ci = cic0 + f * 16 + lk * 4 + reg;
      if (ci >= CI) continue;
      int co = co0 + wid * 16 + lm;
      if (co >= CO) continue;
#pragma unroll
      for (int tap = 0; tap < 9; ++tap)
        slab[(long)co * Kcan + (long)ci * 9 + tap] = acc[f][tap][reg];
    }
  }
  if (db_ws != nullptr && blockIdx.x == 0) {
    db_part += __shfl_down(db_part, 16);
    db_part += __shfl_down(db_part, 32);
    int co = co0 + wid * 16 + lm;
    if (lk == 0 && co < CO) db_ws[(long)ms * CO + co] = db_part;
  }
}

// ---------------------------------------------------------------------------
// Linear forward, bf16: y[M][COp] = x[M][CIk] . w^T + b (optional ReLU).
// w is fp32 canonical [CO][CIcan]; perm[k] maps the (padded, possibly
// NHWC-flattened) k index to the canonical column (-1 => zero pad), so the
// checkpoint-ordered weights never need a materialized repack.
// ---------------------------------------------------------------------------
template <bool HAS_BIAS, bool RELU, int MW>
__global__ __launch_bounds__(256) void linear_fwd_bf16_k(
    const bf16* __restrict__ x, const float* __restrict__ w,
    const float* __restrict__ bias, const int* __restrict__ perm,
    bf16* __restrict__ y, int M, int CIk, int CIcan, int COp, int CO) {
  constexpr int BK = 32;
  const int BM = 4 * MW * 16;
  const int m0 = blockIdx.x * BM;
  const int co0 = blockIdx.y * 16;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lm = l & 15;
  const int lk = l >> 4;

  __shared__ bf16 lds_w[16][BK + 8];

  f32x4 acc[MW] = {};

  for (int k0 = 0; k0 < CIk; k0 += BK) {
    __syncthreads();
    for (int idx = tid; idx < 16 * BK; idx += 256) {
      int co = idx >> 5;
      int kk = idx & 31;
      int k = k0 + kk;
      float v = 0.f;
      if (k < CIk && co0 + co < CO) {
        int ci = perm[k];
        if (ci >= 0) v = w[(co0 + co) * CIcan + ci];
      }
      lds_w[co][kk] = to_bf16(v);
    }
    __syncthreads();

    int kgroups = min(BK, CIk - k0) / 8;
    for (int kg = 0; kg < kgroups; kg += 4) {
      bool gin = lk < (kgroups - kg);
      int gsel = gin ? kg + lk : 0;
      bf16x8 b = gin ? *reinterpret_cast<const bf16x8*>(&lds_w[lm][gsel * 8])
                     : bf8_zero();
#pragma unroll
      for (int mi = 0; mi < MW; ++mi) {
        int m = m0 + (wid * MW + mi) * 16 + lm;
        bf16x8 a = (m < M && gin)
                       ? *reinterpret_cast<const bf16x8*>(
                             &x[(long)m * CIk + k0 + gsel * 8])
                       : bf8_zero();
        acc[mi] = mfma16x16x32bf(a, b, acc[mi]);
      }
    }
  }

#pragma unroll
  for (int mi = 0; mi < MW; ++mi) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int m = m0 + (wid * MW + mi) * 16 + lk * 4 + reg;
      if (m >= M) continue;
      int co = co0 + lm;
      if (co >= COp) continue;
      float v = (co < CO) ? acc[mi][reg] : 0.f;
      if (HAS_BIAS && co < CO) v += bias[co];
      if (RELU) v = fmaxf(v, 0.f);
      y[(long)m * COp + co] = to_bf16(v);
    }
  }
}

// ---------------------------------------------------------------------------
// Linear backward-input, bf16: dx[M][CIk] = dyr[M][COp] . w (relu-masked).
//   A = dyr rows (8 contiguous co; mask from saved y at identical offsets),
//   B = w^T staged via perm into LDS as [ci_k][co] rows.
// ---------------------------------------------------------------------------
template <bool RELU, int MW>
__global__ __launch_bounds__(256) void linear_dx_bf16_k(
    const bf16* __restrict__ dy, const float* __restrict__ w,
    const bf16* __restrict__ yact, const int* __restrict__ perm,
    bf16* __restrict__ dx, int M, int CIk, int CIcan, int COp, int CO) {
  constexpr int BK = 32;
  const int BM = 4 * MW * 16;
  const int m0 = blockIdx.x * BM;
  const int ci0 = blockIdx.y * 16;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lm = l & 15;
  const int lk = l >> 4;

  __shared__ bf16 lds_w[16][BK + 8];   // [ci_k][co-chunk]

  f32x4 acc[MW] = {};

  for (int k0 = 0; k0 < COp; k0 += BK) {
    __syncthreads();
    for (int idx = tid; idx < 16 * BK; idx += 256) {
      int ci = idx >> 5;
      int kk = idx & 31;
      int co = k0 + kk;
      float v = 0.f;
      if (co < CO && ci0 + ci < CIk) {
        int cic = perm[ci0 + ci];
        if (cic >= 0) v = w[co * CIcan + cic];
      }
      lds_w[ci][kk] = to_bf16(v);
    }
    __syncthreads();

    int kgroups = min(BK, COp - k0) / 8;
    for (int kg = 0; kg < kgroups; kg += 4) {
      bool gin = lk < (kgroups - kg);
      int gsel = gin ? kg + lk : 0;
      bf16x8 b = gin ? *reinterpret_cast<const bf16x8*>(&lds_w[lm][gsel * 8])
                     : bf8_zero();
#pragma unroll
      for (int mi = 0; mi < MW; ++mi) {
        int m = m0 + (wid * MW + mi) * 16 + lm;
        bf16x8 a = bf8_zero();
        if (m < M && gin) {
          long off = (long)m * COp + k0 + gsel * 8;
          a = *reinterpret_cast<const bf16x8*>(&dy[off]);
          if (RELU) {
            bf16x8 yv = *reinterpret_cast<const bf16x8*>(&yact[off]);
#pragma unroll
            for (int j = 0; j < 8; ++j)
              if (!(to_f32(yv[j]) > 0.f)) a[j] = to_bf16(0.f);
          }
        }
        acc[mi] = mfma16x16x32bf(a, b, acc[mi]);
      }
    }
  }

#pragma unroll
  for (int mi = 0; mi < MW; ++mi) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int m = m0 + (wid * MW + mi) * 16 + lk * 4 + reg;
      if (m >= M) continue;
      int ci = ci0 + lm;
      if (ci >= CIk) continue;
      dx[(long)m * CIk + ci] = to_bf16(acc[mi][reg]);
    }
  }
}

// ---------------------------------------------------------------------------
// Linear backward-weight (+bias), bf16 -> fp32 canonical via perm.
// Same structure as conv wgrad: transposed LDS chunks, waves on m, partial
// slabs. dyr mask applied at staging time.
// ---------------------------------------------------------------------------
template <bool RELU, bool WANT_BIAS, int KT>
__global__ __launch_bounds__(256) void linear_dw_bf16_k(
    const bf16* __restrict__ dy, const bf16* __restrict__ x,
    const bf16* __restrict__ yact, const int* __restrict__ perm,
    float* __restrict__ dw_ws, float* __restrict__ db_ws,
    int M, int CIk, int CIcan, int COp, int CO, int m_per_split) {
  constexpr int MC = 64;
  const int k0 = blockIdx.x * (KT * 16);   // ci_k tile
  const int co0 = blockIdx.y * 16;
  const int ms = blockIdx.z;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lm = l & 15;
  const int lk = l >> 4;

  __shared__ bf16 lds_xT[4][KT * 16][MC + 8];
  __shared__ bf16 lds_dyT[4][16][MC + 8];
  __shared__ float lds_red[4 * KT * 256];

  f32x4 acc[KT] = {};
  float db_part = 0.f;

  int m_begin = ms * m_per_split;
  int m_end = min(M, m_begin + m_per_split);

  for (int mc = m_begin; mc < m_end; mc += 4 * MC) {
    __syncthreads();
    for (int idx = tid; idx < 4 * MC * (KT * 2); idx += 256) {
      int kg = idx % (KT * 2);
      int rest = idx / (KT * 2);
      int mloc = rest % MC;
      int w = rest / MC;
      int m = mc + w * MC + mloc;
      int k = k0 + kg * 8;
      bf16x8 v = bf8_zero();
      if (m < m_end && k < CIk)
        v = *reinterpret_cast<const bf16x8*>(&x[(long)m * CIk + k]);
#pragma unroll
      for (int j = 0; j < 8; ++j) lds_xT[w][kg * 8 + j][mloc] = v[j];
    }
    for (int idx = tid; idx < 4 * MC * 2; idx += 256) {
      int cg = idx & 1;
      int rest = idx >> 1;
      int mloc = rest % MC;
      int w = rest / MC;
      int m = mc + w * MC + mloc;
      bf16x8 v = bf8_zero();
      if (m < m_end && co0 + cg * 8 < COp) {
        long off = (long)m * COp + co0 + cg * 8;
        v = *reinterpret_cast<const bf16x8*>(&dy[off]);
        if (RELU) {
          bf16x8 yv = *reinterpret_cast<const bf16x8*>(&yact[off]);
#pragma unroll
          for (int j = 0; j < 8; ++j)
            if (!(to_f32(yv[j]) > 0.f)) v[j] = to_bf16(0.f);
        }
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) lds_dyT[w][cg * 8 + j][mloc] = v[j];
    }
    __syncthreads();

    int mlim = min(MC, m_end - mc - wid * MC);
    for (int mm = 0; mm < mlim; mm += 32) {
      bf16x8 b = *reinterpret_cast<const bf16x8*>(&lds_dyT[wid][lm][mm + lk * 8]);
      if (WANT_BIAS) {
#pragma unroll
        for (int j = 0; j < 8; ++j) db_part += to_f32(b[j]);
      }
#pragma unroll
      for (int kf = 0; kf < KT; ++kf) {
        bf16x8 a = *reinterpret_cast<const bf16x8*>(
            &lds_xT[wid][kf * 16 + lm][mm + lk * 8]);
        acc[kf] = mfma16x16x32bf(a, b, acc[kf]);
      }
    }
  }

  __syncthreads();
#pragma unroll
  for (int kf = 0; kf < KT; ++kf)
#pragma unroll
    for (int reg = 0; reg < 4; ++reg)
      lds_red[((wid * KT + kf) * 256) + (lk * 4 + reg) * 16 + lm] = acc[kf][reg];
  __syncthreads();
  if (wid == 0) {
    float* slab = dw_ws + (long)ms * (CO * (long)CIcan);
#pragma unroll
    for (int kf = 0; kf < KT; ++kf) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        int e = (lk * 4 + reg) * 16 + lm;
        float v = lds_red[kf * 256 + e] + lds_red[(KT + kf) * 256 + e] +
                  lds_red[(2 * KT + kf) * 256 + e] + lds_red[(3 * KT + kf) * 256 + e];
        int kk = k0 + kf * 16 + lk * 4 + reg;
        int co = co0 + lm;
        if (kk < CIk && co < CO) {
          int cic = perm[kk];
          if (cic >= 0) slab[co * CIcan + cic] = v;
        }
      }
    }
  }
  if (WANT_BIAS && blockIdx.x == 0) {
    db_part += __shfl_down(db_part, 16);
    db_part += __shfl_down(db_part, 32);
    __syncthreads();
    if (lk == 0) lds_red[wid * 16 + lm] = db_part;
    __syncthreads();
    if (tid < 16 && co0 + tid < CO)
      db_ws[(long)ms * CO + co0 + tid] = lds_red[tid] + lds_red[16 + tid] +
                                         lds_red[32 + tid] + lds_red[48 + tid];
  }
}

// ---------------------------------------------------------------------------
// Fused MaxPool2d(2,2) + ReLU, NHWC bf16. One thread per (n,oh,ow,8-channel
// group): 4 x 16B loads, packed max, relu; u8 argmax per channel (stored as
// one 8-byte group). Backward scatters into the exclusively-owned 2x2 window.
// ---------------------------------------------------------------------------
__global__ void maxpool2x2_relu_bf16_fwd_k(const bf16* __restrict__ x,
                                           bf16* __restrict__ y,
                                           unsigned char* __restrict__ idx,
                                           int N, int H, int W, int Cp,
                                           int OH, int OW) {
  long total = (long)N * OH * OW * (Cp / 8);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int cg = i % (Cp / 8);
    long r = i / (Cp / 8);
    int ow = r % OW;
    long r2 = r / OW;
    int oh = r2 % OH;
    int n = r2 / OH;
    const bf16* p = x + (((long)(n * H + oh * 2) * W + ow * 2) * Cp) + cg * 8;
    bf16x8 v0 = *reinterpret_cast<const bf16x8*>(p);
    bf16x8 v1 = *reinterpret_cast<const bf16x8*>(p + Cp);
    bf16x8 v2 = *reinterpret_cast<const bf16x8*>(p + (long)W * Cp);
    bf16x8 v3 = *reinterpret_cast<const bf16x8*>(p + (long)W * Cp + Cp);
    bf16x8 out;
    unsigned char am[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f0 = to_f32(v0[j]), f1 = to_f32(v1[j]);
      float f2 = to_f32(v2[j]), f3 = to_f32(v3[j]);
      float m01 = fmaxf(f0, f1), m23 = fmaxf(f2, f3);
      float m = fmaxf(m01, m23);
      unsigned char a;
      if (m == f0) a = 0; else if (m == f1) a = 1;
      else if (m == f2) a = 2; else a = 3;
      out[j] = to_bf16(fmaxf(m, 0.f));
      am[j] = a;
    }
    long yoff = (((long)(n * OH + oh) * OW + ow) * Cp) + cg * 8;
    *reinterpret_cast<bf16x8*>(y + yoff) = out;
    *reinterpret_cast<unsigned long long*>(idx + yoff) =
        *reinterpret_cast<unsigned long long*>(am);
  }
}

__global__ void maxpool2x2_relu_bf16_bwd_k(const bf16* __restrict__ dy,
                                           const bf16* __restrict__ y,
                                           const unsigned char* __restrict__ idx,
                                           bf16* __restrict__ dx,
                                           int N, int H, int W, int Cp,
                                           int OH, int OW) {
  long total = (long)N * OH * OW * (Cp / 8);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int cg = i % (Cp / 8);
    long r = i / (Cp / 8);
    int ow = r % OW;
    long r2 = r / OW;
    int oh = r2 % OH;
    int n = r2 / OH;
    long yoff = (((long)(n * OH + oh) * OW + ow) * Cp) + cg * 8;
    bf16x8 g = *reinterpret_cast<const bf16x8*>(dy + yoff);
    bf16x8 yv = *reinterpret_cast<const bf16x8*>(y + yoff);
    unsigned long long am8 =
        *reinterpret_cast<const unsigned long long*>(idx + yoff);
    bf16x8 o0 = bf8_zero(), o1 = bf8_zero(), o2 = bf8_zero(), o3 = bf8_zero();
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gv = (to_f32(yv[j]) > 0.f) ? to_f32(g[j]) : 0.f;
      unsigned char a = (am8 >> (8 * j)) & 0xff;
      bf16 bg = to_bf16(gv);
      if (a == 0) o0[j] = bg; else if (a == 1) o1[j] = bg;
      else if (a == 2) o2[j] = bg; else o3[j] = bg;
    }
    bf16* p = dx + (((long)(n * H + oh * 2) * W + ow * 2) * Cp) + cg * 8;
    *reinterpret_cast<bf16x8*>(p) = o0;
    *reinterpret_cast<bf16x8*>(p + Cp) = o1;
    *reinterpret_cast<bf16x8*>(p + (long)W * Cp) = o2;
    *reinterpret_cast<bf16x8*>(p + (long)W * Cp + Cp) = o3;
  }
}

// ---------------------------------------------------------------------------
// Dropout bf16, vectorized 8-wide; one mix64 hash yields 8 keep-bytes.
// channelwise (Dropout2d on NHWC): rng unit = (n, c) — computed per element
// of the vector (the 8 lanes are 8 consecutive channels of one pixel).
// mask: u8 per element (elementwise) or per (n,c) (channelwise).
// ---------------------------------------------------------------------------
__global__ void bump_u64_k(unsigned long long* ctr) {
  if (threadIdx.x == 0 && blockIdx.x == 0) *ctr += 1;
}

__global__ void dropout_bf16_fwd_k(const bf16* __restrict__ x,
                                   bf16* __restrict__ y,
                                   unsigned char* __restrict__ mask,
                                   long n8, int Cp, long hwc, float p,
                                   int channelwise,
                                   unsigned long long seed,
                                   const unsigned long long* __restrict__ ctr_p) {
  unsigned long long ctr = *ctr_p;
  float inv = 1.f / (1.f - p);
  unsigned int pthresh = (unsigned int)(p * 256.f);
  for (long i8 = (long)blockIdx.x * blockDim.x + threadIdx.x; i8 < n8;
       i8 += (long)gridDim.x * blockDim.x) {
    long i = i8 * 8;
    bf16x8 v = *reinterpret_cast<const bf16x8*>(x + i);
    unsigned long long h;
    if (channelwise) {
      // unit = (n, cgroup): one hash covers the 8 consecutive channels
      long nimg = i / hwc;
      int c8 = (int)((i % Cp) / 8);
      h = mix64(seed ^ mix64(ctr ^ mix64((unsigned long long)(nimg * (Cp / 8) + c8))));
    } else {
      h = mix64(seed ^ mix64(ctr ^ mix64((unsigned long long)i8)));
    }
    bf16x8 o;
    unsigned char mk[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      bool keep = ((h >> (8 * j)) & 0xff) >= pthresh;
      mk[j] = keep ? 1 : 0;
      o[j] = keep ? to_bf16(to_f32(v[j]) * inv) : to_bf16(0.f);
    }
    *reinterpret_cast<bf16x8*>(y + i) = o;
    if (channelwise) {
      // mask indexed [n][Cp]: only the first pixel of each image writes it
      long nimg = i / hwc;
      if (i % hwc < Cp) {
        int c0 = (int)(i % Cp);
        for (int j = 0; j < 8; ++j) mask[nimg * Cp + c0 + j] = mk[j];
      }
    } else {
      *reinterpret_cast<unsigned long long*>(mask + i) =
          *reinterpret_cast<unsigned long long*>(mk);
    }
  }
}

__global__ void dropout_bf16_bwd_k(const bf16* __restrict__ dy,
                                   const unsigned char* __restrict__ mask,
                                   bf16* __restrict__ dx, long n8, int Cp,
                                   long hwc, float p, int channelwise) {
  float inv = 1.f / (1.f - p);
  for (long i8 = (long)blockIdx.x * blockDim.x + threadIdx.x; i8 < n8;
       i8 += (long)gridDim.x * blockDim.x) {
    long i = i8 * 8;
    bf16x8 v = *reinterpret_cast<const bf16x8*>(dy + i);
    unsigned char mk[8];
    if (channelwise) {
      long nimg = i / hwc;
      int c0 = (int)(i % Cp);
      for (int j = 0; j < 8; ++j) mk[j] = mask[nimg * Cp + c0 + j];
    } else {
      *reinterpret_cast<unsigned long long*>(mk) =
          *reinterpret_cast<const unsigned long long*>(mask + i);
    }
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = mk[j] ? to_bf16(to_f32(v[j]) * inv) : to_bf16(0.f);
    *reinterpret_cast<bf16x8*>(dx + i) = o;
  }
}

// ---------------------------------------------------------------------------
// LogSoftmax + mean NLL, bf16 logits [M][COp] (only c < C are real columns).
// logp fp32 [M][C] (the user-facing reference interface); bwd emits bf16
// dx [M][COp] with zeroed padding columns.
// ---------------------------------------------------------------------------
__global__ void logsoftmax_nll_bf16_fwd_k(const bf16* __restrict__ x,
                                          const long* __restrict__ target,
                                          float* __restrict__ logp,
                                          float* __restrict__ loss,
                                          int M, int COp, int C, int with_loss) {
  __shared__ float red[256];
  float part = 0.f;
  for (long m = (long)blockIdx.x * blockDim.x + threadIdx.x; m < M;
       m += (long)gridDim.x * blockDim.x) {
    const bf16* row = x + m * COp;
    float mx = to_f32(row[0]);
    for (int c = 1; c < C; ++c) mx = fmaxf(mx, to_f32(row[c]));
    float s = 0.f;
    for (int c = 0; c < C; ++c) s += __expf(to_f32(row[c]) - mx);
    float lse = mx + __logf(s);
    float* out = logp + m * C;
    for (int c = 0; c < C; ++c) out[c] = to_f32(row[c]) - lse;
    if (with_loss) part -= out[target[m]];
  }
  if (with_loss) {
    red[threadIdx.x] = part;
    __syncthreads();
    for (int s = 128; s > 0; s >>= 1) {
      if (threadIdx.x < s) red[threadIdx.x] += red[threadIdx.x + s];
      __syncthreads();
    }
    if (threadIdx.x == 0) atomicAdd(loss, red[0] / (float)M);
  }
}

__global__ void logsoftmax_nll_bf16_bwd_k(const float* __restrict__ logp,
                                          const long* __restrict__ target,
                                          const float* __restrict__ dloss,
                                          bf16* __restrict__ dx, int M,
                                          int COp, int C) {
  float scale = dloss[0] / (float)M;
  for (long m = (long)blockIdx.x * blockDim.x + threadIdx.x; m < M;
       m += (long)gridDim.x * blockDim.x) {
    const float* lp = logp + m * C;
    bf16* out = dx + m * COp;
    long t = target[m];
    for (int c = 0; c < COp; ++c) {
      float v = 0.f;
      if (c < C) v = (__expf(lp[c]) - (c == t ? 1.f : 0.f)) * scale;
      out[c] = to_bf16(v);
    }
  }
}

// tr-read mapping probe: img [64][16] f32 -> out[l][j] should be
// img[lk*8+j][lm] if tr_read_8m implements A[row=col-idx][m=lk*8+j].
__global__ void tr16_probe_k(const float* __restrict__ img,
                             float* __restrict__ out) {
  __shared__ bf16 lds[64][16];
  int l = threadIdx.x;
  if (l >= 64) return;
  for (int i = l; i < 64 * 16; i += 64) lds[i / 16][i % 16] = to_bf16(img[i]);
  __syncthreads();
  int lm = l & 15, lk = l >> 4;
  bf16x8 a = tr_read_8m(&lds[0][0], l);
  (void)lm;
  for (int j = 0; j < 8; ++j) out[l * 8 + j] = to_f32(a[j]);
}

// ---------------------------------------------------------------------------
// Fused stem: Conv2d(CI<=2, s=1, p=0) + MaxPool2x2 + ReLU in ONE kernel.
// For LeNet conv1 (CI=1) the MFMA path pays 8x padding waste and three full
// activation round-trips (y1 store, pool read, pool store = 2x151MB + 38MB at
// batch 8192); this kernel writes only the pooled output + argmax, computing
// the 4 conv pixels per pool cell directly on the VALU (25*CI FMAs each —
// conv1 is ~1.9 GFLOP at batch 8192, trivial next to the saved bandwidth).
// Thread = (n, pool cell, 8-co group). Backward = conv2d_wgrad_pool_bf16
// (no dgrad: the stem is the input layer).
// ---------------------------------------------------------------------------
template <int CI, int KH, int KW>
__global__ __launch_bounds__(256) void conv_pool_stem_bf16_k(
    const bf16* __restrict__ x, const float* __restrict__ w,
    const float* __restrict__ bias, bf16* __restrict__ p,
    unsigned char* __restrict__ idx,
    int CIp, int H, int W, int COp, int CO,
    int OH2, int OW2, int N) {
  // weights in LDS, fp32: [co][CI*KH*KW]
  extern __shared__ float lds_wf[];
  const int tid = threadIdx.x;
  const int Kc = CI * KH * KW;
  for (int i = tid; i < COp * Kc; i += 256) {
    int co = i / Kc, k = i % Kc;
    lds_wf[i] = (co < CO) ? w[co * Kc + k] : 0.f;
  }
  __syncthreads();

  // thread = (n, pool cell); loops over ALL co groups so the x tile loads
  // once (round 1 re-loaded it per 8-co group), with bf16x2 vector row
  // reads when CIp==1 (guide G13: scalar bf16 loads cost ~2x)
  using bf16x2 = __attribute__((ext_vector_type(2))) __bf16;
  long total = (long)N * OH2 * OW2;
  for (long t = (long)blockIdx.x * blockDim.x + tid; t < total;
       t += (long)gridDim.x * blockDim.x) {
    int ow2 = t % OW2;
    long r2 = t / OW2;
    int oh2 = r2 % OH2;
    int n = r2 / OH2;

    // x tile (KH+1) x (KW+1) x CI in registers; all loop bounds are
    // compile-time so xt stays in VGPRs (guide rule 20: runtime-indexed
    // arrays spill to scratch)
    constexpr int TH = KH + 1, TW = KW + 1;
    float xt[CI * TH * TW];
    const bf16* xb = x + ((long)(n * H + oh2 * 2) * W + ow2 * 2) * CIp;
    if (CI == 1 && TW % 2 == 0 && CIp == 1) {
      // CIp==1 (runtime!) row is contiguous; base is 4-B aligned (even
      // pixel index). Padded inputs (CIp=8, from nchw_to_nhwc_padded) take
      // the strided scalar branch below.
#pragma unroll
      for (int ty = 0; ty < TH; ++ty)
#pragma unroll
        for (int tx = 0; tx < TW; tx += 2) {
          bf16x2 v = *reinterpret_cast<const bf16x2*>(&xb[ty * W + tx]);
          xt[ty * TW + tx] = to_f32(v[0]);
          xt[ty * TW + tx + 1] = to_f32(v[1]);
        }
    } else {
#pragma unroll
      for (int ci = 0; ci < CI; ++ci)
#pragma unroll
        for (int ty = 0; ty < TH; ++ty)
#pragma unroll
          for (int tx = 0; tx < TW; ++tx)
            xt[(ci * TH + ty) * TW + tx] = to_f32(xb[(ty * W + tx) * CIp + ci]);
    }

    long off0 = (((long)(n * OH2 + oh2) * OW2 + ow2) * COp);
    for (int cg = 0; cg < COp / 8; ++cg) {
      bf16x8 pout;
      unsigned char am[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int co = cg * 8 + j;
        float acc[4];
        float bval = (co < CO) ? bias[co] : 0.f;
        acc[0] = acc[1] = acc[2] = acc[3] = bval;
        if (co < CO) {
          const float* wr = &lds_wf[co * Kc];
#pragma unroll
          for (int ci = 0; ci < CI; ++ci)
#pragma unroll
            for (int kh = 0; kh < KH; ++kh)
#pragma unroll
              for (int kw = 0; kw < KW; ++kw) {
                float wv = wr[(ci * KH + kh) * KW + kw];
                const float* xrow = &xt[(ci * TH + kh) * TW + kw];
                acc[0] += wv * xrow[0];
                acc[1] += wv * xrow[1];
                acc[2] += wv * xrow[TW];
                acc[3] += wv * xrow[TW + 1];
              }
        }
        float m01 = fmaxf(acc[0], acc[1]), m23 = fmaxf(acc[2], acc[3]);
        float mx = fmaxf(m01, m23);
        unsigned char a;
        if (mx == acc[0]) a = 0; else if (mx == acc[1]) a = 1;
        else if (mx == acc[2]) a = 2; else a = 3;
        pout[j] = to_bf16((co < CO) ? fmaxf(mx, 0.f) : 0.f);
        am[j] = a;
      }
      long off = off0 + cg * 8;
      *reinterpret_cast<bf16x8*>(p + off) = pout;
      *reinterpret_cast<unsigned long long*>(idx + off) =
          *reinterpret_cast<unsigned long long*>(am);
    }
  }
}

// ---------------------------------------------------------------------------
// wgrad for the fused stem: identical reduction to conv2d_wgrad_bf16 but the
// conv-space dy is reconstructed on the fly from (d_pool, pool_y, idx):
//   dy_conv[n,oh,ow,co] = (idx==quadrant && pool_y>0) ? d_pool : 0.
// ---------------------------------------------------------------------------
template <int KT, int MC = 64>
__global__ __launch_bounds__(256) void conv2d_wgrad_pool_bf16_k(
    const bf16* __restrict__ dpool, const bf16* __restrict__ pool_y,
    const unsigned char* __restrict__ pidx, const bf16* __restrict__ x,
    float* __restrict__ dw_ws, float* __restrict__ db_ws,
    int CIp, int CI, int H, int W, int COp, int CO, int KH, int KW,
    int OH, int OW, int M, int K, int Kcan, int m_per_split, long x_elems) {
  // k-space uses PER-kh-ROW padding: KROW = ceil(KW*CIp/8)*8, K = KH*KROW,
  // so every 8-k A-fragment stays inside one kh row of the patch even when
  // KW*CIp is not a multiple of 8 (the stem's CIp can be 1). Pad k entries
  // read in-row garbage (load base clamped to the tensor) and their dw rows
  // are simply never written back.
  // KT sized so 4*KT*16 >= K with minimal waste (LeNet stem: K=40 -> KT=1;
  // round-1 ran KT=2 = 128 k-slots, 69% dead MFMA/staging).
  constexpr int NKF = 4 * KT;
  const int k0 = blockIdx.x * (NKF * 16);
  const int co0 = blockIdx.y * 16;
  const int ms = blockIdx.z;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lm = l & 15;
  const int lk = l >> 4;
  const int OH2 = OH / 2, OW2 = OW / 2;

  __shared__ bf16 lds_x[NKF][MC][16];
  __shared__ bf16 lds_dy[MC][16];
  __shared__ int lds_mbase[MC];
  __shared__ int lds_mpool[MC];   // pooled-offset | quadrant (COp mult of 8)
  // decode table in LDS (a per-thread array would be runtime-indexed in
  // the staging loop -> scratch, guide rule 20)
  __shared__ int lds_goff[NKF * 2];

  const int KROWV = KW * CIp;                 // valid entries per kh row
  const int KROW = ((KROWV + 7) / 8) * 8;     // padded row length
  const int OHOW = OH * OW;

  if (tid < NKF * 2) {
    int g = tid;
    int k = k0 + g * 8;
    int kcl = k < K ? k : 0;
    int kh = kcl / KROW, r = kcl % KROW;
    lds_goff[g] = kh * W * CIp + r;           // r may run past the real row
  }

  f32x4 acc[KT] = {};
  float db_part = 0.f;

  int m_begin = ms * m_per_split;
  int m_end = min(M, m_begin + m_per_split);

  // buffer-load staging measured SLOWER here (128 -> 138 us,
  // gpurun_out/micro_r2u.txt) unlike every other staging loop: keep the
  // predicated path with the per-element tail
  const bool use_buf = false;
  buf_rsrc_t xr = make_bf16_rsrc(x, 8);
  (void)xr;

  for (int mc = m_begin; mc < m_end; mc += MC) {
    __syncthreads();
    for (int mi = tid; mi < MC; mi += 256) {
      int m = mc + mi;
      int mm = m < m_end ? m : 0;
      int n = mm / OHOW, r = mm % OHOW;
      int oh = r / OW, ow = r % OW;
      lds_mbase[mi] = (n * H + oh) * W * CIp + ow * CIp;  // pad=0, s=1
      int q = (oh & 1) * 2 + (ow & 1);
      lds_mpool[mi] =
          (int)((((long)(n * OH2 + oh / 2) * OW2 + ow / 2) * COp)) | q;
    }
    __syncthreads();
    for (int idx2 = tid; idx2 < NKF * MC * 2; idx2 += 256) {
      int half = idx2 & 1;
      int rest = idx2 >> 1;
      int mloc = rest % MC;
      int kf = rest / MC;
      int g = kf * 2 + half;
      int m = mc + mloc;
      int k = k0 + g * 8;
      bool ok = m < m_end && k < K;
      long a = (long)lds_mbase[mloc] + lds_goff[g];
      bf16x8 v;
      if (use_buf) {
        // hardware bounds-check handles BOTH the dead-m mask and the
        // per-element tensor-tail cut (the round-1 whole-vector clamp
        // shifted still-valid k slots); loads stay countable
        v = buf_load8(xr, a, ok);
      } else {
        v = bf8_zero();
        if (ok) {
          if (a + 8 > x_elems) {
#pragma unroll
            for (int j = 0; j < 8; ++j)
              v[j] = (a + j < x_elems) ? x[a + j] : to_bf16(0.f);
          } else {
            v = *reinterpret_cast<const bf16x8*>(&x[a]);
          }
        }
      }
      *reinterpret_cast<bf16x8*>(&lds_x[kf][mloc][half * 8]) = v;
    }
    for (int idx2 = tid; idx2 < MC * 2; idx2 += 256) {
      int half = idx2 & 1;
      int mloc = idx2 >> 1;
      int m = mc + mloc;
      bf16x8 v = bf8_zero();
      if (m < m_end && co0 + half * 8 < COp) {
        int mp = lds_mpool[mloc];
        unsigned char q = (unsigned char)(mp & 3);
        long off = (long)(mp & ~3) + co0 + half * 8;
        bf16x8 dp = *reinterpret_cast<const bf16x8*>(&dpool[off]);
        bf16x8 py = *reinterpret_cast<const bf16x8*>(&pool_y[off]);
        unsigned long long am8 =
            *reinterpret_cast<const unsigned long long*>(&pidx[off]);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          bool take = (((am8 >> (8 * j)) & 0xff) == q) && (to_f32(py[j]) > 0.f);
          v[j] = take ? dp[j] : to_bf16(0.f);
        }
      }
      *reinterpret_cast<bf16x8*>(&lds_dy[mloc][half * 8]) = v;
    }
    __syncthreads();

#pragma unroll
    for (int mm = 0; mm < MC; mm += 32) {
      bf16x8 b = tr_read_8m(&lds_dy[mm][0], l);
      if (wid == 0) {
#pragma unroll
        for (int j = 0; j < 8; ++j) db_part += to_f32(b[j]);
      }
#pragma unroll
      for (int kf = 0; kf < KT; ++kf) {
        bf16x8 a = tr_read_8m(&lds_x[wid * KT + kf][mm][0], l);
        acc[kf] = mfma16x16x32bf(a, b, acc[kf]);
      }
    }
  }

  {
    float* slab = dw_ws + (long)ms * (CO * (long)Kcan);
#pragma unroll
    for (int kf = 0; kf < KT; ++kf) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        int kk = k0 + (wid * KT + kf) * 16 + lk * 4 + reg;
        int co = co0 + lm;
        if (kk < K && co < CO) {
          int kh = kk / KROW, r = kk % KROW;
          if (r < KROWV) {                     // pad k rows never written
            int kw = r / CIp, ci = r % CIp;
            if (ci < CI)
              slab[((co * CI) + ci) * (KH * KW) + kh * KW + kw] = acc[kf][reg];
          }
        }
      }
    }
  }
  if (blockIdx.x == 0 && wid == 0) {
    db_part += __shfl_down(db_part, 16);
    db_part += __shfl_down(db_part, 32);
    if (lk == 0 && co0 + lm < CO) db_ws[(long)ms * CO + co0 + lm] = db_part;
  }
}

// ---------------------------------------------------------------------------
// LDS-tiled conv, stride 1 (fwd and dgrad), NHWC bf16 — input-stationary.
// A block owns an 8x8 output-pixel tile of one image and NW*16 output
// channels; the input region ((8+KH-1)^2 x up-to-64 input channels) is staged
// into LDS ONCE per ci-chunk, so every im2col re-read of a pixel (KH*KW taps)
// comes from LDS instead of L2 (the gather engine's bound). Weights come from
// a per-call bf16 repack w_rp[tap][out_ch][in_ch] (repack_w_k below): B-frags
// are direct 16-byte loads from L1/L2-resident global memory — no LDS weight
// stage, no per-tap barriers. DGRAD reuses the same kernel: the repack writes
// [flipped tap][ci][co], and an s1 KxK dgrad is an s1 KxK conv with
// pad' = K-1-pad.
// ---------------------------------------------------------------------------
__global__ void repack_w_k(const float* __restrict__ w, bf16* __restrict__ w_rp,
                           int CO, int CI, int KHKW, int Ap, int Bp,
                           int dgrad) {
  // fwd:  w_rp[tap][co][ci] = w[co][ci][tap]        (Ap=COp, Bp=CIp)
  // dgrad: w_rp[KHKW-1-tap][ci][co] = w[co][ci][tap] (Ap=CIp, Bp=COp)
  // Iterates the PADDED destination extent and writes zeros at pad positions
  // itself (a garbage inf in a pad column would turn 0*inf into NaN inside
  // valid accumulators) — the destination can be at::empty, no separate fill
  // launch. Consecutive i => consecutive Bp index, so stores stay coalesced.
  long total = (long)KHKW * Ap * Bp;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int b = i % Bp;
    long r = i / Bp;
    int a = r % Ap;
    int tap_d = r / Ap;
    int co = dgrad ? b : a;
    int ci = dgrad ? a : b;
    int tap = dgrad ? KHKW - 1 - tap_d : tap_d;
    bf16 v = to_bf16(0.f);
    if (co < CO && ci < CI) v = to_bf16(w[((long)co * CI + ci) * KHKW + tap]);
    w_rp[i] = v;
  }
}

template <int KH, bool HAS_BIAS, bool RELU, int NW, int MW, int NIMG = 1,
          int S = 1, bool AFF = false, bool BGLDS = false>
__global__ __launch_bounds__(256) void conv_tiled_s1_bf16_k(
    const bf16* __restrict__ x, const bf16* __restrict__ w_rp,
    const float* __restrict__ bias, bf16* __restrict__ y,
    int CIp, int H, int W, int COp, int COcan,
    int OH, int OW, int pad, int N, int tiles_h, int tiles_w,
    const float* __restrict__ aff = nullptr) {
  // BGLDS: the next tap's 8 KB B panel streams into a 2-deep LDS ring via
  // global_load_lds WHILE the current tap's MFMAs run — the per-tap B
  // loads' L2 latency was 70% wave-parked time (pmc_micro_r2.txt). The
  // panel image is chunk-XOR-swizzled (chunk ^= (co>>1)&7) on BOTH the
  // glds source index and the read so the b128 reads stay conflict-free
  // without per-row padding (glds destinations must be lane-linear).
  // Costs one LDS ring (16 KB -> occupancy 3->2 blocks/CU) + a raw
  // barrier per tap; requires full 64-ci chunks and NW==4.
  // AFF: x is a RAW conv output whose BatchNorm(+ReLU) apply is fused into
  // this staging pass — v = bf16(max(aff[c]*v + aff[CIp+c], 0)); padded
  // channels have aff == 0 so they stay zero. Saves the separate bn_apply
  // write + re-read of the whole activation (docs/ROADMAP.md BN fusion).
  // output tile: 8 rows x (8*MW/NIMG) cols x NIMG images; a wave owns MW
  // 16-pixel fragments (split across columns and images), so every
  // B-fragment read feeds MW MFMAs. NIMG>1 keeps the reuse on small maps
  // (ResNet layer3/4) where wide column tiles would be mostly padding.
  // S=2 (ResNet layer2/3 block-0 3x3 convs): the region grows to
  // (T-1)*2+KH per axis, so the ci chunk halves (CC=32) to stay inside the
  // 64 KB static-LDS block limit.
  constexpr int TH = 8, TW = 8 * (MW / NIMG);
  constexpr int RH = (TH - 1) * S + KH, RW = (TW - 1) * S + KH;
  constexpr int CC = (S == 1) ? 64 : 32;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lm = l & 15;
  const int lk = l >> 4;

  int tw = blockIdx.x % tiles_w;
  int rest = blockIdx.x / tiles_w;
  int th = rest % tiles_h;
  int n = (rest / tiles_h) * NIMG;   // first image of this block's group
  const int oh0 = th * TH, ow0 = tw * TW;
  const int co0 = blockIdx.y * (NW * 16);

  // per-pixel stride padded +8 bf16 (bank spread for the 16-pixel b128
  // reads); ONE shared object (a second one would force vmcnt(0) before
  // every ds_read beside an in-flight glds — guide §5 trap 4a): the BGLDS
  // B-panel ring is carved off the tail
  __shared__ bf16 lds_x[NIMG * RH * RW * (CC + 8) + 8 +
                        (BGLDS ? 2 * 64 * CC : 0)];
  bf16* ldsB = &lds_x[NIMG * RH * RW * (CC + 8) + 8];

  // wave pixel map: fragment mi covers a 2-row x 8-col patch; the block's 4
  // waves cover the 8 rows. pixel (py, px) for (wave, mi, lm):
  //   py = wid*2 + (lm>>3), px = mi*8 + (lm&7)
  const int py = wid * 2 + (lm >> 3);

  f32x4 acc[MW][NW] = {};

  constexpr int KHKW = KH * KH;
  const long x_elems = (long)N * H * W * CIp;
  const bool use_buf = rsrc_fits(x_elems);
  buf_rsrc_t xr = make_bf16_rsrc(x, x_elems);
  for (int cic = 0; cic < CIp; cic += CC) {
    int cc = min(CC, CIp - cic);
    int ccp = cc + 8;
    __syncthreads();
    for (int idx = tid; idx < NIMG * RH * RW * (cc / 8); idx += 256) {
      int g8 = idx % (cc / 8);
      int p = idx / (cc / 8);
      int rx = p % RW;
      int rest2 = p / RW;
      int ry = rest2 % RH;
      int img = rest2 / RH;
      int ih = oh0 * S - pad + ry, iw = ow0 * S - pad + rx;
      bool ok = ih >= 0 && ih < H && iw >= 0 && iw < W && n + img < N;
      long off = ((long)((n + img) * H + ih) * W + iw) * CIp + cic + g8 * 8;
      bf16x8 v;
      if (use_buf) {
        // unconditional hardware-bounds-checked load: keeps the staging
        // loop countable (no per-iteration vmcnt(0) drain — see buf_load8)
        v = buf_load8(xr, off, ok);
      } else {
        v = bf8_zero();
        if (ok) v = *reinterpret_cast<const bf16x8*>(&x[off]);
      }
      if (AFF && ok) {
        const float* sc = &aff[cic + g8 * 8];
        const float* sh = &aff[CIp + cic + g8 * 8];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          v[j] = to_bf16(fmaxf(to_f32(v[j]) * sc[j] + sh[j], 0.f));
      }
      *reinterpret_cast<bf16x8*>(
          &lds_x[(img * RH * RW + ry * RW + rx) * ccp + g8 * 8]) = v;
    }
    __syncthreads();
    // glds one 8 KB tap panel: each wave DMAs two 1 KiB pieces (dest =
    // wave-uniform base + lane*16; the SOURCE index carries the XOR
    // swizzle so reads stay conflict-free on a pad-free panel)
    auto stage_B = [&](int tap, int buf) {
      if constexpr (BGLDS) {
        constexpr int NCH = CC / 8;            // 16-B chunks per co row
        const bf16* wt = w_rp + ((long)tap * COp + co0) * CIp + cic;
        bf16* panel = ldsB + buf * (64 * CC);
#pragma unroll
        for (int i = 0; i < CC / 32; ++i) {
          int s = (wid * (CC / 32) + i) * 64 + l;  // 16-B slot index
          int co = s / NCH;
          int chlog = (s % NCH) ^ ((co >> 1) & (NCH - 1));
          const bf16* src = &wt[(long)co * CIp + chlog * 8];
          __builtin_amdgcn_global_load_lds(
              (const __attribute__((address_space(1))) unsigned int*)src,
              (__attribute__((address_space(3))) unsigned int*)(panel +
                                                               (wid * (CC / 32) +
                                                                i) * 512),
              16, 0, 0);
        }
      }
    };
    if (BGLDS) {
      // 3-deep ring: panels for taps t+1 and t+2 stay in flight across
      // the barrier (counted vmcnt at the tail, guide T3/T4 — never
      // drain to 0 in the main loop)
      stage_B(0, 0);
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
#pragma unroll
    for (int tap = 0; tap < KHKW; ++tap) {
      int kh = tap / KH, kw = tap % KH;
      const bf16* wtap = w_rp + ((long)tap * COp + co0) * CIp + cic;
      const bf16* panel = BGLDS ? ldsB + (tap & 1) * (64 * CC) : nullptr;
      if (BGLDS && tap + 1 < KHKW) stage_B(tap + 1, (tap + 1) & 1);
      // compile-time trip count (remainder chunks handled by the gin
      // predicate): the whole tap x ks nest fully unrolls, so the compiler
      // can hoist and software-pipeline the A/B loads across taps
#pragma unroll
      for (int ks = 0; ks < CC; ks += 32) {
        bool gin = lk * 8 < cc - ks;
        if constexpr (NIMG != 2) {
          // ALL MW A-reads issue BEFORE the MFMA block: the round-1
          // read-then-consume-per-fragment order made hipcc drain
          // lgkmcnt(0) before every 4-MFMA group (~50 exposed LDS cycles
          // per ~70 cycles of matrix work — the measured 19% MfmaUtil;
          // asm audit in the commit). Loads are unconditional at a
          // clamped offset + post-select. Measured: l1 fwd 230->209,
          // l2 208, dgrad l1 175->160 us. The NIMG==2 instantiation
          // REGRESSED with this order (194->234) and keeps the old body.
          int bsel = gin ? ks + lk * 8 : 0;
          bf16x8 bw[NW];
          if constexpr (BGLDS) {
            // B from the prefetched LDS panel (chunk-XOR-swizzled read)
            int chlog = (ks >> 3) + lk;
#pragma unroll
            for (int ni = 0; ni < NW; ++ni) {
              int co = ni * 16 + lm;
              int chst = chlog ^ ((co >> 1) & (CC / 8 - 1));
              bw[ni] = *reinterpret_cast<const bf16x8*>(
                  &panel[co * CC + chst * 8]);
            }
          } else {
#pragma unroll
            for (int ni = 0; ni < NW; ++ni) {
              bf16x8 t = *reinterpret_cast<const bf16x8*>(
                  &wtap[(long)(ni * 16 + lm) * CIp + bsel]);
              bw[ni] = gin ? t : bf8_zero();
            }
          }
          bf16x8 av[MW];
#pragma unroll
          for (int mi = 0; mi < MW; ++mi) {
            int px = (mi % (MW / NIMG)) * 8 + (lm & 7);
            int img = mi / (MW / NIMG);
            const bf16* xpix =
                &lds_x[(img * RH * RW + (py * S + kh) * RW + (px * S + kw)) *
                       ccp];
            bf16x8 t = *reinterpret_cast<const bf16x8*>(&xpix[bsel]);
            av[mi] = gin ? t : bf8_zero();
          }
#pragma unroll
          for (int mi = 0; mi < MW; ++mi)
#pragma unroll
            for (int ni = 0; ni < NW; ++ni)
              acc[mi][ni] = mfma16x16x32bf(av[mi], bw[ni], acc[mi][ni]);
        } else {
          bf16x8 bw[NW];
          if constexpr (BGLDS) {
            int chlog = (ks >> 3) + lk;
#pragma unroll
            for (int ni = 0; ni < NW; ++ni) {
              int co = ni * 16 + lm;
              int chst = chlog ^ ((co >> 1) & (CC / 8 - 1));
              bw[ni] = *reinterpret_cast<const bf16x8*>(
                  &panel[co * CC + chst * 8]);
            }
          } else {
#pragma unroll
            for (int ni = 0; ni < NW; ++ni)
              bw[ni] = gin ? *reinterpret_cast<const bf16x8*>(
                                 &wtap[(long)(ni * 16 + lm) * CIp + ks + lk * 8])
                           : bf8_zero();
          }
#pragma unroll
          for (int mi = 0; mi < MW; ++mi) {
            int px = (mi % (MW / NIMG)) * 8 + (lm & 7);
            int img = mi / (MW / NIMG);
            const bf16* xpix =
                &lds_x[(img * RH * RW + (py * S + kh) * RW + (px * S + kw)) *
                       ccp];
            bf16x8 a = gin ? *reinterpret_cast<const bf16x8*>(&xpix[ks + lk * 8])
                           : bf8_zero();
#pragma unroll
            for (int ni = 0; ni < NW; ++ni)
              acc[mi][ni] = mfma16x16x32bf(a, bw[ni], acc[mi][ni]);
          }
        }
      }
      if (BGLDS) {
        // depth 1: drain + raw barrier per tap (a 3-deep ring with
        // counted vmcnt measured WORSE — 12.98 vs 12.53 ms on the b256
        // step; the next tap's panel has the whole MFMA block to fly)
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
      }
    }
  }

  // epilogue: D row = lk*4+reg indexes the fragment's 16-pixel space
#pragma unroll
  for (int mi = 0; mi < MW; ++mi) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int mo = lk * 4 + reg;                 // 0..15 within fragment
      int oy = oh0 + wid * 2 + (mo >> 3);
      int ox = ow0 + (mi % (MW / NIMG)) * 8 + (mo & 7);
      int nimg = n + mi / (MW / NIMG);
      if (oy >= OH || ox >= OW || nimg >= N) continue;
      long yb = ((long)(nimg * OH + oy) * OW + ox) * COp;
#pragma unroll
      for (int ni = 0; ni < NW; ++ni) {
        int co = co0 + ni * 16 + lm;
        if (co >= COp) continue;
        float v = (co < COcan) ? acc[mi][ni][reg] : 0.f;
        if (HAS_BIAS && co < COcan) v += bias[co];
        if (RELU) v = fmaxf(v, 0.f);
        y[yb + co] = to_bf16(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Region-stationary STEM wgrad (7x7 s2, <=4 real input channels, pad 3,
// H==2*OH): the generic engine's im2col gather reads M*K*2B ~ 2.5 GB here
// (49x the raw input) — the single most expensive kernel of the ResNet
// step at 918 us. Channels pack to 4 (as in the stem fwd) so one tap ROW
// is k=(kw*4+ci) padded to 32 = two 16-k fragments; a block owns all
// 7x32 k and 64 co, marching m in 96-pixel chunks of ONE output row
// (OW >= MC guarantees a chunk spans <= 2 rows -> 9 input rows + pad
// region, 17 KB). dy fragments hoist to VGPRs once per chunk (tap-row
// invariant), per-kh A-images build LDS->LDS into the vacated dy planes
// (b64 pairs: the 8-byte pixel stride leaves vec8 alignment only at even
// columns). Same tr16 fragment recipe and partial-slab fold as the other
// wgrad engines.
// ---------------------------------------------------------------------------
template <int MC>
__global__ __launch_bounds__(256) void conv2d_wgrad_stem_bf16_k(
    const bf16* __restrict__ dy, const bf16* __restrict__ x,
    float* __restrict__ dw_ws, float* __restrict__ db_ws, int CI, int COp,
    int CO, int OH, int OW, int M, int Kcan, int m_per_split) {
  const int co0 = blockIdx.y * 64;
  const int ms = blockIdx.z;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lm = l & 15;
  const int Wi = 2 * OW;
  const int W2 = Wi + 6;  // 3 pad cols each side
  constexpr int PXROWS = 9, PXCOLS = 262;  // OW <= 128

  __shared__ bf16 lds_reg[PXROWS * PXCOLS * 4];  // [row][col][4ci]
  __shared__ bf16 lds_img_raw[4 * MC * 16 + 3 * 16];
  __shared__ int lds_mm[MC];  // packed (px<<8 | oy+1)
  auto plane = [&](int p) { return &lds_img_raw[p * (MC * 16 + 16)]; };
  constexpr int MSTEPS = MC / 32;

  const int m_begin = ms * m_per_split;
  const int m_end = min(M, m_begin + m_per_split);

  f32x4 acc[2][7] = {};  // [kf][kh]
  float db_part = 0.f;

  const long x_elems = 2L * (M / OW) * Wi * 8;
  const long dy_elems = (long)M * COp;
  const bool use_buf = rsrc_fits(x_elems) && rsrc_fits(dy_elems);
  buf_rsrc_t xr = make_bf16_rsrc(x, x_elems);
  buf_rsrc_t dyr = make_bf16_rsrc(dy, dy_elems);

  for (int mc = m_begin; mc < m_end; mc += MC) {
    const int g0 = mc / OW;
    const int mlast = min(mc + MC - 1, m_end - 1);
    const int rows = 2 * (mlast / OW - g0) + 7;
    __syncthreads();
    if (tid < MC) {
      int m = mc + tid;
      int mm = m < m_end ? m : mc;
      int g = mm / OW, ox = mm % OW;
      int px = (2 * (g - g0)) * PXCOLS + 2 * ox;  // kh/kw offsets added later
      int oy = g % OH;
      lds_mm[tid] = (px << 8) | (m < m_end ? oy + 1 : 0);
    }
    // x region rows gi in [2*g0-3, ...]: 4-packed channels, b64 stores
    // (buffer loads: countable, no per-stage vmcnt(0) — see buf_load8)
    for (int idx = tid; idx < rows * W2; idx += 256) {
      int c = idx % W2;
      int r = idx / W2;
      int gi = 2 * g0 - 3 + r;
      int iw = c - 3;
      bool ok = gi >= 0 && gi < 2 * (M / OW) && iw >= 0 && iw < Wi;
      long off = ((long)gi * Wi + iw) * 8;
      bf16x4 v;
      if (use_buf) {
        v = buf_load4(xr, off, ok);
      } else {
        v = bf16x4{};
        if (ok) v = *reinterpret_cast<const bf16x4*>(&x[off]);
      }
      *reinterpret_cast<bf16x4*>(&lds_reg[(r * PXCOLS + c) * 4]) = v;
    }
    // dy planes [ni][MC][16co]
    for (int idx = tid; idx < 4 * MC * 2; idx += 256) {
      int half = idx & 1;
      int rest = idx >> 1;
      int mloc = rest % MC;
      int ni = rest / MC;
      int m = mc + mloc;
      int cb = co0 + ni * 16 + half * 8;
      bool ok = m < m_end && cb < COp;
      bf16x8 v;
      if (use_buf) {
        v = buf_load8(dyr, (long)m * COp + cb, ok);
      } else {
        v = bf8_zero();
        if (ok) v = *reinterpret_cast<const bf16x8*>(&dy[(long)m * COp + cb]);
      }
      *reinterpret_cast<bf16x8*>(&plane(ni)[mloc * 16 + half * 8]) = v;
    }
    __syncthreads();

    bf16x8 bfr[MSTEPS];
#pragma unroll
    for (int s = 0; s < MSTEPS; ++s)
      bfr[s] = tr_read_8m(plane(wid) + s * 32 * 16, l);
    if (db_ws != nullptr) {
#pragma unroll
      for (int s = 0; s < MSTEPS; ++s)
#pragma unroll
        for (int j = 0; j < 8; ++j) db_part += to_f32(bfr[s][j]);
    }
    __syncthreads();

    // A-image build for tap row kh: entry (m, kw*4+ci); two b64 region
    // reads (adjacent columns) make one 16-byte image write
    auto build = [&](int kh, int buf) {
      for (int idx = tid; idx < MC * 4; idx += 256) {
        int g8 = idx & 3;          // (plane f = g8>>1, half = g8&1)
        int mloc = idx >> 2;
        int mm = lds_mm[mloc];
        int oy = (mm & 255) - 1;
        bf16x4 v0 = {}, v1 = {};
        int iy = 2 * oy + kh - 3;
        if (oy >= 0 && iy >= 0 && iy < 2 * OH) {
          // region row = 2*(g-g0)+kh, col = 2*ox + kw with kw = g8*2 (+1)
          int base = (mm >> 8) + kh * PXCOLS + (g8 << 1);
          v0 = *reinterpret_cast<const bf16x4*>(&lds_reg[base * 4]);
          v1 = *reinterpret_cast<const bf16x4*>(&lds_reg[(base + 1) * 4]);
        }
        bf16x8 v;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          v[j] = v0[j];
          v[4 + j] = v1[j];
        }
        *reinterpret_cast<bf16x8*>(
            &plane((g8 >> 1) * 2 + buf)[mloc * 16 + (g8 & 1) * 8]) = v;
      }
    };

    build(0, 0);
    __syncthreads();
#pragma unroll
    for (int kh = 0; kh < 7; ++kh) {
      int buf = kh & 1;
      if (kh + 1 < 7) build(kh + 1, buf ^ 1);
      // paired tr-reads before their MFMAs + alternating accumulators
      // (wait-collapse; same fix as conv2d_wgrad_rs_bf16_k)
#pragma unroll
      for (int s = 0; s < MSTEPS; ++s) {
        bf16x8 a0 = tr_read_8m(plane(0 * 2 + buf) + s * 32 * 16, l);
        bf16x8 a1 = tr_read_8m(plane(1 * 2 + buf) + s * 32 * 16, l);
        acc[0][kh] = mfma16x16x32bf(a0, bfr[s], acc[0][kh]);
        acc[1][kh] = mfma16x16x32bf(a1, bfr[s], acc[1][kh]);
      }
      __syncthreads();
    }
  }

  // slab: D row (lk*4+reg) = k-local within the kf frag, col (lm) = co
  float* slab = dw_ws + (long)ms * (CO * (long)Kcan);
  const int lk = l >> 4;
#pragma unroll
  for (int f = 0; f < 2; ++f) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int kidx = f * 16 + lk * 4 + reg;  // kw*4+ci
      int kw = kidx >> 2, ci = kidx & 3;
      if (kw >= 7 || ci >= CI) continue;
      int co = co0 + wid * 16 + lm;
      if (co >= CO) continue;
#pragma unroll
      for (int kh = 0; kh < 7; ++kh)
        slab[(long)co * Kcan + (long)ci * 49 + kh * 7 + kw] = acc[f][kh][reg];
    }
  }
  if (db_ws != nullptr) {
    db_part += __shfl_down(db_part, 16);
    db_part += __shfl_down(db_part, 32);
    int co = co0 + wid * 16 + lm;
    if (lk == 0 && co < CO) db_ws[(long)ms * CO + co] = db_part;
  }
}

// ---------------------------------------------------------------------------
// LDS-tiled stride-2 DGRAD (parity classes): for stride 2, dx pixel
// (2y+p, 2x+q) only receives taps of matching parity, and per class the
// whole dgrad collapses to a stride-1 2x2 conv over dy:
//     tap (a,b) in {0,1}^2  <->  kh = p + pad - 2a, kw = q + pad - 2b
// (invalid kh/kw map to zero weights in the repack — this covers both the
// 3x3 pad-1 convs, where classes carry 1/2/2/4 live taps, and the 1x1
// pad-0 downsample, where only class (0,0) tap (0,0) is live). dy is
// HALF-resolution, so the staged region for an 8-col tile is tiny
// (9 x (TW+1) x 72 bf16) and every dy element is read from LDS for all
// four taps. The gather parity engine this replaces was ~11x off its
// bandwidth bound (profiles/resnet18_pmc_final2_r1.txt: 3% MFMA, scattered
// per-pixel loads). Epilogue writes dx[2y+p][2x+q] directly.
// blockIdx.z = class; per-class output dims Hc x Wc derived in-kernel.
// ---------------------------------------------------------------------------
template <int NW, int MW, int NIMG, bool BGLDS = false>
__global__ __launch_bounds__(256) void conv_tiled_dgrad_s2_bf16_k(
    const bf16* __restrict__ dy, const bf16* __restrict__ wc,
    bf16* __restrict__ dx, int COp, int OH, int OW, int CIp, int CIcan,
    int H, int W, int N, int tiles_h, int tiles_w, int tap_mask) {
  constexpr int TH = 8, TW = 8 * (MW / NIMG);
  constexpr int RH = TH + 1, RW = TW + 1;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lm = l & 15;
  const int lk = l >> 4;

  int tw = blockIdx.x % tiles_w;
  int rest = blockIdx.x / tiles_w;
  int th = rest % tiles_h;
  int n = (rest / tiles_h) * NIMG;
  const int oh0 = th * TH, ow0 = tw * TW;
  const int ci0 = blockIdx.y * (NW * 16);
  const int p = blockIdx.z >> 1, q = blockIdx.z & 1;
  const int Hc = (H - p + 1) >> 1, Wc = (W - q + 1) >> 1;
  const bf16* wcls = wc + (long)blockIdx.z * 4 * CIp * COp;

  __shared__ bf16 lds_y[NIMG * RH * RW * 72 + 8 + (BGLDS ? 2 * 64 * 64 : 0)];
  bf16* ldsB = &lds_y[NIMG * RH * RW * 72 + 8];

  const int py = wid * 2 + (lm >> 3);
  f32x4 acc[MW][NW] = {};

  // a class with no live taps (1x1 downsample, classes 1-3) skips straight
  // to the zero-writing epilogue
  const int k_end = ((tap_mask >> (blockIdx.z * 4)) & 15) ? COp : 0;
  const long dy_elems = (long)N * OH * OW * COp;
  const bool use_buf = rsrc_fits(dy_elems);
  buf_rsrc_t dyr = make_bf16_rsrc(dy, dy_elems);
  for (int cic = 0; cic < k_end; cic += 64) {
    int cc = min(64, COp - cic);
    int ccp = cc + 8;
    __syncthreads();
    for (int idx = tid; idx < NIMG * RH * RW * (cc / 8); idx += 256) {
      int g8 = idx % (cc / 8);
      int pp = idx / (cc / 8);
      int rx = pp % RW;
      int rest2 = pp / RW;
      int ry = rest2 % RH;
      int img = rest2 / RH;
      int ih = oh0 + ry, iw = ow0 + rx;
      bool ok = ih < OH && iw < OW && n + img < N;
      long off = ((long)((n + img) * OH + ih) * OW + iw) * COp + cic + g8 * 8;
      bf16x8 v;
      if (use_buf) {
        v = buf_load8(dyr, off, ok);   // countable, no per-stage vmcnt(0)
      } else {
        v = bf8_zero();
        if (ok) v = *reinterpret_cast<const bf16x8*>(&dy[off]);
      }
      *reinterpret_cast<bf16x8*>(
          &lds_y[(img * RH * RW + ry * RW + rx) * ccp + g8 * 8]) = v;
    }
    __syncthreads();
    // glds-pipelined tap weight panels (same recipe as conv_tiled_s1's
    // BGLDS; dead taps still stage/barrier so the loop stays uniform)
    auto stage_B = [&](int tap, int buf) {
      if constexpr (BGLDS) {
        const bf16* wt = wcls + ((long)tap * CIp + ci0) * COp + cic;
        bf16* panel = ldsB + buf * (64 * 64);
#pragma unroll
        for (int i = 0; i < 2; ++i) {
          int sl = (wid * 2 + i) * 64 + l;
          int ci = sl >> 3;
          int chlog = (sl & 7) ^ ((ci >> 1) & 7);
          const bf16* src = &wt[(long)ci * COp + chlog * 8];
          __builtin_amdgcn_global_load_lds(
              (const __attribute__((address_space(1))) unsigned int*)src,
              (__attribute__((address_space(3))) unsigned int*)(panel +
                                                               (wid * 2 + i) *
                                                                   512),
              16, 0, 0);
        }
      }
    };
    if (BGLDS) {
      stage_B(0, 0);
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
#pragma unroll
    for (int tap = 0; tap < 4; ++tap) {
      const bf16* panel = BGLDS ? ldsB + (tap & 1) * (64 * 64) : nullptr;
      if (BGLDS && tap + 1 < 4) stage_B(tap + 1, (tap + 1) & 1);
      // live-tap mask (host bit cls*4+tap): classes carry 1/2/2/4 live
      // taps for 3x3 pad-1 and a single live tap for 1x1 — skipping the
      // zero-weight ones is ~44% of the MFMA work (block-uniform branch)
      bool live = ((tap_mask >> (blockIdx.z * 4 + tap)) & 1) != 0;
      if (!BGLDS && !live) continue;
      int a = tap >> 1, b = tap & 1;
      const bf16* wtap = wcls + ((long)tap * CIp + ci0) * COp + cic;
      if (!live) {  // BGLDS: keep the stage/barrier tail uniform
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        continue;
      }
#pragma unroll
      for (int ks = 0; ks < 64; ks += 32) {
        // hoisted A reads + clamped unconditional loads (wait-collapse,
        // same fix as conv_tiled_s1_bf16_k)
        bool gin = lk * 8 < cc - ks;
        int bsel = gin ? ks + lk * 8 : 0;
        bf16x8 bw[NW];
        if constexpr (BGLDS) {
          int chlog = (ks >> 3) + lk;
#pragma unroll
          for (int ni = 0; ni < NW; ++ni) {
            int ci = ni * 16 + lm;
            int chst = chlog ^ ((ci >> 1) & 7);
            bw[ni] = *reinterpret_cast<const bf16x8*>(
                &panel[ci * 64 + chst * 8]);
          }
        } else {
#pragma unroll
          for (int ni = 0; ni < NW; ++ni) {
            bf16x8 t = *reinterpret_cast<const bf16x8*>(
                &wtap[(long)(ni * 16 + lm) * COp + bsel]);
            bw[ni] = gin ? t : bf8_zero();
          }
        }
        bf16x8 av[MW];
#pragma unroll
        for (int mi = 0; mi < MW; ++mi) {
          int px = (mi % (MW / NIMG)) * 8 + (lm & 7);
          int img = mi / (MW / NIMG);
          const bf16* ypix =
              &lds_y[(img * RH * RW + (py + a) * RW + (px + b)) * ccp];
          bf16x8 t = *reinterpret_cast<const bf16x8*>(&ypix[bsel]);
          av[mi] = gin ? t : bf8_zero();
        }
#pragma unroll
        for (int mi = 0; mi < MW; ++mi)
#pragma unroll
          for (int ni = 0; ni < NW; ++ni)
            acc[mi][ni] = mfma16x16x32bf(av[mi], bw[ni], acc[mi][ni]);
      }
      if (BGLDS) {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
      }
    }
  }

#pragma unroll
  for (int mi = 0; mi < MW; ++mi) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int mo = lk * 4 + reg;
      int oy = oh0 + wid * 2 + (mo >> 3);
      int ox = ow0 + (mi % (MW / NIMG)) * 8 + (mo & 7);
      int nimg = n + mi / (MW / NIMG);
      if (oy >= Hc || ox >= Wc || nimg >= N) continue;
      long xb = ((long)(nimg * H + 2 * oy + p) * W + 2 * ox + q) * CIp;
#pragma unroll
      for (int ni = 0; ni < NW; ++ni) {
        int ci = ci0 + ni * 16 + lm;
        if (ci >= CIp) continue;
        dx[xb + ci] = to_bf16(ci < CIcan ? acc[mi][ni][reg] : 0.f);
      }
    }
  }
}

// w[CO][CI][KH][KW] fp32 -> wc[class][a*2+b][ci][co] bf16 with
// kh = p + pad - 2a, kw = q + pad - 2b (invalid taps and channel pads
// written as zeros; destination can be at::empty)
__global__ void repack_w_dgrad_s2_k(const float* __restrict__ w,
                                    bf16* __restrict__ wc, int CO, int CI,
                                    int KH, int KW, int CIp, int COp,
                                    int pad) {
  long percls = (long)4 * CIp * COp;
  long total = 4 * percls;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int co = i % COp;
    long r = i / COp;
    int ci = r % CIp;
    long r2 = r / CIp;
    int tap = r2 & 3;
    int cls = r2 >> 2;
    int pcl = cls >> 1, qcl = cls & 1;
    int kh = pcl + pad - 2 * (tap >> 1);
    int kw = qcl + pad - 2 * (tap & 1);
    bf16 v = to_bf16(0.f);
    if (kh >= 0 && kh < KH && kw >= 0 && kw < KW && co < CO && ci < CI)
      v = to_bf16(w[(((long)co * CI + ci) * KH + kh) * KW + kw]);
    wc[i] = v;
  }
}

// ---------------------------------------------------------------------------
// LDS-tiled stride-2 STEM conv (ResNet conv1: 7x7 s2, CIp==8). The gather
// engine re-reads each input pixel ~KH*KW/4 times from L2 here (486 us on
// the b256 stem, the largest single forward kernel); instead a block stages
// the whole ((TH-1)*2+KH) x ((TW-1)*2+KH) x 8ch input region for an 8x32
// output tile into LDS once (23 KB) and k-orders the GEMM as (kh | kw*8+ci):
// every 8-k group is ONE tap's 8 channels = one aligned vec8 LDS read, with
// the kw axis row-padded to 8 taps (KROW=64) so a 32-k MFMA step spans
// exactly 4 predicated taps. Weights repack to [kh][co][kw*8+ci]
// (repack_w_stem_k) making B-fragments contiguous 16-byte loads. MW=4 pixel
// fragments x NW=4 co-tiles -> 16 MFMAs per 32-k step.
// ---------------------------------------------------------------------------
template <int KH, bool HAS_BIAS, bool RELU>
__global__ __launch_bounds__(256) void conv_stem_s2_bf16_k(
    const bf16* __restrict__ x, const bf16* __restrict__ w_rp,
    const float* __restrict__ bias, bf16* __restrict__ y, int H, int W,
    int COp, int COcan, int OH, int OW, int pad, int N, int tiles_h,
    int tiles_w) {
  constexpr int TH = 8, TW = 32, MW = 4, NW = 4;
  // channels packed to 4 (CI <= 4): k per row = KH*4 <= 32 = ONE MFMA step
  // per tap row. +1 region pad column: the kw==7 lane half reads it freely
  // (its B column is repack-zeroed, so the product is 0 either way).
  constexpr int RH = (TH - 1) * 2 + KH, RW = (TW - 1) * 2 + KH + 1;
  constexpr int KROW = 32;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lm = l & 15;
  const int lk = l >> 4;

  int tw = blockIdx.x % tiles_w;
  int rest = blockIdx.x / tiles_w;
  int th = rest % tiles_h;
  int n = rest / tiles_h;
  const int oh0 = th * TH, ow0 = tw * TW;
  const int co0 = blockIdx.y * (NW * 16);

  __shared__ bf16 lds_x[RH * RW * 4 + 8];

  // stage the input region ONCE, dropping input channels 4..7 (pure pad):
  // one 8-byte store per region pixel (buffer loads: countable waits)
  const int ih0 = oh0 * 2 - pad, iw0 = ow0 * 2 - pad;
  const long x_elems = (long)N * H * W * 8;
  const bool use_buf = rsrc_fits(x_elems);
  buf_rsrc_t xr = make_bf16_rsrc(x, x_elems);
  for (int idx = tid; idx < RH * RW; idx += 256) {
    int rx = idx % RW, ry = idx / RW;
    int ih = ih0 + ry, iw = iw0 + rx;
    bool ok = ih >= 0 && ih < H && iw >= 0 && iw < W;
    long off = ((long)(n * H + ih) * W + iw) * 8;
    bf16x4 v;
    if (use_buf) {
      v = buf_load4(xr, off, ok);
    } else {
      v = bf16x4{};
      if (ok) v = *reinterpret_cast<const bf16x4*>(&x[off]);
    }
    *reinterpret_cast<bf16x4*>(&lds_x[idx * 4]) = v;
  }
  __syncthreads();

  const int py = wid * 2 + (lm >> 3);  // fragment row within the tile
  f32x4 acc[MW][NW] = {};
  // software-pipelined B loads: next row's weights issue before this row's
  // MFMAs so the 16-MFMA step never stalls on L1/L2 latency
  const bf16* wbase = w_rp + (long)(co0 + lm) * KROW + lk * 8;
  bf16x8 bw[NW];
#pragma unroll
  for (int ni = 0; ni < NW; ++ni)
    bw[ni] = *reinterpret_cast<const bf16x8*>(&wbase[(long)ni * 16 * KROW]);
#pragma unroll
  for (int kh = 0; kh < KH; ++kh) {
    bf16x8 bc[NW];
#pragma unroll
    for (int ni = 0; ni < NW; ++ni) bc[ni] = bw[ni];
    if (kh + 1 < KH) {
      const bf16* wn = wbase + (long)(kh + 1) * COp * KROW;
#pragma unroll
      for (int ni = 0; ni < NW; ++ni)
        bw[ni] = *reinterpret_cast<const bf16x8*>(&wn[(long)ni * 16 * KROW]);
    }
#pragma unroll
    for (int mi = 0; mi < MW; ++mi) {
      // k_local = lk*8+j -> tap kw = lk*2 + j/4, ci = j&3: the vec8 spans
      // two adjacent region pixels' 4-channel packs — contiguous, aligned
      int px = mi * 8 + (lm & 7);
      bf16x8 a = *reinterpret_cast<const bf16x8*>(
          &lds_x[((py * 2 + kh) * RW + px * 2 + lk * 2) * 4]);
#pragma unroll
      for (int ni = 0; ni < NW; ++ni)
        acc[mi][ni] = mfma16x16x32bf(a, bc[ni], acc[mi][ni]);
    }
  }

#pragma unroll
  for (int mi = 0; mi < MW; ++mi) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int mo = lk * 4 + reg;
      int oy = oh0 + wid * 2 + (mo >> 3);
      int ox = ow0 + mi * 8 + (mo & 7);
      if (oy >= OH || ox >= OW) continue;
      long yb = ((long)(n * OH + oy) * OW + ox) * COp;
#pragma unroll
      for (int ni = 0; ni < NW; ++ni) {
        int co = co0 + ni * 16 + lm;
        if (co >= COp) continue;
        float v = (co < COcan) ? acc[mi][ni][reg] : 0.f;
        if (HAS_BIAS && co < COcan) v += bias[co];
        if (RELU) v = fmaxf(v, 0.f);
        y[yb + co] = to_bf16(v);
      }
    }
  }
}

// w[CO][CI][KH][KW] fp32 -> w_rp[kh][co][kw*4+ci] bf16 (CI <= 4), the kw
// axis row-padded to KROW=32 k; all pads written as zeros (destination can
// be at::empty)
__global__ void repack_w_stem_k(const float* __restrict__ w,
                                bf16* __restrict__ w_rp, int CO, int CI,
                                int KH, int KW, int COp) {
  long total = (long)KH * COp * 32;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int kwci = i & 31;
    long r = i >> 5;
    int co = r % COp;
    int kh = r / COp;
    int kw = kwci >> 2, ci = kwci & 3;
    bf16 v = to_bf16(0.f);
    if (co < CO && ci < CI && kw < KW)
      v = to_bf16(w[(((long)co * CI + ci) * KH + kh) * KW + kw]);
    w_rp[i] = v;
  }
}

// ---------------------------------------------------------------------------
// Batched weight repack: ONE launch refreshes every cached packed-weight
// buffer after the optimizer step (grid.y = cache entry). Replaces ~2 small
// repack launches per conv per step (mostly launch tail at these extents)
// with a single kernel. Kinds mirror repack_w_k / repack_w_stem_k /
// repack_w_dgrad_s2_k above — same indexing, same zero-padding rules.
// ---------------------------------------------------------------------------
struct PackDesc {
  const float* w;
  bf16* rp;
  long extent;
  long base;  // prefix offset in the flattened batched-refresh index space
  int kind;  // 0 = repack_w (fwd/dgrad s1), 1 = stem, 2 = dgrad_s2 parity
  int CO, CI, KH, KW, Ap, Bp, fp;  // fp: kind0 dgrad flag / kind2 pad
};

__global__ void repack_all_k(const PackDesc* __restrict__ ds, int n,
                             long total) {
  // flattened 1:1 thread<->element over the SUM of entry extents (a
  // blocks-per-entry grid.y layout measured 200 us: the per-entry block cap
  // starved the big entries while small entries idled). Entry found by
  // binary search on the prefix offsets — 6 scalar-ish steps, block-uniform
  // except at entry seams.
  // bases are 256-aligned (host pads the flattened space), so a BLOCK maps
  // to exactly one entry: search on the block-uniform offset -> the whole
  // lookup compiles to scalar loads, one chain per wave not per lane
  long i0 = (long)blockIdx.x * blockDim.x;
  if (i0 >= total) return;
  int lo = 0, hi = n - 1;
  while (lo < hi) {
    int mid = (lo + hi + 1) >> 1;
    if (ds[mid].base <= i0) lo = mid;
    else hi = mid - 1;
  }
  const PackDesc d = ds[lo];
  long j = i0 + threadIdx.x - d.base;
  if (j >= d.extent) return;  // alignment gap past the entry tail
  bf16 v = to_bf16(0.f);
  if (d.kind == 0) {
    int KHKW = d.KH * d.KW;
    int b = j % d.Bp;
    long r = j / d.Bp;
    int a = r % d.Ap;
    int tap_d = r / d.Ap;
    int co = d.fp ? b : a, ci = d.fp ? a : b;
    int tap = d.fp ? KHKW - 1 - tap_d : tap_d;
    if (co < d.CO && ci < d.CI)
      v = to_bf16(d.w[((long)co * d.CI + ci) * KHKW + tap]);
  } else if (d.kind == 1) {
    int kwci = j & 31;
    long r = j >> 5;
    int co = r % d.Ap;
    int kh = r / d.Ap;
    int kw = kwci >> 2, ci = kwci & 3;
    if (co < d.CO && ci < d.CI && kw < d.KW)
      v = to_bf16(d.w[(((long)co * d.CI + ci) * d.KH + kh) * d.KW + kw]);
  } else {
    int co = j % d.Bp;
    long r = j / d.Bp;
    int ci = r % d.Ap;
    long r2 = r / d.Ap;
    int tap = r2 & 3;
    int cls = r2 >> 2;
    int kh = (cls >> 1) + d.fp - 2 * (tap >> 1);
    int kw = (cls & 1) + d.fp - 2 * (tap & 1);
    if (kh >= 0 && kh < d.KH && kw >= 0 && kw < d.KW && co < d.CO &&
        ci < d.CI)
      v = to_bf16(d.w[(((long)co * d.CI + ci) * d.KH + kh) * d.KW + kw]);
  }
  d.rp[j] = v;
}

// ---------------------------------------------------------------------------
// Wide-class LogSoftmax+NLL (ResNet: C=1000): one WAVE per row, lanes split
// the class axis, shfl tree reductions. (The thread-per-row kernel above is
// for tiny C like LeNet's 10.)
// ---------------------------------------------------------------------------
__global__ void logsoftmax_nll_bf16_wide_fwd_k(
    const bf16* __restrict__ x, const long* __restrict__ target,
    float* __restrict__ logp, float* __restrict__ loss, int M, int COp, int C,
    int with_loss) {
  __shared__ float red[4];
  int wid = threadIdx.x >> 6;
  int l = threadIdx.x & 63;
  float part = 0.f;
  for (long m = (long)blockIdx.x * 4 + wid; m < M; m += (long)gridDim.x * 4) {
    const bf16* row = x + m * COp;
    float mx = -3.4e38f;
    for (int c = l; c < C; c += 64) mx = fmaxf(mx, to_f32(row[c]));
    for (int off = 32; off > 0; off >>= 1) mx = fmaxf(mx, __shfl_xor(mx, off));
    float s = 0.f;
    for (int c = l; c < C; c += 64) s += __expf(to_f32(row[c]) - mx);
    for (int off = 32; off > 0; off >>= 1) s += __shfl_xor(s, off);
    float lse = mx + __logf(s);
    float* out = logp + m * C;
    for (int c = l; c < C; c += 64) out[c] = to_f32(row[c]) - lse;
    if (with_loss && l == 0) part -= to_f32(row[target[m]]) - lse;
  }
  if (with_loss) {
    if (l == 0) red[wid] = part;
    __syncthreads();
    if (threadIdx.x == 0)
      atomicAdd(loss, (red[0] + red[1] + red[2] + red[3]) / (float)M);
  }
}

__global__ void logsoftmax_nll_bf16_wide_bwd_k(
    const float* __restrict__ logp, const long* __restrict__ target,
    const float* __restrict__ dloss, bf16* __restrict__ dx, int M, int COp,
    int C) {
  float scale = dloss[0] / (float)M;
  long total = (long)M * COp;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int c = i % COp;
    long m = i / COp;
    float v = 0.f;
    if (c < C) v = (__expf(logp[m * C + c]) - (c == target[m] ? 1.f : 0.f)) * scale;
    dx[i] = to_bf16(v);
  }
}

// ---------------------------------------------------------------------------
// Stride-2 dgrad, parity-specialized (ResNet downsample convs). For s=2 a
// dx pixel (ih, iw) only receives taps with kh === (ih+pad) (mod 2) and
// kw === (iw+pad) (mod 2). Grouping dx pixels by parity class (blockIdx.z)
// makes the valid-tap set WAVE-UNIFORM: the kernel loops just that subset
// (1/2/2/4 taps of 9 for 3x3) with no divisibility predicates, unlike the
// generic dgrad which scans all KH*KW*COp with per-lane checks.
// ---------------------------------------------------------------------------
template <int MW, int NW>
__global__ __launch_bounds__(256) void conv2d_dgrad_s2_bf16_k(
    const bf16* __restrict__ dy, const float* __restrict__ w,
    bf16* __restrict__ dx,
    int CIp, int CI, int H, int W, int COp, int CO, int KH, int KW,
    int OH, int OW, int pad, int N) {
  constexpr int BK = 32;
  const int BM = 4 * MW * 16;
  const int cy = blockIdx.z >> 1, cx = blockIdx.z & 1;
  const int Hc = (H - cy + 1) / 2, Wc = (W - cx + 1) / 2;
  const long Mc = (long)N * Hc * Wc;   // pixels in this parity class
  const int m0 = blockIdx.x * BM;
  const int ci0 = blockIdx.y * (NW * 16);
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lm = l & 15;
  const int lk = l >> 4;

  // valid tap grid for this class
  const int kh0 = (cy + pad) & 1;
  const int kw0 = (cx + pad) & 1;
  const int nkh = (KH - kh0 + 1) / 2;
  const int nkw = (KW - kw0 + 1) / 2;
  const int Kp = nkh * nkw * COp;      // reduced k' space

  __shared__ bf16 lds_w[NW * 16][BK + 8];
  __shared__ int lds_kdec[BK / 8];     // (kh<<20)|(kw<<10)|co0g

  int ih[MW], iw[MW], nn[MW];
  bool mval[MW];
#pragma unroll
  for (int mi = 0; mi < MW; ++mi) {
    long m = m0 + (wid * MW + mi) * 16 + lm;
    mval[mi] = m < Mc;
    long mm = mval[mi] ? m : 0;
    long per_img = (long)Hc * Wc;
    nn[mi] = (int)(mm / per_img);
    int r = (int)(mm % per_img);
    ih[mi] = cy + 2 * (r / Wc);
    iw[mi] = cx + 2 * (r % Wc);
  }

  f32x4 acc[MW][NW] = {};

  for (int k0 = 0; k0 < Kp; k0 += BK) {
    __syncthreads();
    for (int idx = tid; idx < NW * 16 * BK; idx += 256) {
      int ci = idx >> 5;
      int kk = idx & 31;
      int k = k0 + kk;
      float v = 0.f;
      if (k < Kp && ci0 + ci < CI) {
        int t = k / COp, co = k % COp;
        int kh = kh0 + 2 * (t / nkw), kw = kw0 + 2 * (t % nkw);
        if (co < CO)
          v = w[((co * CI) + ci0 + ci) * (KH * KW) + kh * KW + kw];
      }
      lds_w[ci][kk] = to_bf16(v);
    }
    if (tid < BK / 8) {
      int k = k0 + tid * 8;
      int kcl = k < Kp ? k : 0;
      int t = kcl / COp, co0g = kcl % COp;
      int kh = kh0 + 2 * (t / nkw), kw = kw0 + 2 * (t % nkw);
      lds_kdec[tid] = (kh << 20) | (kw << 10) | co0g;
    }
    __syncthreads();

    int kgroups = min(BK, Kp - k0) / 8;
    for (int kg = 0; kg < kgroups; kg += 4) {
      bool gin = lk < (kgroups - kg);
      int gsel = gin ? kg + lk : 0;
      int kdec = lds_kdec[gsel];
      int kh = kdec >> 20, kw = (kdec >> 10) & 0x3ff, co0g = kdec & 0x3ff;
      bf16x8 bw[NW];
#pragma unroll
      for (int ni = 0; ni < NW; ++ni)
        bw[ni] = gin ? *reinterpret_cast<const bf16x8*>(&lds_w[ni * 16 + lm][gsel * 8])
                     : bf8_zero();
#pragma unroll
      for (int mi = 0; mi < MW; ++mi) {
        bf16x8 a = bf8_zero();
        if (mval[mi] & gin) {
          int oh = (ih[mi] + pad - kh) >> 1;
          int ow = (iw[mi] + pad - kw) >> 1;
          if (oh >= 0 && oh < OH && ow >= 0 && ow < OW)
            a = *reinterpret_cast<const bf16x8*>(
                &dy[((long)(nn[mi] * OH + oh) * OW + ow) * COp + co0g]);
        }
#pragma unroll
        for (int ni = 0; ni < NW; ++ni)
          acc[mi][ni] = mfma16x16x32bf(a, bw[ni], acc[mi][ni]);
      }
    }
  }

#pragma unroll
  for (int mi = 0; mi < MW; ++mi) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      long m = m0 + (wid * MW + mi) * 16 + lk * 4 + reg;
      if (m >= Mc) continue;
      long per_img = (long)Hc * Wc;
      int n = (int)(m / per_img);
      int r = (int)(m % per_img);
      int oy = cy + 2 * (r / Wc), ox = cx + 2 * (r % Wc);
      long xb = ((long)(n * H + oy) * W + ox) * CIp;
#pragma unroll
      for (int ni = 0; ni < NW; ++ni) {
        int ci = ci0 + ni * 16 + lm;
        if (ci >= CIp) continue;
        dx[xb + ci] = to_bf16(ci < CI ? acc[mi][ni][reg] : 0.f);
      }
    }
  }
}
