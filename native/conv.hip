// Conv2d fwd / bwd-input / bwd-weight as fused-im2col implicit-GEMM on MFMA
// (BASELINE.json:5 "Conv2d fwd/bwd via im2col-GEMM ...
// hand-written HIP for CDNA4 with MFMA/LDS tiling").
//
// fp32 path, NCHW, exact-f32 MFMA (v_mfma_f32_16x16x4_f32 — guide §3: the f32
// vector-rate matrix path, ~2.4x a VALU GEMM at identical numerics).
//
// GEMM views (M = N*OH*OW output pixels, K = CI*KH*KW, canonical k order
// (ci,kh,kw) == NCHW patch order, so weights need NO repack):
//   fwd   : D[co][m]  = sum_k  W[co][k]      * im2col(x)[m][k]
//   dgrad : D[ci][m'] = sum_k' Wflip[ci][k'] * col(dy)[m'][k']   (k'=(co,kh,kw))
//   wgrad : D[k][co]  = sum_m  im2col(x)[m][k] * dy[m][co]       (split-M + atomicAdd)
//
// Operand/store orientation is chosen so every global 16-lane access group
// walks the fastest (ow/iw) NCHW axis => coalesced gathers and stores.
#include "common.h"

// ---------------------------------------------------------------------------
// Forward. Block = 256 threads = 4 waves; each wave owns MW=2 m-frags of 16
// rows => BM = 128 output pixels per block; one 16-wide co tile per block.
// Weights + k-decode staged in LDS per BK=32 chunk.
// ---------------------------------------------------------------------------
template <bool HAS_BIAS, bool RELU, bool BOUNDS, int NW>
__global__ __launch_bounds__(256) void conv2d_fwd_f32_k(
    const float* __restrict__ x, const float* __restrict__ w,
    const float* __restrict__ bias, float* __restrict__ y,
    int CI, int H, int W, int CO, int KH, int KW, int OH, int OW,
    int stride, int pad, int M, int K) {
  constexpr int BM = 128, BK = 32, MW = 2;
  const int m0 = blockIdx.x * BM;
  const int co0 = blockIdx.y * (NW * 16);
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lm = l & 15;   // B col (m) / A row (co)
  const int lk = l >> 4;   // k slice 0..3

  __shared__ float lds_w[BK][NW * 16];
  __shared__ int lds_koff[BK];    // ci*H*W + kh*W + kw
  __shared__ int lds_khw[BK];     // (kh<<16)|kw, only used if BOUNDS

  const int HW = H * W;
  const int OHOW = OH * OW;

  int xbase[MW];
  int ybase[MW];
  int ih0[MW], iw0[MW];
  bool mval[MW];
#pragma unroll
  for (int mi = 0; mi < MW; ++mi) {
    int m = m0 + (wid * MW + mi) * 16 + lm;
    mval[mi] = m < M;
    int mm = mval[mi] ? m : 0;
    int n = mm / OHOW, r = mm % OHOW;
    int oh = r / OW, ow = r % OW;
    ih0[mi] = oh * stride - pad;
    iw0[mi] = ow * stride - pad;
    xbase[mi] = n * CI * HW + ih0[mi] * W + iw0[mi];
    ybase[mi] = n * CO * OHOW + r;
  }

  f32x4 acc[MW][NW] = {};

  for (int k0 = 0; k0 < K; k0 += BK) {
    __syncthreads();
    for (int idx = tid; idx < BK * NW * 16; idx += 256) {
      int kk = idx / (NW * 16), co = idx % (NW * 16);
      int k = k0 + kk;
      lds_w[kk][co] = (k < K && co0 + co < CO) ? w[(co0 + co) * K + k] : 0.f;
    }
    if (tid < BK) {
      int k = k0 + tid;
      int kcl = k < K ? k : 0;
      int ci = kcl / (KH * KW), r = kcl % (KH * KW);
      int kh = r / KW, kw = r % KW;
      lds_koff[tid] = ci * HW + kh * W + kw;
      lds_khw[tid] = (kh << 16) | kw;
    }
    __syncthreads();

    int klim = min(BK, K - k0);
    for (int kk = 0; kk + 4 <= klim; kk += 4) {
      int koff = lds_koff[kk + lk];
      float a[NW];
#pragma unroll
      for (int ni = 0; ni < NW; ++ni) a[ni] = lds_w[kk + lk][ni * 16 + lm];
#pragma unroll
      for (int mi = 0; mi < MW; ++mi) {
        bool ok = mval[mi];
        if (BOUNDS) {
          int khw = lds_khw[kk + lk];
          int kh = khw >> 16, kw = khw & 0xffff;
          int ih = ih0[mi] + kh, iw = iw0[mi] + kw;
          ok = ok && ih >= 0 && ih < H && iw >= 0 && iw < W;
        }
        float b = ok ? x[xbase[mi] + koff] : 0.f;
#pragma unroll
        for (int ni = 0; ni < NW; ++ni)
          acc[mi][ni] = mfma16x16x4(a[ni], b, acc[mi][ni]);
      }
    }
    int ktail = klim & 3;
    if (ktail) {
      int kk = klim - ktail;
      bool kin = lk < ktail;
      int koff = kin ? lds_koff[kk + lk] : 0;
      float a[NW];
#pragma unroll
      for (int ni = 0; ni < NW; ++ni)
        a[ni] = kin ? lds_w[kk + lk][ni * 16 + lm] : 0.f;
#pragma unroll
      for (int mi = 0; mi < MW; ++mi) {
        bool ok = mval[mi] && kin;
        if (BOUNDS && ok) {
          int khw = lds_khw[kk + lk];
          int kh = khw >> 16, kw = khw & 0xffff;
          int ih = ih0[mi] + kh, iw = iw0[mi] + kw;
          ok = ih >= 0 && ih < H && iw >= 0 && iw < W;
        }
        float b = ok ? x[xbase[mi] + koff] : 0.f;
#pragma unroll
        for (int ni = 0; ni < NW; ++ni)
          acc[mi][ni] = mfma16x16x4(a[ni], b, acc[mi][ni]);
      }
    }
  }

#pragma unroll
  for (int mi = 0; mi < MW; ++mi) {
    if (!mval[mi]) continue;
#pragma unroll
    for (int ni = 0; ni < NW; ++ni) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        int co = co0 + ni * 16 + lk * 4 + reg;
        if (co >= CO) continue;
        float v = acc[mi][ni][reg];
        if (HAS_BIAS) v += bias[co];
        if (RELU) v = fmaxf(v, 0.f);
        y[ybase[mi] + co * OHOW] = v;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Backward-input (dgrad). Same skeleton; m' = (n, ih, iw), k' = (co, kh, kw),
// A = flipped weights staged as [k'][ci-tile], B = dy gathered with the
// transposed-conv index map (zero where oh/ow fall outside or off-stride).
// ---------------------------------------------------------------------------
template <bool S1, int MW>
__global__ __launch_bounds__(256) void conv2d_dgrad_f32_k(
    const float* __restrict__ dy, const float* __restrict__ w,
    float* __restrict__ dx,
    int CI, int H, int W, int CO, int KH, int KW, int OH, int OW,
    int stride, int pad, int Mp, int Kp) {
  constexpr int BM = 4 * MW * 16, BK = 32;
  const int m0 = blockIdx.x * BM;
  const int ci0 = blockIdx.y * 16;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lm = l & 15;
  const int lk = l >> 4;

  __shared__ float lds_w[BK][16];   // Wflip^T [k'][ci]
  __shared__ int lds_kdec[BK];      // (co<<16)|(kh<<8)|kw

  const int HW = H * W;
  const int OHOW = OH * OW;
  const int KHKW = KH * KW;

  int dybase[MW];  // n*CO*OHOW
  int ih[MW], iw[MW];
  int dxbase[MW];
  bool mval[MW];
#pragma unroll
  for (int mi = 0; mi < MW; ++mi) {
    int m = m0 + (wid * MW + mi) * 16 + lm;
    mval[mi] = m < Mp;
    int mm = mval[mi] ? m : 0;
    int n = mm / HW, r = mm % HW;
    ih[mi] = r / W;
    iw[mi] = r % W;
    dybase[mi] = n * CO * OHOW;
    dxbase[mi] = n * CI * HW + r;
  }

  f32x4 acc[MW] = {};

  for (int k0 = 0; k0 < Kp; k0 += BK) {
    __syncthreads();
    for (int idx = tid; idx < BK * 16; idx += 256) {
      int kk = idx >> 4, ci = idx & 15;
      int k = k0 + kk;
      float v = 0.f;
      if (k < Kp && ci0 + ci < CI) {
        int co = k / KHKW, r = k % KHKW;
        int kh = r / KW, kw = r % KW;
        // flipped kernel: w[co][ci][KH-1-kh][KW-1-kw]
        v = w[((co * CI + ci0 + ci) * KH + (KH - 1 - kh)) * KW + (KW - 1 - kw)];
      }
      lds_w[kk][ci] = v;
    }
    if (tid < BK) {
      int k = k0 + tid;
      int kcl = k < Kp ? k : 0;
      int co = kcl / KHKW, r = kcl % KHKW;
      int kh = r / KW, kw = r % KW;
      lds_kdec[tid] = (co << 16) | (kh << 8) | kw;
    }
    __syncthreads();

    int klim = min(BK, Kp - k0);
    for (int kk = 0; kk < klim; kk += 4) {
      bool kin = lk < (klim - kk);
      int kdec = lds_kdec[kin ? kk + lk : 0];
      float a = kin ? lds_w[kk + lk][lm] : 0.f;
      int co = kdec >> 16, kh = (kdec >> 8) & 0xff, kw = kdec & 0xff;
      // flipped map: dx[ih] gets dy[oh] where ih = oh*s - p + (KH-1-kh)' ...
      // with the flip above: oh = (ih + pad - (KH-1-kh)) / s; using kdec's
      // (kh,kw) as the FLIPPED position directly:
#pragma unroll
      for (int mi = 0; mi < MW; ++mi) {
        float b = 0.f;
        if (mval[mi] & kin) {
          int tH = ih[mi] + pad - (KH - 1 - kh);
          int tW = iw[mi] + pad - (KW - 1 - kw);
          if (S1) {
            if (tH >= 0 && tH < OH && tW >= 0 && tW < OW)
              b = dy[dybase[mi] + co * OHOW + tH * OW + tW];
          } else {
            if (tH >= 0 && tW >= 0 && tH % stride == 0 && tW % stride == 0) {
              int oh = tH / stride, ow = tW / stride;
              if (oh < OH && ow < OW)
                b = dy[dybase[mi] + co * OHOW + oh * OW + ow];
            }
          }
        }
        acc[mi] = mfma16x16x4(a, b, acc[mi]);
      }
    }
  }

#pragma unroll
  for (int mi = 0; mi < MW; ++mi) {
    if (!mval[mi]) continue;
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int ci = ci0 + lk * 4 + reg;
      if (ci >= CI) continue;
      dx[dxbase[mi] + ci * HW] = acc[mi][reg];
    }
  }
}

// ---------------------------------------------------------------------------
// Backward-weight (wgrad) + bias. Each block owns a [k-tile x 16co] output
// tile and reduces an M-split; the 4 waves stream DISJOINT m-chunks in
// parallel (full wave utilization at any K, unlike a waves-on-k split that
// idles 3/4 of the block for K < 64), each accumulating the full k-tile
// (KT 16-row frags). Wave partials are reduced through LDS so each block
// issues ONE atomicAdd per dw element (atomic traffic matters: dw is tiny and
// thousands of M-split blocks hit the same cache lines).
// dy chunk is LDS-transposed (coalesced loads, conflict-free MFMA reads);
// per-m pixel decode is precomputed into LDS once per chunk.
// ---------------------------------------------------------------------------
template <bool BOUNDS, bool WANT_BIAS, int KT>
__global__ __launch_bounds__(256) void conv2d_wgrad_f32_k(
    const float* __restrict__ dy, const float* __restrict__ x,
    float* __restrict__ dw, float* __restrict__ db,
    int CI, int H, int W, int CO, int KH, int KW, int OH, int OW,
    int stride, int pad, int M, int K, int m_per_split) {
  constexpr int MC = 256;  // super-chunk: 4 waves x 64 m
  const int k0 = blockIdx.x * (KT * 16);
  const int co0 = blockIdx.y * 16;
  const int ms = blockIdx.z;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lm = l & 15;
  const int lk = l >> 4;

  __shared__ float lds_dy[MC][17];  // [m][co] transposed, padded
  __shared__ int lds_mbase[MC];     // n*CI*HW + ih0*W + iw0
  __shared__ int lds_mhw[MC];       // biased (ih0, iw0) for bounds

  const int HW = H * W;
  const int OHOW = OH * OW;

  // per-lane k decode (A row = lm), fixed over the whole reduction
  int koff[KT], khh[KT], kww[KT];
  bool kval[KT];
#pragma unroll
  for (int kf = 0; kf < KT; ++kf) {
    int k = k0 + kf * 16 + lm;
    kval[kf] = k < K;
    int kcl = kval[kf] ? k : 0;
    int ci = kcl / (KH * KW), kr = kcl % (KH * KW);
    khh[kf] = kr / KW;
    kww[kf] = kr % KW;
    koff[kf] = ci * HW + khh[kf] * W + kww[kf];
  }

  f32x4 acc[KT] = {};
  float db_part = 0.f;

  int m_begin = ms * m_per_split;
  int m_end = min(M, m_begin + m_per_split);

  // countable (buffer) loads for the x gathers + dy staging: the previous
  // exec-predicated forms forced vmcnt(0) drains per consumer (common.h)
  const long x_elems = (long)(M / OHOW) * CI * HW;
  const long dy_elems = (long)(M / OHOW) * CO * OHOW;
  const bool use_buf = rsrc_fits_f32(x_elems) && rsrc_fits_f32(dy_elems);
  buf_rsrc_t xr = make_f32_rsrc(x, x_elems);
  buf_rsrc_t dyrs = make_f32_rsrc(dy, dy_elems);

  for (int mc = m_begin; mc < m_end; mc += MC) {
    __syncthreads();
    // stage dy[mc:mc+256][co0:+16] transposed (reads coalesced over m)
    for (int idx = tid; idx < MC * 16; idx += 256) {
      int co = idx >> 8;  // idx = co*256 + mloc
      int mloc = idx & 255;
      int m = mc + mloc;
      bool ok = m < m_end && co0 + co < CO;
      int mm = ok ? m : 0;
      int n = mm / OHOW, r = mm % OHOW;
      long off = (long)(n * CO + co0 + co) * OHOW + r;
      float v;
      if (use_buf) {
        v = buf_loadf(dyrs, off, ok);
      } else {
        v = 0.f;
        if (ok) v = dy[off];
      }
      lds_dy[mloc][co] = v;
    }
    // stage per-m x-base decode
    if (tid < MC) {
      int m = mc + tid;
      int mm = m < m_end ? m : 0;
      int n = mm / OHOW, r = mm % OHOW;
      int oh = r / OW, ow = r % OW;
      int ih0 = oh * stride - pad, iw0 = ow * stride - pad;
      lds_mbase[tid] = n * CI * HW + ih0 * W + iw0;
      lds_mhw[tid] = ((ih0 + 16384) << 16) | (iw0 + 16384);
    }
    __syncthreads();

    // wave `wid` consumes m rows [wid*64, wid*64+64)
    int wbase = wid * 64;
    int mlim = min(64, m_end - mc - wbase);  // may be negative on tail
    for (int mm = 0; mm < mlim; mm += 4) {
      bool min_ok = (mm + lk) < mlim;
      int msel = wbase + (min_ok ? mm + lk : 0);
      float b = min_ok ? lds_dy[msel][lm] : 0.f;  // B[k=m][col=co]
      if (WANT_BIAS) db_part += b;
      int mb = lds_mbase[msel];
      int hw = lds_mhw[msel];
#pragma unroll
      for (int kf = 0; kf < KT; ++kf) {
        bool ok = kval[kf] & min_ok;
        if (BOUNDS && ok) {
          int ih = (hw >> 16) - 16384 + khh[kf];
          int iw = (hw & 0xffff) - 16384 + kww[kf];
          ok = ih >= 0 && ih < H && iw >= 0 && iw < W;
        }
        float a;
        if (use_buf) {
          a = buf_loadf(xr, (long)mb + koff[kf], ok);
        } else {
          a = 0.f;
          if (ok) a = x[mb + koff[kf]];
        }
        acc[kf] = mfma16x16x4(a, b, acc[kf]);
      }
    }
  }

  // cross-wave reduction in LDS (reuse lds_dy: 4 waves x KT x 256 floats)
  float* red = &lds_dy[0][0];  // >= 4*KT*256 floats available (KT<=4)
  __syncthreads();
#pragma unroll
  for (int kf = 0; kf < KT; ++kf) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      red[(wid * KT + kf) * 256 + (lk * 4 + reg) * 16 + lm] = acc[kf][reg];
    }
  }
  __syncthreads();
  // wave 0 folds the other 3 waves; result goes to this split's PARTIAL slab
  // (dw points at workspace [n_splits][CO*K] — plain stores, no atomics: the
  // deterministic reduce kernel below folds the split axis in fixed order).
  if (wid == 0) {
    float* slab = dw + (long)ms * (CO * (long)K);
#pragma unroll
    for (int kf = 0; kf < KT; ++kf) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        int e = (lk * 4 + reg) * 16 + lm;
        float v = red[kf * 256 + e] + red[(KT + kf) * 256 + e] +
                  red[(2 * KT + kf) * 256 + e] + red[(3 * KT + kf) * 256 + e];
        int kk = k0 + kf * 16 + lk * 4 + reg;
        int co = co0 + lm;
        if (kk < K && co < CO) slab[co * K + kk] = v;
      }
    }
  }
  if (WANT_BIAS && blockIdx.x == 0) {
    // db: every wave summed its own m rows; fold lk groups then waves
    db_part += __shfl_down(db_part, 16);
    db_part += __shfl_down(db_part, 32);
    __syncthreads();
    if (lk == 0) red[wid * 16 + lm] = db_part;
    __syncthreads();
    if (tid < 16 && co0 + tid < CO)
      db[(long)ms * CO + co0 + tid] =
          red[tid] + red[16 + tid] + red[32 + tid] + red[48 + tid];
  }
}

// Fold the split axis of the wgrad partial slabs in fixed order
// (deterministic — no float-atomic reassociation): out[e] = sum_s in[s*E+e].
// Tiled so small E with large S still fills the chip: a block owns a 16-wide
// e-tile; its 16 thread-columns stride the s axis and fold through LDS in a
// FIXED tree order (column c holds s ≡ c (mod 16), folded c..c+8..).
__global__ void reduce_splits_k(const float* __restrict__ in,
                                float* __restrict__ out, long E, int S) {
  // block (x = 16-wide e-tile, y = z of Z): folds s in {z, z+Z, ...} in a
  // FIXED order (deterministic) into out[z*E + e]. Z = gridDim.y; the host
  // helper runs one pass (Z=1) for small S or two passes (Z=16 partials,
  // then Z=1 over the 16 partials) so large split counts still fill the chip.
  __shared__ float red[16][17];
  int Z = gridDim.y;
  int z = blockIdx.y;
  int e0 = blockIdx.x * 16;
  int el = threadIdx.x & 15;
  int sl = threadIdx.x >> 4;
  long e = e0 + el;
  float v = 0.f;
  if (e < E)
    for (int s = z + sl * Z; s < S; s += 16 * Z) v += in[(long)s * E + e];
  red[sl][el] = v;
  __syncthreads();
  for (int step = 8; step > 0; step >>= 1) {
    if (sl < step) red[sl][el] += red[sl + step][el];
    __syncthreads();
  }
  if (sl == 0 && e < E) out[(long)z * E + e] = red[0][el];
}

// Single-LAUNCH fold for small element counts (BN statistics slabs,
// bias-gradient slabs): a wave owns one output element, its 64 lanes stride
// the split axis, and the cross-lane fold is a fixed xor tree — still a
// deterministic fixed order for any given (E, S). Replaces the two-pass
// launch pair where dispatch latency, not bandwidth, is the cost.
__global__ __launch_bounds__(256) void reduce_splits_wave_k(
    const float* __restrict__ in, float* __restrict__ out, long E, int S) {
  long e = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
  if (e >= E) return;
  int l = threadIdx.x & 63;
  float v = 0.f;
  for (int s = l; s < S; s += 64) v += in[(long)s * E + e];
#pragma unroll
  for (int d = 32; d > 0; d >>= 1) v += __shfl_xor(v, d);
  if (l == 0) out[e] = v;
}

// ---------------------------------------------------------------------------
// Whole-image-LDS conv (fwd and dgrad), NCHW fp32 — the LeNet-class fp32
// engine (fp32 at the bf16 engine's structural quality).
// The generic f32 gather kernels above pay one SCALAR global gather per lane
// per MFMA k-slice (NCHW has no contiguous channel vectors), measured
// 478 us/call on the conv2 dgrad at batch 8192 — pure L1/L2 latency. At
// LeNet scale whole input images are tiny (conv2 fwd: 5.8 KB/image fp32),
// so a block stages its 2-3 images AND the whole repacked weight panel in
// LDS once, then the k loop runs barrier-free: per mfma_f32_16x16x4 k-slice
// each lane does one ds_read_b32 from the image (~2 cyc) instead of a
// global gather. fp32 MFMA is exact f32 at the 155 TF vector rate
// (guide §3), so the fp32 dtype contract is preserved bit-for-bit vs an
// fmaf chain.
//   A (lane l): A[m = l&15][k = l>>4] — one float
//   B (lane l): B[k = l>>4][c = l&15] from lds_w[k][OCHp]
//   D (lane l): col c = l&15, row m = (l>>4)*4 + reg
// DGRAD consumes dy images with the transposed weight panel and the
// (pad-kh, pad-kw) tap predicate; fwd is routed pad==0 only (LeNet).
// ---------------------------------------------------------------------------
template <bool DGRAD, bool HAS_BIAS, int MW, int NW, int IMG_MAX>
__global__ __launch_bounds__(256) void conv_smallimg_f32_k(
    const float* __restrict__ x, const float* __restrict__ w,
    const float* __restrict__ bias, float* __restrict__ y,
    int CI, int H, int W, int CO, int KH, int KW, int OH, int OW,
    int pad, int M, int K, int N) {
  const int ICH = DGRAD ? CO : CI;        // channels of the staged tensor
  const int IH = DGRAD ? OH : H;
  const int IW = DGRAD ? OW : W;
  const int OCH = DGRAD ? CI : CO;        // valid output columns
  const int OCHp = (OCH + 7) / 8 * 8;
  const int OHo = DGRAD ? H : OH;         // output map dims
  const int OWo = DGRAD ? W : OW;
  const int ohow = OHo * OWo;             // out pixels per image
  const int img_elems = ICH * IH * IW;
  const int KHKW = KH * KW;

  extern __shared__ __attribute__((aligned(16))) char smemc[];
  float* lds_img = reinterpret_cast<float*>(smemc);      // [img][ICH][IH][IW]
  float* lds_w = lds_img + IMG_MAX * img_elems;          // [K][OCHp]
  int* lds_off = reinterpret_cast<int*>(lds_w + (long)K * OCHp);  // [K]
  int* lds_dk = lds_off + K;                             // [K] (dgrad)

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lm = l & 15;
  const int lk = l >> 4;
  constexpr int BM = 4 * MW * 16;
  const int m0 = blockIdx.x * BM;
  const int img_lo = m0 / ohow;
  const int img_n =
      min(IMG_MAX, (min(m0 + BM, M) - 1) / ohow - img_lo + 1);

  {  // stage images (linear copy; float4 when the image size permits)
    const float* src = x + (long)img_lo * img_elems;
    long total = (long)img_n * img_elems;
    if ((img_elems & 3) == 0) {
      for (long i = tid; i < (total >> 2); i += 256)
        reinterpret_cast<float4*>(lds_img)[i] =
            reinterpret_cast<const float4*>(src)[i];
    } else {
      for (long i = tid; i < total; i += 256) lds_img[i] = src[i];
    }
  }
  for (int idx = tid; idx < K * OCHp; idx += 256) {  // weight panel
    int k = idx / OCHp, c = idx % OCHp;
    float v = 0.f;
    if (c < OCH) {
      if (DGRAD) {
        v = w[((long)(k / KHKW) * CI + c) * KHKW + k % KHKW];
      } else {
        v = w[(long)c * K + k];   // w[CO][CI*KH*KW], k already canonical
      }
    }
    lds_w[idx] = v;
  }
  for (int k = tid; k < K; k += 256) {  // per-k tables
    int r = k % KHKW;
    int kh = r / KW, kw = r % KW;
    if (DGRAD) {
      lds_off[k] = ((k / KHKW) * OH + pad - kh) * OW + (pad - kw);
      lds_dk[k] = ((pad - kh) << 16) | ((pad - kw) & 0xffff);
    } else {
      lds_off[k] = ((k / KHKW) * H + kh) * W + kw;
    }
  }
  __syncthreads();

  // per-fragment m decode (this lane's A row is m = ... + lm)
  int mbase[MW], mih[MW], miw[MW];
  bool mok[MW];
#pragma unroll
  for (int mi = 0; mi < MW; ++mi) {
    int m = m0 + (wid * MW + mi) * 16 + lm;
    mok[mi] = m < M;
    int mm = mok[mi] ? m : m0;
    int nimg = mm / ohow, r = mm % ohow;
    int oy = r / OWo, ox = r % OWo;
    mih[mi] = oy;
    miw[mi] = ox;
    // fwd: element = img + xoff[k] + oy*W + ox   (pad 0, stride 1)
    // dgrad: element = img + aoff[k] + ih*OW + iw
    mbase[mi] = (nimg - img_lo) * img_elems +
                (DGRAD ? oy * OW + ox : oy * W + ox);
  }

  f32x4 acc[MW][NW] = {};
#pragma unroll 2
  for (int k0 = 0; k0 < K; k0 += 4) {
    int k = k0 + lk;
    bool kin = k < K;
    int kc = kin ? k : 0;
    int aoff = lds_off[kc];
    float bw[NW];
#pragma unroll
    for (int ni = 0; ni < NW; ++ni) {
      int c = ni * 16 + lm;
      bw[ni] = (kin && c < OCHp) ? lds_w[kc * OCHp + c] : 0.f;
    }
    int dkh = 0, dkw = 0;
    if (DGRAD) {
      int dk = lds_dk[kc];
      dkh = dk >> 16;
      dkw = (short)(dk & 0xffff);
    }
#pragma unroll
    for (int mi = 0; mi < MW; ++mi) {
      bool ok = mok[mi] & kin;
      if (DGRAD) {
        int tH = mih[mi] + dkh, tW = miw[mi] + dkw;
        ok = ok && (unsigned)tH < (unsigned)OH && (unsigned)tW < (unsigned)OW;
      }
      float a = lds_img[ok ? mbase[mi] + aoff : 0];
      if (!ok) a = 0.f;
#pragma unroll
      for (int ni = 0; ni < NW; ++ni)
        acc[mi][ni] = mfma16x16x4(a, bw[ni], acc[mi][ni]);
    }
  }

  // epilogue: NCHW scattered stores (channel stride ohow)
#pragma unroll
  for (int mi = 0; mi < MW; ++mi) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int m = m0 + (wid * MW + mi) * 16 + lk * 4 + reg;
      if (m >= M) continue;
      int nimg = m / ohow, r = m % ohow;
#pragma unroll
      for (int ni = 0; ni < NW; ++ni) {
        int c = ni * 16 + lm;
        if (c >= OCH) continue;
        float v = acc[mi][ni][reg];
        if (HAS_BIAS) v += bias[c];
        y[((long)nimg * OCH + c) * ohow + r] = v;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Fused fp32 stem: Conv2d(CI==1, s1 p0) + MaxPool2x2 + ReLU in one kernel
// (the fp32-tier port of conv_pool_stem_bf16_k — VERDICT item 3). NCHW in
// and out. Thread = (n, pool cell), all CO groups; x tile rows load as
// float2 vectors (row base is an even pixel -> 8-B aligned). Writes only
// the pooled output + argmax: the separate conv-y round trip at batch 8192
// was 2x189 MB of fp32 traffic.
// ---------------------------------------------------------------------------
template <int KH, int KW>
__global__ __launch_bounds__(256) void conv_pool_stem_f32_k(
    const float* __restrict__ x, const float* __restrict__ w,
    const float* __restrict__ bias, float* __restrict__ p,
    unsigned char* __restrict__ idx,
    int H, int W, int CO, int OH2, int OW2, int N) {
  extern __shared__ float lds_wf[];   // [CO][KH*KW]
  const int tid = threadIdx.x;
  constexpr int Kc = KH * KW;
  for (int i = tid; i < CO * Kc; i += 256) lds_wf[i] = w[i];
  __syncthreads();

  constexpr int TH = KH + 1, TW = KW + 1;
  const long ohow2 = (long)OH2 * OW2;
  long total = (long)N * ohow2;
  for (long t = (long)blockIdx.x * blockDim.x + tid; t < total;
       t += (long)gridDim.x * blockDim.x) {
    int ow2 = t % OW2;
    long r2 = t / OW2;
    int oh2 = r2 % OH2;
    int n = r2 / OH2;

    float xt[TH * TW];
    const float* xb = x + (long)n * H * W + (oh2 * 2) * W + ow2 * 2;
#pragma unroll
    for (int ty = 0; ty < TH; ++ty)
#pragma unroll
      for (int tx = 0; tx < TW; tx += 2) {
        float2 v = *reinterpret_cast<const float2*>(&xb[ty * W + tx]);
        xt[ty * TW + tx] = v.x;
        xt[ty * TW + tx + 1] = v.y;
      }

    for (int co = 0; co < CO; ++co) {
      float a0, a1, a2, a3;
      float bv = bias[co];
      a0 = a1 = a2 = a3 = bv;
      const float* wr = &lds_wf[co * Kc];
#pragma unroll
      for (int kh = 0; kh < KH; ++kh)
#pragma unroll
        for (int kw = 0; kw < KW; ++kw) {
          float wv = wr[kh * KW + kw];
          const float* xr = &xt[kh * TW + kw];
          a0 += wv * xr[0];
          a1 += wv * xr[1];
          a2 += wv * xr[TW];
          a3 += wv * xr[TW + 1];
        }
      float m01 = fmaxf(a0, a1), m23 = fmaxf(a2, a3);
      float mx = fmaxf(m01, m23);
      unsigned char am;
      if (mx == a0) am = 0; else if (mx == a1) am = 1;
      else if (mx == a2) am = 2; else am = 3;
      long off = ((long)n * CO + co) * ohow2 + oh2 * OW2 + ow2;
      p[off] = fmaxf(mx, 0.f);
      idx[off] = am;
    }
  }
}

// ---------------------------------------------------------------------------
// Fused fp32 stem wgrad: reconstructs the conv-space dy from
// (d_pool, pool_y, argmax) on the fly — dy[n,co,oh,ow] = (idx==quadrant &&
// pool_y>0) ? d_pool : 0 — and reduces dw/db with exact-f32 MFMA
// (16x16x4: D[k<=48][co<=16] over 4-m slices). Waves split the m chunk
// (wave-local compute after one cooperative stage), each writing its own
// partial slab row; reduce_splits folds ms*4 slabs deterministically.
// k space is per-kh-row padded like the bf16 twin: K = KH*8 (CI==1).
// ---------------------------------------------------------------------------
template <int MC>
__global__ __launch_bounds__(256) void conv2d_wgrad_pool_f32_k(
    const float* __restrict__ dpool, const float* __restrict__ pool_y,
    const unsigned char* __restrict__ pidx, const float* __restrict__ x,
    float* __restrict__ dw_ws, float* __restrict__ db_ws,
    int H, int W, int CO, int KH, int KW,
    int OH, int OW, int M, int K, int Kcan, int m_per_split, long x_elems) {
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lm = l & 15;
  const int lk = l >> 4;
  const int ms = blockIdx.z;
  const int co0 = blockIdx.y * 16;
  const int OH2 = OH / 2, OW2 = OW / 2;
  const int OHOW = OH * OW;
  const int KROW = 8;                  // CI==1, KW<=8
  const int NKF = 3;                   // 48 k slots >= K=40

  __shared__ float lds_x[MC][NKF * 16 + 1];   // [m][48k] (+1: bank stagger)
  __shared__ float lds_dy[MC][16 + 1];        // [m][16co]
  __shared__ int lds_mb[MC];                  // x base of the patch
  __shared__ int lds_mpq[MC];                 // pooled offset | quadrant

  f32x4 acc[NKF] = {};
  float db_part = 0.f;

  const int m_begin = ms * m_per_split;
  const int m_end = min(M, m_begin + m_per_split);
  const int mw0 = wid * (MC / 4);      // this wave's m-slice of each chunk

  for (int mc = m_begin; mc < m_end; mc += MC) {
    __syncthreads();
    for (int mi = tid; mi < MC; mi += 256) {
      int m = mc + mi;
      int mm = m < m_end ? m : 0;
      int n = mm / OHOW, r = mm % OHOW;
      int oh = r / OW, ow = r % OW;
      lds_mb[mi] = (n * H + oh) * W + ow;
      int q = (oh & 1) * 2 + (ow & 1);
      // pooled offset is per-channel-plane: final index adds co*OH2*OW2
      lds_mpq[mi] = ((((n * CO + co0) * OH2 + oh / 2) * OW2 + ow / 2) << 2) | q;
    }
    __syncthreads();
    // x patch rows: [m][kh*8 + kw], 8-wide padded rows (3 valid beyond KW=5
    // read in-row garbage, never written back)
    // mloc FASTEST across threads: consecutive lanes then read consecutive
    // conv pixels (stride-1 x / stride-0-or-1 pooled offsets) — with co or
    // kh fastest every gating load is a separate NCHW-strided cacheline
    // (measured 394 us/step, worse than the unfused pair it replaced)
    for (int idx2 = tid; idx2 < MC * KH * 2; idx2 += 256) {
      int mloc = idx2 % MC;
      int rest = idx2 / MC;
      int kh = rest >> 1;
      int half = rest & 1;
      int m = mc + mloc;
      float4 v = {0.f, 0.f, 0.f, 0.f};
      if (m < m_end) {
        long a = (long)lds_mb[mloc] + kh * W + half * 4;
        // per-ELEMENT tail bound: a whole-vector clamp would shift the
        // still-valid kw=4 slot for the last image's final rows (measured:
        // 7/250 dw elements off by ~2.4e-2 relative)
        v.x = (a < x_elems) ? x[a] : 0.f;
        v.y = (a + 1 < x_elems) ? x[a + 1] : 0.f;
        v.z = (a + 2 < x_elems) ? x[a + 2] : 0.f;
        v.w = (a + 3 < x_elems) ? x[a + 3] : 0.f;
      }
      float* dst = &lds_x[mloc][kh * 8 + half * 4];
      dst[0] = v.x; dst[1] = v.y; dst[2] = v.z; dst[3] = v.w;
    }
    // gated conv-space dy [m][16co] (co plane stride OH2*OW2 in dpool)
    for (int idx2 = tid; idx2 < MC * 16; idx2 += 256) {
      int mloc = idx2 % MC;
      int co = idx2 / MC;
      int m = mc + mloc;
      float v = 0.f;
      if (m < m_end && co0 + co < CO) {
        int mpq = lds_mpq[mloc];
        long off = (long)(mpq >> 2) + (long)co * (OH2 * OW2);
        unsigned char q = (unsigned char)(mpq & 3);
        if (pidx[off] == q && pool_y[off] > 0.f) v = dpool[off];
      }
      lds_dy[mloc][co] = v;
    }
    __syncthreads();

    // wave-local MFMA over its MC/4 m rows: A[k][m-slice], B[m-slice][co]
    for (int mstep = 0; mstep < MC / 4 / 4; ++mstep) {
      int mrow = mw0 + mstep * 4 + lk;
      float b = lds_dy[mrow][lm];
      db_part += b;
#pragma unroll
      for (int kf = 0; kf < NKF; ++kf) {
        float a = lds_x[mrow][kf * 16 + lm];
        acc[kf] = mfma16x16x4(a, b, acc[kf]);
      }
    }
  }

  // per-wave partial slab: split index ms*4 + wid
  {
    float* slab = dw_ws + (long)(ms * 4 + wid) * (CO * (long)Kcan);
#pragma unroll
    for (int kf = 0; kf < NKF; ++kf) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        int kk = kf * 16 + lk * 4 + reg;
        int co = co0 + lm;
        if (kk < K && co < CO) {
          int kh = kk / KROW, r = kk % KROW;
          if (r < KW)
            slab[(long)co * Kcan + kh * KW + r] = acc[kf][reg];
        }
      }
    }
  }
  {
    // db: each lane holds partials for co=lm over its (lk, wave) m rows
    db_part += __shfl_down(db_part, 16);
    db_part += __shfl_down(db_part, 32);
    if (lk == 0 && co0 + lm < CO)
      db_ws[(long)(ms * 4 + wid) * CO + co0 + lm] = db_part;
  }
}
