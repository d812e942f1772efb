// ResNet-family support kernels, NHWC bf16 (BASELINE.json config 5):
// BatchNorm2d (training stats, fp32 accumulation), MaxPool 3x3/s2/p1 with
// gather-style deterministic backward, global average pool, fused
// residual-add+ReLU, standalone ReLU. Channel axis is fastest (NHWC) so every
// per-channel op vectorizes as bf16x8 with f32x8-style unpacked math.
#include "common.h"
// bf16 types come from bf16.hip (this file is included after it in ext.hip)

// ---------------------------------------------------------------------------
// BatchNorm2d forward, training mode.
//  pass 1: per-channel sum / sumsq partials (f32), atomicAdd into stats[C][2]
//  pass 2: mean/invstd + running-stat update (tiny)
//  pass 3: normalize y = (x - mean) * invstd * gamma + beta  (+opt ReLU)
// ---------------------------------------------------------------------------
__global__ void bn_stats_k(const bf16* __restrict__ x, float* __restrict__ stats,
                           long npix, int Cp) {
  // thread t owns channel group cg = t % ncg and marches pixels with its
  // pixel-lane: consecutive threads cover a FULL contiguous pixel row
  // (coalesced), unlike a per-channel-group grid (16B per 128B line).
  int ncg = Cp / 8;                  // <= 256 (Cp <= 2048)
  int cg = threadIdx.x % ncg;
  int pl = threadIdx.x / ncg;
  int lanes = blockDim.x / ncg;      // pixel-lanes per block
  __shared__ float red[256][8];
  float s[8] = {}, ss[8] = {};
  if (pl < lanes) {
    for (long i = (long)blockIdx.x * lanes + pl; i < npix;
         i += (long)gridDim.x * lanes) {
      bf16x8 v = *reinterpret_cast<const bf16x8*>(&x[i * Cp + cg * 8]);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = to_f32(v[j]);
        s[j] += f;
        ss[j] += f * f;
      }
    }
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) red[threadIdx.x][j] = s[j];
  __syncthreads();
  if (pl == 0) {
    for (int l = 1; l < lanes; ++l)
#pragma unroll
      for (int j = 0; j < 8; ++j) s[j] += red[cg + l * ncg][j];
  }
  __syncthreads();
#pragma unroll
  for (int j = 0; j < 8; ++j) red[threadIdx.x][j] = ss[j];
  __syncthreads();
  if (pl == 0) {
    for (int l = 1; l < lanes; ++l)
#pragma unroll
      for (int j = 0; j < 8; ++j) ss[j] += red[cg + l * ncg][j];
    // per-block partial slab (plain stores; deterministic split-reduce
    // folds the block axis — atomics on [C][2] serialize on 1-2 lines)
    float* slab = stats + (long)blockIdx.x * (2 * 8 * ncg);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      slab[(cg * 8 + j) * 2 + 0] = s[j];
      slab[(cg * 8 + j) * 2 + 1] = ss[j];
    }
  }
}

__global__ void bn_finalize_k(const float* __restrict__ stats,
                              float* __restrict__ mean,
                              float* __restrict__ invstd,
                              float* __restrict__ running_mean,
                              float* __restrict__ running_var,
                              long npix, int C, int Cp, float eps,
                              float momentum, int update_running) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= Cp) return;
  if (c >= C) {  // zero the channel pads here (saves an at::zeros launch;
                 // downstream kernels multiply pads by these values)
    mean[c] = 0.f;
    invstd[c] = 0.f;
    return;
  }
  float m = stats[c * 2] / (float)npix;
  float var = stats[c * 2 + 1] / (float)npix - m * m;
  var = fmaxf(var, 0.f);
  mean[c] = m;
  invstd[c] = rsqrtf(var + eps);
  if (update_running) {
    float unbiased = var * (float)npix / (float)max(1L, npix - 1);
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * m;
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
  }
}

// Fused slab-fold + finalize: ONE launch replacing reduce_splits x2 +
// bn_finalize_k (3 dispatches of launch latency per BN forward). A wave owns
// one channel; its 64 lanes stride the per-block partial slabs reading the
// (sum, sumsq) pair as one 8-byte load, then fold through a fixed xor tree
// (deterministic order for a given S) and lane 0 does the finalize math.
__global__ __launch_bounds__(256) void bn_finalize_fold_k(
    const float* __restrict__ ws, float* __restrict__ mean,
    float* __restrict__ invstd, float* __restrict__ running_mean,
    float* __restrict__ running_var, long npix, int C, int Cp, float eps,
    float momentum, int update_running, int S,
    const float* __restrict__ gamma = nullptr,
    const float* __restrict__ beta = nullptr,
    float* __restrict__ aff = nullptr) {
  // ONE channel per 256-lane block (round 2: the 4-channels-per-block form
  // ran on <=128 blocks — chip underfilled, 15.7 us/call at trivial work).
  // aff != nullptr: additionally emit the per-channel affine the BN-apply
  // FUSION consumes (aff[c] = gamma*invstd, aff[Cp+c] = beta - mean*scale;
  // pads -> 0 so fused staging zeroes padded channels) — no extra launch.
  int c = blockIdx.x;
  int t = threadIdx.x;
  if (c >= Cp) return;
  if (c >= C) {  // zero the channel pads (downstream kernels multiply by them)
    if (t == 0) {
      mean[c] = 0.f;
      invstd[c] = 0.f;
      if (aff != nullptr) {
        aff[c] = 0.f;
        aff[Cp + c] = 0.f;
      }
    }
    return;
  }
  long E = (long)Cp * 2;
  float s = 0.f, ss = 0.f;
  for (int sp = t; sp < S; sp += 256) {
    const float* p = ws + (long)sp * E + (long)c * 2;
    s += p[0];
    ss += p[1];
  }
#pragma unroll
  for (int d = 32; d > 0; d >>= 1) {
    s += __shfl_xor(s, d);
    ss += __shfl_xor(ss, d);
  }
  __shared__ float red[4][2];
  if ((t & 63) == 0) {
    red[t >> 6][0] = s;
    red[t >> 6][1] = ss;
  }
  __syncthreads();
  if (t != 0) return;
  s = red[0][0] + red[1][0] + red[2][0] + red[3][0];
  ss = red[0][1] + red[1][1] + red[2][1] + red[3][1];
  float m = s / (float)npix;
  float var = fmaxf(ss / (float)npix - m * m, 0.f);
  float is = rsqrtf(var + eps);
  mean[c] = m;
  invstd[c] = is;
  if (aff != nullptr) {
    float sc = gamma[c] * is;
    aff[c] = sc;
    aff[Cp + c] = beta[c] - m * sc;
  }
  if (update_running) {
    float unbiased = var * (float)npix / (float)max(1L, npix - 1);
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * m;
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
  }
}

// Backward twin: folds the (sum_dy, sum_dy_xhat) partial slabs in one launch
// and ALSO writes dbeta/dgamma densely (their previous extraction from
// red[Cp][2] cost two strided-copy kernels per BN layer).
__global__ __launch_bounds__(256) void bn_bwd_fold_k(
    const float* __restrict__ ws, float* __restrict__ red_out,
    float* __restrict__ dbeta, float* __restrict__ dgamma, int C, int Cp,
    int S) {
  // one channel per 256-lane block (see bn_finalize_fold_k); also writes
  // dbeta/dgamma densely
  int c = blockIdx.x;
  int t = threadIdx.x;
  if (c >= Cp) return;
  long E = (long)Cp * 2;
  float s0 = 0.f, s1 = 0.f;
  for (int sp = t; sp < S; sp += 256) {
    const float* p = ws + (long)sp * E + (long)c * 2;
    s0 += p[0];
    s1 += p[1];
  }
#pragma unroll
  for (int d = 32; d > 0; d >>= 1) {
    s0 += __shfl_xor(s0, d);
    s1 += __shfl_xor(s1, d);
  }
  __shared__ float red[4][2];
  if ((t & 63) == 0) {
    red[t >> 6][0] = s0;
    red[t >> 6][1] = s1;
  }
  __syncthreads();
  if (t != 0) return;
  s0 = red[0][0] + red[1][0] + red[2][0] + red[3][0];
  s1 = red[0][1] + red[1][1] + red[2][1] + red[3][1];
  red_out[c * 2] = s0;
  red_out[c * 2 + 1] = s1;
  if (c < C) {
    dbeta[c] = s0;
    dgamma[c] = s1;
  }
}

template <bool RELU>
__global__ void bn_apply_k(const bf16* __restrict__ x, bf16* __restrict__ y,
                           const float* __restrict__ mean,
                           const float* __restrict__ invstd,
                           const float* __restrict__ gamma,
                           const float* __restrict__ beta, long npix, int Cp,
                           int C) {
  int ncg = Cp / 8;
  int cg = threadIdx.x % ncg;
  int pl = threadIdx.x / ncg;
  int lanes = blockDim.x / ncg;
  if (pl >= lanes) return;
  float sc[8], sh[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int c = cg * 8 + j;
    if (c < C) {
      sc[j] = gamma[c] * invstd[c];
      sh[j] = beta[c] - mean[c] * sc[j];
    } else {
      sc[j] = 0.f;
      sh[j] = 0.f;
    }
  }
  for (long i = (long)blockIdx.x * lanes + pl; i < npix;
       i += (long)gridDim.x * lanes) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(&x[i * Cp + cg * 8]);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = to_f32(v[j]) * sc[j] + sh[j];
      if (RELU) f = fmaxf(f, 0.f);
      o[j] = to_bf16(f);
    }
    *reinterpret_cast<bf16x8*>(&y[i * Cp + cg * 8]) = o;
  }
}

// eval mode: same apply with running stats (host passes mean=running_mean,
// invstd = rsqrt(running_var+eps) precomputed by bn_finalize-eval path)

// ---------------------------------------------------------------------------
// BatchNorm backward.
//  reduce: dbeta[c] = sum dy ; dgamma[c] = sum dy * xhat
//  apply : dx = gamma*invstd * (dy - dbeta/npix - xhat * dgamma/npix)
// ---------------------------------------------------------------------------
// GATE: 0 = none, 1 = read the saved relu output (yact), 2 = RECOMPUTE the
// relu gate from (x, affine) — the fused-apply path never materializes
// y_bn, and recomputing (scale*x+shift > 0) is 2 VALU instead of a whole
// extra tensor read.
template <int GATE>
__global__ void bn_bwd_reduce_k(const bf16* __restrict__ dy,
                                const bf16* __restrict__ x,
                                const bf16* __restrict__ yact,
                                const float* __restrict__ mean,
                                const float* __restrict__ invstd,
                                float* __restrict__ red_out,  // [C][2] partial
                                long npix, int Cp,
                                const float* __restrict__ gamma = nullptr,
                                const float* __restrict__ beta = nullptr) {
  int ncg = Cp / 8;
  int cg = threadIdx.x % ncg;
  int pl = threadIdx.x / ncg;
  int lanes = blockDim.x / ncg;
  __shared__ float red[256][8];
  float db[8] = {}, dg[8] = {};
  float mu[8], is[8], sc[8], sh[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    mu[j] = mean[cg * 8 + j];
    is[j] = invstd[cg * 8 + j];
    if (GATE == 2) {
      sc[j] = gamma[cg * 8 + j] * is[j];
      sh[j] = beta[cg * 8 + j] - mu[j] * sc[j];
    }
  }
  if (pl < lanes) {
    for (long i = (long)blockIdx.x * lanes + pl; i < npix;
         i += (long)gridDim.x * lanes) {
      bf16x8 g = *reinterpret_cast<const bf16x8*>(&dy[i * Cp + cg * 8]);
      bf16x8 v = *reinterpret_cast<const bf16x8*>(&x[i * Cp + cg * 8]);
      bf16x8 ya;
      if (GATE == 1) ya = *reinterpret_cast<const bf16x8*>(&yact[i * Cp + cg * 8]);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float gf = to_f32(g[j]);
        float xf = to_f32(v[j]);
        if (GATE == 1 && !(to_f32(ya[j]) > 0.f)) gf = 0.f;
        // the fused forward stored bf16(max(sc*x+sh, 0)): gate on the SAME
        // bf16-rounded value so backward matches the forward bit-for-bit
        if (GATE == 2 && !(to_f32(to_bf16(sc[j] * xf + sh[j])) > 0.f)) gf = 0.f;
        db[j] += gf;
        dg[j] += gf * (xf - mu[j]) * is[j];
      }
    }
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) red[threadIdx.x][j] = db[j];
  __syncthreads();
  if (pl == 0)
    for (int l = 1; l < lanes; ++l)
#pragma unroll
      for (int j = 0; j < 8; ++j) db[j] += red[cg + l * ncg][j];
  __syncthreads();
#pragma unroll
  for (int j = 0; j < 8; ++j) red[threadIdx.x][j] = dg[j];
  __syncthreads();
  if (pl == 0) {
    for (int l = 1; l < lanes; ++l)
#pragma unroll
      for (int j = 0; j < 8; ++j) dg[j] += red[cg + l * ncg][j];
    float* slab = red_out + (long)blockIdx.x * (2 * 8 * ncg);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      slab[(cg * 8 + j) * 2 + 0] = db[j];
      slab[(cg * 8 + j) * 2 + 1] = dg[j];
    }
  }
}

template <int GATE>
__global__ void bn_bwd_apply_k(const bf16* __restrict__ dy,
                               const bf16* __restrict__ x,
                               const bf16* __restrict__ yact,
                               const float* __restrict__ mean,
                               const float* __restrict__ invstd,
                               const float* __restrict__ gamma,
                               const float* __restrict__ red,
                               bf16* __restrict__ dx, long npix, int Cp, int C,
                               const float* __restrict__ beta = nullptr) {
  int ncg = Cp / 8;
  int cg = threadIdx.x % ncg;
  int pl = threadIdx.x / ncg;
  int lanes = blockDim.x / ncg;
  if (pl >= lanes) return;
  float mu[8], is[8], gsc[8], mdb[8], mdg[8], sh[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int c = cg * 8 + j;
    if (c < C) {
      mu[j] = mean[c];
      is[j] = invstd[c];
      gsc[j] = gamma[c] * is[j];
      mdb[j] = red[c * 2] / (float)npix;
      mdg[j] = red[c * 2 + 1] / (float)npix;
      if (GATE == 2) sh[j] = beta[c] - mu[j] * gsc[j];
    } else {
      mu[j] = is[j] = gsc[j] = mdb[j] = mdg[j] = sh[j] = 0.f;
    }
  }
  for (long i = (long)blockIdx.x * lanes + pl; i < npix;
       i += (long)gridDim.x * lanes) {
    bf16x8 g = *reinterpret_cast<const bf16x8*>(&dy[i * Cp + cg * 8]);
    bf16x8 v = *reinterpret_cast<const bf16x8*>(&x[i * Cp + cg * 8]);
    bf16x8 ya;
    if (GATE == 1) ya = *reinterpret_cast<const bf16x8*>(&yact[i * Cp + cg * 8]);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = to_f32(g[j]);
      float xf = to_f32(v[j]);
      if (GATE == 1 && !(to_f32(ya[j]) > 0.f)) gf = 0.f;
      if (GATE == 2 && !(to_f32(to_bf16(gsc[j] * xf + sh[j])) > 0.f)) gf = 0.f;
      float xhat = (xf - mu[j]) * is[j];
      float val = gsc[j] * (gf - mdb[j] - xhat * mdg[j]);
      o[j] = to_bf16(val);
    }
    *reinterpret_cast<bf16x8*>(&dx[i * Cp + cg * 8]) = o;
  }
}

// ---------------------------------------------------------------------------
// MaxPool kernel=3 stride=2 pad=1 (ResNet stem), NHWC bf16, argmax u8 (0..8).
// Backward is GATHER-style (each input pixel checks the <=4 windows that can
// contain it) — deterministic, no atomics despite overlapping windows.
// ---------------------------------------------------------------------------
__global__ void maxpool3x3s2_fwd_k(const bf16* __restrict__ x,
                                   bf16* __restrict__ y,
                                   unsigned char* __restrict__ idx, int N,
                                   int H, int W, int Cp, int OH, int OW) {
  long total = (long)N * OH * OW * (Cp / 8);
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += (long)gridDim.x * blockDim.x) {
    int cg = t % (Cp / 8);
    long r = t / (Cp / 8);
    int ow = r % OW;
    long r2 = r / OW;
    int oh = r2 % OH;
    int n = r2 / OH;
    int ih0 = oh * 2 - 1, iw0 = ow * 2 - 1;
    float best[8];
    unsigned char arg[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) { best[j] = -3.4e38f; arg[j] = 0; }
    for (int kh = 0; kh < 3; ++kh) {
      int ih = ih0 + kh;
      if (ih < 0 || ih >= H) continue;
      for (int kw = 0; kw < 3; ++kw) {
        int iw = iw0 + kw;
        if (iw < 0 || iw >= W) continue;
        bf16x8 v = *reinterpret_cast<const bf16x8*>(
            &x[((long)(n * H + ih) * W + iw) * Cp + cg * 8]);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = to_f32(v[j]);
          if (f > best[j]) { best[j] = f; arg[j] = (unsigned char)(kh * 3 + kw); }
        }
      }
    }
    long off = (((long)(n * OH + oh) * OW + ow) * Cp) + cg * 8;
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = to_bf16(best[j]);
    *reinterpret_cast<bf16x8*>(&y[off]) = o;
    unsigned char a8[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) a8[j] = arg[j];
    *reinterpret_cast<unsigned long long*>(&idx[off]) =
        *reinterpret_cast<unsigned long long*>(a8);
  }
}

template <int PH>
__global__ void maxpool3x3s2_bwd_k(const bf16* __restrict__ dy,
                                   const unsigned char* __restrict__ idx,
                                   bf16* __restrict__ dx, int N, int H, int W,
                                   int Cp, int OH, int OW) {
  // PAIRED-COLUMN gather (k3/s2/p1). A thread owns the adjacent input pixel
  // pair (iw=2k, 2k+1) of one row: the pair's column candidates collapse to
  // ow=k (serves both: kw=1 for the even pixel, kw=2 for the odd) and
  // ow=k+1 (kw=0, odd pixel only) — 2 window loads produce 2 outputs, and
  // the two 16-byte stores are CONTIGUOUS (a stride-2 per-pixel parity
  // split was measured write-bound: every 128 B line half-covered).
  // Rows keep the parity split (even ih: 1 candidate row, odd: 2), so the
  // avg load count is 1.5/pixel vs 4 for the flat 2x2 set. Lower bounds
  // never trip; ow=k+1 and the odd class's second row are hardware-bounds-
  // checked buffer loads (OOB reads 0 and idx 0 pairs with want!=0 rows).
  int Hc = (H - PH + 1) / 2, W2 = (W + 1) / 2;
  long total = (long)N * Hc * W2 * (Cp / 8);
  const long dy_elems = (long)N * OH * OW * Cp;
  const bool use_buf = rsrc_fits(dy_elems);
  buf_rsrc_t dyr = make_bf16_rsrc(dy, dy_elems);
  buf_rsrc_t idxr = make_u8_rsrc(idx, dy_elems);
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += (long)gridDim.x * blockDim.x) {
    int cg = t % (Cp / 8);
    long r = t / (Cp / 8);
    int k = r % W2;
    long r2 = r / W2;
    int ihc = r2 % Hc;
    int n = r2 / Hc;
    int ih = 2 * ihc + PH;
    float acce[8] = {}, acco[8] = {};
#pragma unroll
    for (int a = 0; a < (PH ? 2 : 1); ++a) {
      int oh = PH ? (ih - 1) / 2 + a : ih / 2;
      int kh = ih - (oh * 2 - 1);
      bool okr = (a == 0 || oh < OH);
      long off0 = (((long)(n * OH + oh) * OW + k) * Cp) + cg * 8;
      unsigned long long a0, a1;
      bf16x8 g0, g1;
      bool ok1 = okr && (k + 1 < OW);
      if (use_buf) {
        a0 = buf_load_u64(idxr, off0, okr);
        g0 = buf_load8(dyr, off0, okr);
        a1 = buf_load_u64(idxr, off0 + Cp, ok1);
        g1 = buf_load8(dyr, off0 + Cp, ok1);
      } else {
        a0 = a1 = 0;
        g0 = g1 = bf8_zero();
        if (okr) {
          a0 = *reinterpret_cast<const unsigned long long*>(&idx[off0]);
          g0 = *reinterpret_cast<const bf16x8*>(&dy[off0]);
        }
        if (ok1) {
          a1 = *reinterpret_cast<const unsigned long long*>(&idx[off0 + Cp]);
          g1 = *reinterpret_cast<const bf16x8*>(&dy[off0 + Cp]);
        }
      }
      unsigned char we = (unsigned char)(kh * 3 + 1);   // even pixel: kw=1@k
      unsigned char wo0 = (unsigned char)(kh * 3 + 2);  // odd: kw=2@k
      unsigned char wo1 = (unsigned char)(kh * 3 + 0);  // odd: kw=0@k+1
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        unsigned char b0 = (a0 >> (8 * j)) & 0xff;
        float f0 = to_f32(g0[j]);
        if (b0 == we) acce[j] += f0;
        if (b0 == wo0) acco[j] += f0;
        if (((a1 >> (8 * j)) & 0xff) == wo1) acco[j] += to_f32(g1[j]);
      }
    }
    bf16x8 oe, oo;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      oe[j] = to_bf16(acce[j]);
      oo[j] = to_bf16(acco[j]);
    }
    long doff = ((long)(n * H + ih) * W + 2 * k) * Cp + cg * 8;
    *reinterpret_cast<bf16x8*>(&dx[doff]) = oe;
    if (2 * k + 1 < W) *reinterpret_cast<bf16x8*>(&dx[doff + Cp]) = oo;
  }
}

// ---------------------------------------------------------------------------
// Global average pool [N,H,W,Cp] -> [N,Cp]; bwd broadcasts dy/(H*W).
// ---------------------------------------------------------------------------
__global__ void gap_fwd_k(const bf16* __restrict__ x, bf16* __restrict__ y,
                          int N, long HW, int Cp) {
  long total = (long)N * (Cp / 8);
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += (long)gridDim.x * blockDim.x) {
    int cg = t % (Cp / 8);
    int n = t / (Cp / 8);
    float acc[8] = {};
    const bf16* base = x + (long)n * HW * Cp + cg * 8;
    for (long i = 0; i < HW; ++i) {
      bf16x8 v = *reinterpret_cast<const bf16x8*>(base + i * Cp);
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += to_f32(v[j]);
    }
    bf16x8 o;
    float inv = 1.f / (float)HW;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = to_bf16(acc[j] * inv);
    *reinterpret_cast<bf16x8*>(&y[(long)n * Cp + cg * 8]) = o;
  }
}

__global__ void gap_bwd_k(const bf16* __restrict__ dy, bf16* __restrict__ dx,
                          int N, long HW, int Cp) {
  long total = (long)N * HW * (Cp / 8);
  float inv = 1.f / (float)HW;
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += (long)gridDim.x * blockDim.x) {
    int cg = t % (Cp / 8);
    long r = t / (Cp / 8);
    long n = r / HW;
    bf16x8 g = *reinterpret_cast<const bf16x8*>(&dy[n * Cp + cg * 8]);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = to_bf16(to_f32(g[j]) * inv);
    *reinterpret_cast<bf16x8*>(&dx[t * 8]) = o;
  }
}

// ---------------------------------------------------------------------------
// Residual add (+ReLU) and standalone ReLU, bf16 vec8.
// add_relu: out = relu(a + b); bwd: da = db = dy * (out > 0).
// ---------------------------------------------------------------------------
template <bool RELU>
__global__ void add_bf16_k(const bf16* __restrict__ a, const bf16* __restrict__ b,
                           bf16* __restrict__ out, long n8) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    bf16x8 va = *reinterpret_cast<const bf16x8*>(a + i * 8);
    bf16x8 vb = *reinterpret_cast<const bf16x8*>(b + i * 8);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = to_f32(va[j]) + to_f32(vb[j]);
      if (RELU) f = fmaxf(f, 0.f);
      o[j] = to_bf16(f);
    }
    *reinterpret_cast<bf16x8*>(out + i * 8) = o;
  }
}

// Residual add with the BN-APPLY of one or both inputs FUSED (a = bn2's raw
// conv output, b = the downsample conv's raw output): the separate
// bn_apply passes (write + re-read of each whole tensor) disappear. Inputs
// round to bf16 after their affine so the result is bit-identical to the
// unfused apply-then-add.
template <bool AFFA, bool AFFB>
__global__ void add_aff_relu_bf16_k(
    const bf16* __restrict__ a, const bf16* __restrict__ b,
    bf16* __restrict__ out, long n8, int Cp8,
    const float* __restrict__ affa, const float* __restrict__ affb, int Cp) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    int cg = (int)(i % Cp8) * 8;
    bf16x8 va = *reinterpret_cast<const bf16x8*>(a + i * 8);
    bf16x8 vb = *reinterpret_cast<const bf16x8*>(b + i * 8);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float fa = to_f32(va[j]);
      float fb = to_f32(vb[j]);
      if (AFFA) fa = to_f32(to_bf16(fa * affa[cg + j] + affa[Cp + cg + j]));
      if (AFFB) fb = to_f32(to_bf16(fb * affb[cg + j] + affb[Cp + cg + j]));
      o[j] = to_bf16(fmaxf(fa + fb, 0.f));
    }
    *reinterpret_cast<bf16x8*>(out + i * 8) = o;
  }
}

__global__ void relu_gate_bf16_k(const bf16* __restrict__ dy,
                                 const bf16* __restrict__ out,
                                 bf16* __restrict__ dx, long n8) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    bf16x8 g = *reinterpret_cast<const bf16x8*>(dy + i * 8);
    bf16x8 v = *reinterpret_cast<const bf16x8*>(out + i * 8);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = (to_f32(v[j]) > 0.f) ? g[j] : to_bf16(0.f);
    *reinterpret_cast<bf16x8*>(dx + i * 8) = o;
  }
}

__global__ void relu_bf16_k(const bf16* __restrict__ x, bf16* __restrict__ y,
                            long n8) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(x + i * 8);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = to_bf16(fmaxf(to_f32(v[j]), 0.f));
    *reinterpret_cast<bf16x8*>(y + i * 8) = o;
  }
}
