// Elementwise + pooling + loss + optimizer kernels.
#include "common.h"

// ---------------------------------------------------------------------------
// ReLU fwd/bwd — float4-vectorized grid-stride (guide G13).
// ---------------------------------------------------------------------------
__global__ void relu_fwd_k(const float* __restrict__ x, float* __restrict__ y,
                           long n) {
  long i4 = (long)(blockIdx.x) * blockDim.x + threadIdx.x;
  long n4 = n >> 2;
  for (; i4 < n4; i4 += (long)gridDim.x * blockDim.x) {
    float4 v = reinterpret_cast<const float4*>(x)[i4];
    v.x = fmaxf(v.x, 0.f); v.y = fmaxf(v.y, 0.f);
    v.z = fmaxf(v.z, 0.f); v.w = fmaxf(v.w, 0.f);
    reinterpret_cast<float4*>(y)[i4] = v;
  }
  long tail = n & 3;
  long base = n - tail;
  long t = (long)(blockIdx.x) * blockDim.x + threadIdx.x;
  if (t < tail) y[base + t] = fmaxf(x[base + t], 0.f);
}

__global__ void relu_bwd_k(const float* __restrict__ dy,
                           const float* __restrict__ y,
                           float* __restrict__ dx, long n) {
  long i4 = (long)(blockIdx.x) * blockDim.x + threadIdx.x;
  long n4 = n >> 2;
  for (; i4 < n4; i4 += (long)gridDim.x * blockDim.x) {
    float4 g = reinterpret_cast<const float4*>(dy)[i4];
    float4 v = reinterpret_cast<const float4*>(y)[i4];
    g.x = v.x > 0.f ? g.x : 0.f; g.y = v.y > 0.f ? g.y : 0.f;
    g.z = v.z > 0.f ? g.z : 0.f; g.w = v.w > 0.f ? g.w : 0.f;
    reinterpret_cast<float4*>(dx)[i4] = g;
  }
  long tail = n & 3;
  long base = n - tail;
  long t = (long)(blockIdx.x) * blockDim.x + threadIdx.x;
  if (t < tail) dx[base + t] = y[base + t] > 0.f ? dy[base + t] : 0.f;
}

// ---------------------------------------------------------------------------
// Fused MaxPool2d(2,2) + ReLU (reference order relu(max_pool2d(x,2)); relu is
// monotone so one kernel computes both, saving a full activation round-trip).
// argmax stored u8 (0..3) for the backward scatter; maxval<=0 routes nothing.
// ---------------------------------------------------------------------------
__global__ void maxpool2x2_relu_fwd_k(const float* __restrict__ x,
                                      float* __restrict__ y,
                                      unsigned char* __restrict__ idx,
                                      int NC, int H, int W, int OH, int OW) {
  long total = (long)NC * OH * OW;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int ow = i % OW;
    long r = i / OW;
    int oh = r % OH;
    long nc = r / OH;
    const float* p = x + (nc * H + oh * 2) * W + ow * 2;
    float v0 = p[0], v1 = p[1], v2 = p[W], v3 = p[W + 1];
    float m01 = fmaxf(v0, v1), m23 = fmaxf(v2, v3);
    float m = fmaxf(m01, m23);
    unsigned char a;
    if (m == v0) a = 0; else if (m == v1) a = 1; else if (m == v2) a = 2; else a = 3;
    y[i] = fmaxf(m, 0.f);
    idx[i] = a;
  }
}

__global__ void maxpool2x2_relu_bwd_k(const float* __restrict__ dy,
                                      const float* __restrict__ y,
                                      const unsigned char* __restrict__ idx,
                                      float* __restrict__ dx,
                                      int NC, int H, int W, int OH, int OW) {
  long total = (long)NC * OH * OW;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int ow = i % OW;
    long r = i / OW;
    int oh = r % OH;
    long nc = r / OH;
    float* p = dx + (nc * H + oh * 2) * W + ow * 2;
    float g = (y[i] > 0.f) ? dy[i] : 0.f;  // relu gate
    unsigned char a = idx[i];
    // this thread exclusively owns its 2x2 input window: write all 4 slots
    p[0] = a == 0 ? g : 0.f;
    p[1] = a == 1 ? g : 0.f;
    p[W] = a == 2 ? g : 0.f;
    p[W + 1] = a == 3 ? g : 0.f;
  }
}

// ---------------------------------------------------------------------------
// LogSoftmax (+ fused mean-NLL) over dim=1 of [M][C]; C is tiny (10) — one
// thread per row, block-level loss reduction, single atomicAdd per block.
// ---------------------------------------------------------------------------
__global__ void logsoftmax_nll_fwd_k(const float* __restrict__ x,
                                     const long* __restrict__ target,
                                     float* __restrict__ logp,
                                     float* __restrict__ loss,  // [1], zeroed
                                     int M, int C, int with_loss) {
  __shared__ float red[256];
  float part = 0.f;
  for (long m = (long)blockIdx.x * blockDim.x + threadIdx.x; m < M;
       m += (long)gridDim.x * blockDim.x) {
    const float* row = x + m * C;
    float mx = row[0];
    for (int c = 1; c < C; ++c) mx = fmaxf(mx, row[c]);
    float s = 0.f;
    for (int c = 0; c < C; ++c) s += __expf(row[c] - mx);
    float lse = mx + __logf(s);
    float* out = logp + m * C;
    for (int c = 0; c < C; ++c) out[c] = row[c] - lse;
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

// dx = (exp(logp) - onehot(target)) * dloss / M
__global__ void logsoftmax_nll_bwd_k(const float* __restrict__ logp,
                                     const long* __restrict__ target,
                                     const float* __restrict__ dloss,
                                     float* __restrict__ dx, int M, int C) {
  float scale = dloss[0] / (float)M;
  for (long m = (long)blockIdx.x * blockDim.x + threadIdx.x; m < M;
       m += (long)gridDim.x * blockDim.x) {
    const float* lp = logp + m * C;
    float* out = dx + m * C;
    long t = target[m];
    for (int c = 0; c < C; ++c)
      out[c] = (__expf(lp[c]) - (c == t ? 1.f : 0.f)) * scale;
  }
}

// generic logsoftmax backward: dx = dy - exp(logp) * rowsum(dy)
__global__ void logsoftmax_bwd_k(const float* __restrict__ dy,
                                 const float* __restrict__ logp,
                                 float* __restrict__ dx, int M, int C) {
  for (long m = (long)blockIdx.x * blockDim.x + threadIdx.x; m < M;
       m += (long)gridDim.x * blockDim.x) {
    const float* g = dy + m * C;
    const float* lp = logp + m * C;
    float s = 0.f;
    for (int c = 0; c < C; ++c) s += g[c];
    float* out = dx + m * C;
    for (int c = 0; c < C; ++c) out[c] = g[c] - __expf(lp[c]) * s;
  }
}

// ---------------------------------------------------------------------------
// Dropout — counter-based RNG (no state tensors), u8 mask saved for bwd.
// channelwise (Dropout2d): one decision per (n, c), mask shaped [N*C].
// ---------------------------------------------------------------------------
__global__ void bump_u64_f32path_k(unsigned long long* ctr) {
  if (threadIdx.x == 0 && blockIdx.x == 0) *ctr += 1;
}

__global__ void dropout_fwd_k(const float* __restrict__ x, float* __restrict__ y,
                              unsigned char* __restrict__ mask, long n,
                              long hw /* 1 for elementwise */, float p,
                              unsigned long long seed,
                              const unsigned long long* __restrict__ ctr_p) {
  unsigned long long ctr = *ctr_p;
  float inv = 1.f / (1.f - p);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    long unit = i / hw;  // rng unit: element, or (n,c) plane when hw = H*W
    bool keep = rng_uniform(seed, ctr, (unsigned long long)unit) >= p;
    y[i] = keep ? x[i] * inv : 0.f;
    if (i % hw == 0) mask[unit] = keep ? 1 : 0;
  }
}

__global__ void dropout_bwd_k(const float* __restrict__ dy,
                              const unsigned char* __restrict__ mask,
                              float* __restrict__ dx, long n, long hw, float p) {
  float inv = 1.f / (1.f - p);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    dx[i] = mask[i / hw] ? dy[i] * inv : 0.f;
  }
}

// ---------------------------------------------------------------------------
// Fused multi-tensor SGD+momentum: ONE launch for the whole
// parameter set. chunks[b] = (tensor_idx, elem_offset); ptrs[t] =
// (param, grad, momentum, numel). buf = mu*buf + g ; p -= lr*buf.
// The grad is consumed here and nothing reads it again before the next
// backward accumulates into it, so the kernel also resets it to zero — this
// replaces a per-param zero_() launch train (62 fills/step on ResNet-18)
// with stores fused into the update loop.
// ---------------------------------------------------------------------------
__global__ void fused_sgd_k(const long* __restrict__ ptrs,  // [T][4]
                            const int* __restrict__ chunks,  // [NC][2]
                            int chunk_elems, float lr, float momentum) {
  int ti = chunks[blockIdx.x * 2];
  int off = chunks[blockIdx.x * 2 + 1];
  float* p = reinterpret_cast<float*>(ptrs[ti * 4 + 0]);
  float* g = reinterpret_cast<float*>(ptrs[ti * 4 + 1]);
  float* mb = reinterpret_cast<float*>(ptrs[ti * 4 + 2]);
  long n = ptrs[ti * 4 + 3];
  long end = min((long)off + chunk_elems, n);
  for (long i = off + threadIdx.x; i < end; i += blockDim.x) {
    float b = momentum * mb[i] + g[i];
    mb[i] = b;
    g[i] = 0.f;
    p[i] -= lr * b;
  }
}
