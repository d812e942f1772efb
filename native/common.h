// Common device helpers for the dist_mnist_amd HIP/CDNA4 kernel core.
// Target: MI355X (gfx950) ONLY. wave64; MFMA per guides in /opt/skills/guides.
#pragma once
#include <hip/hip_runtime.h>

#define DEV __device__ __forceinline__

using f32x4 = __attribute__((ext_vector_type(4))) float;
using f32x16 = __attribute__((ext_vector_type(16))) float;
using s16x4 = __attribute__((ext_vector_type(4))) short;
using s16x8 = __attribute__((ext_vector_type(8))) short;

#ifndef WAVE
#define WAVE 64
#endif

DEV int lane_id() { return threadIdx.x & (WAVE - 1); }
DEV int wave_id() { return threadIdx.x / WAVE; }

// ceil-div
constexpr int cdiv(int a, int b) { return (a + b - 1) / b; }

// ---------------------------------------------------------------------------
// f32 MFMA wrappers (exact f32 at the 157 TF vector-rate — guide §3):
//   v_mfma_f32_16x16x4_f32: A[row=l&15][k=l>>4], B[k=l>>4][col=l&15],
//   C/D: col = l&15, row = (l>>4)*4 + reg.
// ---------------------------------------------------------------------------
DEV f32x4 mfma16x16x4(float a, float b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, c, 0, 0, 0);
}

// ---------------------------------------------------------------------------
// Hardware-bounds-checked buffer loads (guide T8): an if-guarded gather is
// exec-predicated and UNCOUNTABLE for hipcc's s_waitcnt bookkeeping, which
// forces a vmcnt(0) full drain at every consumer. A raw buffer load needs no
// predication — a dead lane's offset is pushed past num_records and the
// hardware returns 0. Callers gate on rsrc*_fits (u32 num_records limit).
// ---------------------------------------------------------------------------
using buf_rsrc_t = __amdgpu_buffer_rsrc_t;
DEV buf_rsrc_t make_f32_rsrc(const float* p, long n_elems) {
  return __builtin_amdgcn_make_buffer_rsrc((void*)p, (short)0,
                                           (int)(n_elems * 4), 0x00020000);
}
DEV bool rsrc_fits_f32(long n_elems) { return n_elems * 4 < (1LL << 31); }
DEV float buf_loadf(buf_rsrc_t r, long elem_off, bool ok) {
  int boff = ok ? (int)(elem_off * 4) : 0x7ffffff0;
  // bit_cast, NOT assignment: the b32 builtin's return type is integral on
  // this clang, and an implicit int->float VALUE conversion of the loaded
  // bits silently corrupts every gather
  auto t = __builtin_amdgcn_raw_buffer_load_b32(r, boff, 0, 0);
  static_assert(sizeof(t) == 4);
  return __builtin_bit_cast(float, t);
}

// counter-based RNG (splitmix64 finalizer) for dropout masks: pure function
// of (seed, stream, index) — reproducible across world sizes and replays.
DEV unsigned long long mix64(unsigned long long z) {
  z += 0x9E3779B97F4A7C15ull;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  return z ^ (z >> 31);
}

DEV float rng_uniform(unsigned long long seed, unsigned long long stream,
                      unsigned long long idx) {
  unsigned long long h = mix64(seed ^ mix64(stream ^ mix64(idx)));
  // top 24 bits -> [0,1)
  return (float)(h >> 40) * (1.0f / 16777216.0f);
}
