// dist_mnist_amd native extension — host bindings for the CDNA4 kernel core.
// Single TU: includes the kernel files and exposes torch-tensor entry points.
// Build: hipcc --offload-arch=gfx950 (in-tree, see setup.py / __graft_entry__).
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "conv.hip"
#include "linear.hip"
#include "elementwise.hip"
#include "bf16.hip"
#include "resnet.hip"
#include "comm.hip"

namespace {

#define CHECK_F32(t) \
  TORCH_CHECK((t).is_cuda() && (t).is_contiguous() && (t).scalar_type() == at::kFloat, \
              #t " must be contiguous f32 CUDA tensor")
#define CHECK_I64(t) \
  TORCH_CHECK((t).is_cuda() && (t).is_contiguous() && (t).scalar_type() == at::kLong, \
              #t " must be contiguous i64 CUDA tensor")

inline hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

// deterministic fold of wgrad partial slabs: 1 pass for small S, wave-per-
// element for small E (one launch), 2 tiled passes otherwise
inline void launch_reduce_splits(const at::Tensor& ws, at::Tensor& out, long E,
                                 int S) {
  if (S > 64 && E <= 4096) {
    hipLaunchKernelGGL(reduce_splits_wave_k, dim3((int)cdiv(E, 4)), dim3(256),
                       0, cur_stream(), ws.data_ptr<float>(),
                       out.data_ptr<float>(), E, S);
  } else if (S > 64) {
    auto mid = at::empty({16, E}, out.options());
    hipLaunchKernelGGL(reduce_splits_k, dim3((int)cdiv(E, 16), 16), dim3(256),
                       0, cur_stream(), ws.data_ptr<float>(),
                       mid.data_ptr<float>(), E, S);
    hipLaunchKernelGGL(reduce_splits_k, dim3((int)cdiv(E, 16), 1), dim3(256),
                       0, cur_stream(), mid.data_ptr<float>(),
                       out.data_ptr<float>(), E, 16);
  } else {
    hipLaunchKernelGGL(reduce_splits_k, dim3((int)cdiv(E, 16), 1), dim3(256),
                       0, cur_stream(), ws.data_ptr<float>(),
                       out.data_ptr<float>(), E, S);
  }
}

// ---------------------------------------------------------------------------
at::Tensor conv2d_fwd(at::Tensor x, at::Tensor w, at::Tensor bias, long stride,
                      long pad) {
  CHECK_F32(x);
  CHECK_F32(w);
  bool has_bias = bias.numel() > 0;
  if (has_bias) CHECK_F32(bias);
  int N = x.size(0), CI = x.size(1), H = x.size(2), W = x.size(3);
  int CO = w.size(0), KH = w.size(2), KW = w.size(3);
  TORCH_CHECK(w.size(1) == CI, "conv2d channel mismatch");
  int OH = (H + 2 * pad - KH) / stride + 1;
  int OW = (W + 2 * pad - KW) / stride + 1;
  auto y = at::empty({N, CO, OH, OW}, x.options());
  int M = N * OH * OW, K = CI * KH * KW;
  // Whole-image-LDS engine (LeNet-class fp32 shapes): images + the whole
  // weight panel stay LDS-resident, per-k gathers become ds_read_b32
  // (conv_smallimg_f32_k) — replaces the scalar global gathers below.
  {
    int ohow = OH * OW;
    long img_elems = (long)CI * H * W;
    constexpr int IMGM = 4;
    int span = (192 % ohow == 0) ? 192 / ohow : 192 / ohow + 2;
    long lds = IMGM * img_elems * 4 + (long)K * ((CO + 7) / 8 * 8) * 4 +
               2L * K * 4;
    if (stride == 1 && pad == 0 && K <= 512 && CO <= 32 && span <= IMGM &&
        lds <= 100000) {
      int snw = CO > 16 ? 2 : 1;
      dim3 g(cdiv(M, 192), 1);
      auto go = [&](auto hb, auto nwc) {
        hipLaunchKernelGGL((conv_smallimg_f32_k<false, decltype(hb)::value, 3,
                                                decltype(nwc)::value, IMGM>),
                           g, dim3(256), lds, cur_stream(),
                           x.data_ptr<float>(), w.data_ptr<float>(),
                           has_bias ? bias.data_ptr<float>() : nullptr,
                           y.data_ptr<float>(), CI, H, W, CO, KH, KW, OH, OW,
                           (int)pad, M, K, N);
      };
      using T = std::true_type;
      using F = std::false_type;
      using N1 = std::integral_constant<int, 1>;
      using N2 = std::integral_constant<int, 2>;
      if (has_bias) { if (snw == 2) go(T{}, N2{}); else go(T{}, N1{}); }
      else { if (snw == 2) go(F{}, N2{}); else go(F{}, N1{}); }
      return y;
    }
  }
  int nw = (CO > 16) ? 2 : 1;   // multi-co tiles amortize the im2col gather
  dim3 grid(cdiv(M, 128), cdiv(CO, nw * 16));
  bool bounds = pad > 0;
  auto launch = [&](auto hb, auto rl, auto bd) {
    auto go = [&](auto nwc) {
      hipLaunchKernelGGL((conv2d_fwd_f32_k<decltype(hb)::value, decltype(rl)::value,
                                           decltype(bd)::value, decltype(nwc)::value>),
                         grid, dim3(256), 0, cur_stream(), x.data_ptr<float>(),
                         w.data_ptr<float>(),
                         has_bias ? bias.data_ptr<float>() : nullptr,
                         y.data_ptr<float>(), CI, H, W, CO, KH, KW, OH, OW,
                         (int)stride, (int)pad, M, K);
    };
    if (nw == 2) go(std::integral_constant<int, 2>{});
    else go(std::integral_constant<int, 1>{});
  };
  using T = std::true_type;
  using F = std::false_type;
  if (has_bias && bounds) launch(T{}, F{}, T{});
  else if (has_bias) launch(T{}, F{}, F{});
  else if (bounds) launch(F{}, F{}, T{});
  else launch(F{}, F{}, F{});
  return y;
}

at::Tensor conv2d_bwd_input(at::Tensor dy, at::Tensor w, long H, long W,
                            long stride, long pad) {
  CHECK_F32(dy);
  CHECK_F32(w);
  int N = dy.size(0), CO = dy.size(1), OH = dy.size(2), OW = dy.size(3);
  int CI = w.size(1), KH = w.size(2), KW = w.size(3);
  auto dx = at::empty({N, CI, H, W}, dy.options());
  int Mp = N * H * W, Kp = CO * KH * KW;
  // Whole-image-LDS dgrad (LeNet-class): dy images + transposed weight
  // panel LDS-resident (see conv_smallimg_f32_k; 478 us -> the scalar
  // global gathers were the bound)
  {
    int ohow = H * W;
    long img_elems = (long)CO * OH * OW;
    constexpr int IMGM = 4;
    int span = (256 % ohow == 0) ? 256 / ohow : 256 / ohow + 2;
    long lds = IMGM * img_elems * 4 + (long)Kp * ((CI + 7) / 8 * 8) * 4 +
               2L * Kp * 4;
    // MEASURED NEUTRAL-TO-WORSE (516 vs 478 us on LeNet conv2 dgrad,
    // gpurun_out/prof_lenet_fp32_r2h.txt): LDS-resident images do not move
    // the bound (same pattern as the bf16 smallK experiments — the
    // per-MFMA dependent gather chain, not where the gather lands).
    // Env-gated for A/B only.
    // Default: MW=8 whole-image-LDS dgrad for CI<=16 — measured 431 vs the
    // 470 us gather and 510 us MW=4 (gpurun_out/ab_dgrad.txt; the wider
    // per-wave m-tile amortizes the per-k table reads + predicates).
    // DMA_F32_SMALLIMG_DGRAD=0 forces the gather engine for A/B.
    const char* smv = getenv("DMA_F32_SMALLIMG_DGRAD");
    int smw = smv ? atoi(smv) : 8;
    if (stride == 1 && Kp <= 512 && CI <= 32 && span <= IMGM &&
        lds <= 100000 && smw != 0) {
      int snw = CI > 16 ? 2 : 1;
      if (smw == 8 && snw == 1) {
        long lds8 = 5 * img_elems * 4 + (long)Kp * ((CI + 7) / 8 * 8) * 4 +
                    2L * Kp * 4;
        dim3 g(cdiv(Mp, 512), 1);
        hipLaunchKernelGGL((conv_smallimg_f32_k<true, false, 8, 1, 5>),
                           g, dim3(256), lds8, cur_stream(),
                           dy.data_ptr<float>(), w.data_ptr<float>(), nullptr,
                           dx.data_ptr<float>(), CI, (int)H, (int)W, CO, KH,
                           KW, OH, OW, (int)pad, Mp, Kp, N);
        return dx;
      }
      dim3 g(cdiv(Mp, 256), 1);
      auto go = [&](auto nwc) {
        hipLaunchKernelGGL((conv_smallimg_f32_k<true, false, 4,
                                                decltype(nwc)::value, IMGM>),
                           g, dim3(256), lds, cur_stream(),
                           dy.data_ptr<float>(), w.data_ptr<float>(), nullptr,
                           dx.data_ptr<float>(), CI, (int)H, (int)W, CO, KH,
                           KW, OH, OW, (int)pad, Mp, Kp, N);
      };
      if (snw == 2) go(std::integral_constant<int, 2>{});
      else go(std::integral_constant<int, 1>{});
      return dx;
    }
  }
  int mw = (Mp >= 1 << 16) ? 4 : 2;  // amortize LDS weight/kdec reads
  dim3 grid(cdiv(Mp, mw * 64), cdiv(CI, 16));
  auto launch = [&](auto s1c, auto mwc) {
    hipLaunchKernelGGL((conv2d_dgrad_f32_k<decltype(s1c)::value, decltype(mwc)::value>),
                       grid, dim3(256), 0, cur_stream(), dy.data_ptr<float>(),
                       w.data_ptr<float>(), dx.data_ptr<float>(), CI, H, W, CO,
                       KH, KW, OH, OW, (int)stride, (int)pad, Mp, Kp);
  };
  using T = std::true_type;
  using F = std::false_type;
  using M2 = std::integral_constant<int, 2>;
  using M4 = std::integral_constant<int, 4>;
  if (stride == 1) { if (mw == 4) launch(T{}, M4{}); else launch(T{}, M2{}); }
  else { if (mw == 4) launch(F{}, M4{}); else launch(F{}, M2{}); }
  return dx;
}

std::vector<at::Tensor> conv2d_bwd_weight(at::Tensor dy, at::Tensor x, long KH,
                                          long KW, long stride, long pad,
                                          bool want_bias) {
  CHECK_F32(dy);
  CHECK_F32(x);
  int N = x.size(0), CI = x.size(1), H = x.size(2), W = x.size(3);
  int CO = dy.size(1), OH = dy.size(2), OW = dy.size(3);
  auto dw = at::empty({CO, CI, KH, KW}, x.options());
  auto db = at::empty({want_bias ? CO : 0}, x.options());
  int M = N * OH * OW, K = CI * KH * KW;
  bool small_k = K <= 32;
  int ktile = small_k ? 32 : 64;
  int kt = cdiv(K, ktile), ct = cdiv(CO, 16);
  long dw_elems = (long)CO * K;
  int ms_cap = (int)std::max<long>(1, (256L << 20) / (4 * dw_elems));
  int ms = std::max(1, std::min({2048 / std::max(1, kt * ct), cdiv(M, 256), ms_cap}));
  int m_per_split = cdiv(cdiv(M, ms), 256) * 256;
  ms = cdiv(M, m_per_split);
  dim3 grid(kt, ct, ms);
  bool bounds = pad > 0;
  // partial slabs per M-split; folded deterministically below
  auto dw_ws = at::empty({ms, dw_elems}, x.options());
  auto db_ws = at::empty({want_bias ? ms : 0, want_bias ? CO : 0}, x.options());
  auto launch = [&](auto bd, auto wb, auto ktc) {
    hipLaunchKernelGGL((conv2d_wgrad_f32_k<decltype(bd)::value, decltype(wb)::value,
                                           decltype(ktc)::value>),
                       grid, dim3(256), 0, cur_stream(), dy.data_ptr<float>(),
                       x.data_ptr<float>(), dw_ws.data_ptr<float>(),
                       want_bias ? db_ws.data_ptr<float>() : nullptr, CI, H, W, CO,
                       KH, KW, OH, OW, (int)stride, (int)pad, M, K, m_per_split);
  };
  using T = std::true_type;
  using F = std::false_type;
  using K2 = std::integral_constant<int, 2>;
  using K4 = std::integral_constant<int, 4>;
  if (small_k) {
    if (bounds && want_bias) launch(T{}, T{}, K2{});
    else if (bounds) launch(T{}, F{}, K2{});
    else if (want_bias) launch(F{}, T{}, K2{});
    else launch(F{}, F{}, K2{});
  } else {
    if (bounds && want_bias) launch(T{}, T{}, K4{});
    else if (bounds) launch(T{}, F{}, K4{});
    else if (want_bias) launch(F{}, T{}, K4{});
    else launch(F{}, F{}, K4{});
  }
  {
    launch_reduce_splits(dw_ws, dw, dw_elems, ms);
    if (want_bias) launch_reduce_splits(db_ws, db, CO, ms);
  }
  return {dw, db};
}

// ---------------------------------------------------------------------------
std::vector<at::Tensor> conv_pool_stem_f32(at::Tensor x, at::Tensor w,
                                           at::Tensor bias) {
  CHECK_F32(x);
  CHECK_F32(w);
  CHECK_F32(bias);
  int N = x.size(0), CI = x.size(1), H = x.size(2), W = x.size(3);
  int CO = w.size(0), KH = w.size(2), KW = w.size(3);
  TORCH_CHECK(CI == 1 && KH == 5 && KW == 5,
              "f32 stem specialized for CI=1 5x5 (LeNet conv1)");
  int OH = H - KH + 1, OW = W - KW + 1;
  TORCH_CHECK(OH % 2 == 0 && OW % 2 == 0);
  int OH2 = OH / 2, OW2 = OW / 2;
  auto p = at::empty({N, CO, OH2, OW2}, x.options());
  auto idx = at::empty({N, CO, OH2, OW2}, x.options().dtype(at::kByte));
  long total = (long)N * OH2 * OW2;
  int blocks = (int)std::min<long>(4096, cdiv(total, 256));
  size_t lds = (size_t)CO * KH * KW * sizeof(float);
  hipLaunchKernelGGL((conv_pool_stem_f32_k<5, 5>), dim3(blocks), dim3(256),
                     lds, cur_stream(), x.data_ptr<float>(),
                     w.data_ptr<float>(), bias.data_ptr<float>(),
                     p.data_ptr<float>(), idx.data_ptr<unsigned char>(),
                     H, W, CO, OH2, OW2, N);
  return {p, idx};
}

std::vector<at::Tensor> conv2d_wgrad_pool_f32(at::Tensor dpool,
                                              at::Tensor pool_y,
                                              at::Tensor pidx, at::Tensor x,
                                              long KH, long KW) {
  CHECK_F32(dpool);
  CHECK_F32(pool_y);
  CHECK_F32(x);
  int N = x.size(0), H = x.size(2), W = x.size(3);
  int CO = dpool.size(1), OH2 = dpool.size(2), OW2 = dpool.size(3);
  int OH = OH2 * 2, OW = OW2 * 2;
  TORCH_CHECK(x.size(1) == 1 && KW <= 8 && KH * 8 <= 48,
              "f32 wgrad_pool: CI=1, KW<=8");
  auto opts = at::TensorOptions().dtype(at::kFloat).device(x.device());
  long Kcan = KH * KW;
  auto dw = at::empty({CO, 1, KH, KW}, opts);
  auto db = at::empty({CO}, opts);
  int M = N * OH * OW;
  int K = (int)KH * 8;
  constexpr int MC = 128;
  int ct = cdiv(CO, 16);
  // 1024 blocks fill the chip at the kernel's 4-blocks/CU LDS occupancy;
  // each wave writes its own slab so the fold sees ms*4 splits
  int ms = std::max(1, std::min(1024 / ct, cdiv(M, 256)));
  int m_per_split = cdiv(cdiv(M, ms), MC) * MC;
  ms = cdiv(M, m_per_split);
  // each WAVE writes its own slab row -> ms*4 splits for the fold
  auto dw_ws = at::empty({ms * 4, CO * Kcan}, opts);
  auto db_ws = at::empty({ms * 4, CO}, opts);
  hipLaunchKernelGGL((conv2d_wgrad_pool_f32_k<MC>), dim3(1, ct, ms),
                     dim3(256), 0, cur_stream(), dpool.data_ptr<float>(),
                     pool_y.data_ptr<float>(),
                     pidx.data_ptr<unsigned char>(), x.data_ptr<float>(),
                     dw_ws.data_ptr<float>(), db_ws.data_ptr<float>(),
                     H, W, CO, (int)KH, (int)KW, OH, OW, M, K, (int)Kcan,
                     m_per_split, (long)x.numel());
  launch_reduce_splits(dw_ws, dw, CO * Kcan, ms * 4);
  launch_reduce_splits(db_ws, db, CO, ms * 4);
  return {dw, db};
}

// ---------------------------------------------------------------------------
at::Tensor linear_fwd(at::Tensor x, at::Tensor w, at::Tensor bias, bool relu) {
  CHECK_F32(x);
  CHECK_F32(w);
  bool has_bias = bias.numel() > 0;
  if (has_bias) CHECK_F32(bias);
  int M = x.size(0), CI = x.size(1), CO = w.size(0);
  TORCH_CHECK(w.size(1) == CI, "linear shape mismatch");
  auto y = at::empty({M, CO}, x.options());
  // BN=16 when BN=64 would underfill the chip (skinny CO, e.g. fc1)
  int bn = (cdiv(M, 64) * cdiv(CO, 64) < 512 && CO <= 64) ? 16 : 64;
  dim3 grid(cdiv(M, 64), cdiv(CO, bn));
  auto launch = [&](auto hb, auto rl) {
    auto go = [&](auto bnc) {
      hipLaunchKernelGGL((linear_fwd_f32_k<decltype(hb)::value, decltype(rl)::value,
                                           decltype(bnc)::value>),
                         grid, dim3(256), 0, cur_stream(), x.data_ptr<float>(),
                         w.data_ptr<float>(),
                         has_bias ? bias.data_ptr<float>() : nullptr,
                         y.data_ptr<float>(), M, CI, CO);
    };
    if (bn == 16) go(std::integral_constant<int, 16>{});
    else go(std::integral_constant<int, 64>{});
  };
  using T = std::true_type;
  using F = std::false_type;
  if (has_bias && relu) launch(T{}, T{});
  else if (has_bias) launch(T{}, F{});
  else if (relu) launch(F{}, T{});
  else launch(F{}, F{});
  return y;
}

std::vector<at::Tensor> linear_bwd(at::Tensor dy, at::Tensor x, at::Tensor w,
                                   at::Tensor yact, bool relu, bool want_bias) {
  CHECK_F32(dy);
  CHECK_F32(x);
  CHECK_F32(w);
  int M = x.size(0), CI = x.size(1), CO = w.size(0);
  auto dx = at::empty({M, CI}, x.options());
  auto dw = at::empty({CO, CI}, x.options());
  auto db = at::empty({want_bias ? CO : 0}, x.options());
  {
    dim3 grid(cdiv(M, 64), cdiv(CI, 64));
    if (relu)
      hipLaunchKernelGGL((linear_dx_f32_k<true>), grid, dim3(256), 0,
                         cur_stream(), dy.data_ptr<float>(), w.data_ptr<float>(),
                         yact.data_ptr<float>(), dx.data_ptr<float>(), M, CI, CO);
    else
      hipLaunchKernelGGL((linear_dx_f32_k<false>), grid, dim3(256), 0,
                         cur_stream(), dy.data_ptr<float>(), w.data_ptr<float>(),
                         yact.data_ptr<float>(), dx.data_ptr<float>(), M, CI, CO);
  }
  {
    int ct = cdiv(CO, 16), it = cdiv(CI, 64);
    int ms = std::max(1, std::min(2048 / std::max(1, ct * it), cdiv(M, 64)));
    int m_per_split = cdiv(cdiv(M, ms), 4) * 4;
    ms = cdiv(M, m_per_split);
    dim3 grid(ct, it, ms);
    auto dw_ws = at::empty({ms, (long)CO * CI}, x.options());
    auto db_ws = at::empty({want_bias ? ms : 0, want_bias ? CO : 0}, x.options());
    auto launch = [&](auto rl, auto wb) {
      hipLaunchKernelGGL((linear_dw_f32_k<decltype(rl)::value, decltype(wb)::value>),
                         grid, dim3(256), 0, cur_stream(), dy.data_ptr<float>(),
                         x.data_ptr<float>(), yact.data_ptr<float>(),
                         dw_ws.data_ptr<float>(),
                         want_bias ? db_ws.data_ptr<float>() : nullptr, M, CI, CO,
                         m_per_split);
    };
    using T = std::true_type;
    using F = std::false_type;
    if (relu && want_bias) launch(T{}, T{});
    else if (relu) launch(T{}, F{});
    else if (want_bias) launch(F{}, T{});
    else launch(F{}, F{});
    launch_reduce_splits(dw_ws, dw, (long)CO * CI, ms);
    if (want_bias) launch_reduce_splits(db_ws, db, CO, ms);
  }
  return {dx, dw, db};
}

// ---------------------------------------------------------------------------
std::vector<at::Tensor> maxpool2x2_relu_fwd(at::Tensor x) {
  CHECK_F32(x);
  int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  TORCH_CHECK(H % 2 == 0 && W % 2 == 0, "maxpool2x2 needs even H, W");
  int OH = H / 2, OW = W / 2;
  auto y = at::empty({N, C, OH, OW}, x.options());
  auto idx = at::empty({N, C, OH, OW}, x.options().dtype(at::kByte));
  long total = (long)N * C * OH * OW;
  int blocks = (int)std::min<long>(2048, cdiv(total, 256));
  hipLaunchKernelGGL(maxpool2x2_relu_fwd_k, dim3(blocks), dim3(256), 0,
                     cur_stream(), x.data_ptr<float>(), y.data_ptr<float>(),
                     idx.data_ptr<unsigned char>(), N * C, H, W, OH, OW);
  return {y, idx};
}

at::Tensor maxpool2x2_relu_bwd(at::Tensor dy, at::Tensor y, at::Tensor idx,
                               long H, long W) {
  CHECK_F32(dy);
  CHECK_F32(y);
  int N = dy.size(0), C = dy.size(1), OH = dy.size(2), OW = dy.size(3);
  auto dx = at::empty({N, C, H, W}, dy.options());
  long total = (long)N * C * OH * OW;
  int blocks = (int)std::min<long>(2048, cdiv(total, 256));
  hipLaunchKernelGGL(maxpool2x2_relu_bwd_k, dim3(blocks), dim3(256), 0,
                     cur_stream(), dy.data_ptr<float>(), y.data_ptr<float>(),
                     idx.data_ptr<unsigned char>(), dx.data_ptr<float>(), N * C,
                     (int)H, (int)W, OH, OW);
  return dx;
}

// ---------------------------------------------------------------------------
at::Tensor relu_fwd(at::Tensor x) {
  CHECK_F32(x);
  auto y = at::empty_like(x);
  long n = x.numel();
  int blocks = (int)std::min<long>(2048, cdiv(std::max<long>(n / 4, 1), 256));
  hipLaunchKernelGGL(relu_fwd_k, dim3(blocks), dim3(256), 0, cur_stream(),
                     x.data_ptr<float>(), y.data_ptr<float>(), n);
  return y;
}

at::Tensor relu_bwd(at::Tensor dy, at::Tensor y) {
  CHECK_F32(dy);
  CHECK_F32(y);
  auto dx = at::empty_like(dy);
  long n = dy.numel();
  int blocks = (int)std::min<long>(2048, cdiv(std::max<long>(n / 4, 1), 256));
  hipLaunchKernelGGL(relu_bwd_k, dim3(blocks), dim3(256), 0, cur_stream(),
                     dy.data_ptr<float>(), y.data_ptr<float>(),
                     dx.data_ptr<float>(), n);
  return dx;
}

// ---------------------------------------------------------------------------
std::vector<at::Tensor> logsoftmax_nll_fwd(at::Tensor x, at::Tensor target) {
  CHECK_F32(x);
  CHECK_I64(target);
  int M = x.size(0), C = x.size(1);
  auto logp = at::empty_like(x);
  auto loss = at::zeros({}, x.options());
  int blocks = (int)std::min<long>(1024, cdiv(M, 256));
  hipLaunchKernelGGL(logsoftmax_nll_fwd_k, dim3(blocks), dim3(256), 0,
                     cur_stream(), x.data_ptr<float>(), target.data_ptr<long>(),
                     logp.data_ptr<float>(), loss.data_ptr<float>(), M, C, 1);
  return {logp, loss};
}

at::Tensor logsoftmax_nll_bwd(at::Tensor logp, at::Tensor target,
                              at::Tensor dloss) {
  CHECK_F32(logp);
  CHECK_I64(target);
  CHECK_F32(dloss);
  int M = logp.size(0), C = logp.size(1);
  auto dx = at::empty_like(logp);
  int blocks = (int)std::min<long>(1024, cdiv(M, 256));
  hipLaunchKernelGGL(logsoftmax_nll_bwd_k, dim3(blocks), dim3(256), 0,
                     cur_stream(), logp.data_ptr<float>(),
                     target.data_ptr<long>(), dloss.data_ptr<float>(),
                     dx.data_ptr<float>(), M, C);
  return dx;
}

at::Tensor logsoftmax_fwd(at::Tensor x) {
  CHECK_F32(x);
  int M = x.size(0), C = x.size(1);
  auto logp = at::empty_like(x);
  int blocks = (int)std::min<long>(1024, cdiv(M, 256));
  hipLaunchKernelGGL(logsoftmax_nll_fwd_k, dim3(blocks), dim3(256), 0,
                     cur_stream(), x.data_ptr<float>(), nullptr,
                     logp.data_ptr<float>(), nullptr, M, C, 0);
  return logp;
}

at::Tensor logsoftmax_bwd(at::Tensor dy, at::Tensor logp) {
  CHECK_F32(dy);
  CHECK_F32(logp);
  int M = dy.size(0), C = dy.size(1);
  auto dx = at::empty_like(dy);
  int blocks = (int)std::min<long>(1024, cdiv(M, 256));
  hipLaunchKernelGGL(logsoftmax_bwd_k, dim3(blocks), dim3(256), 0, cur_stream(),
                     dy.data_ptr<float>(), logp.data_ptr<float>(),
                     dx.data_ptr<float>(), M, C);
  return dx;
}

// ---------------------------------------------------------------------------
std::vector<at::Tensor> dropout_fwd(at::Tensor x, double p, bool channelwise,
                                    long seed, at::Tensor ctr) {
  CHECK_I64(ctr);
  CHECK_F32(x);
  long n = x.numel();
  long hw = 1;
  long mask_n = n;
  if (channelwise) {
    TORCH_CHECK(x.dim() == 4, "channelwise dropout needs NCHW");
    hw = (long)x.size(2) * x.size(3);
    mask_n = (long)x.size(0) * x.size(1);
  }
  auto y = at::empty_like(x);
  auto mask = at::empty({mask_n}, x.options().dtype(at::kByte));
  int blocks = (int)std::min<long>(2048, cdiv(n, 256));
  hipLaunchKernelGGL(bump_u64_f32path_k, dim3(1), dim3(64), 0, cur_stream(),
                     reinterpret_cast<unsigned long long*>(ctr.data_ptr<long>()));
  hipLaunchKernelGGL(dropout_fwd_k, dim3(blocks), dim3(256), 0, cur_stream(),
                     x.data_ptr<float>(), y.data_ptr<float>(),
                     mask.data_ptr<unsigned char>(), n, hw, (float)p,
                     (unsigned long long)seed,
                     reinterpret_cast<const unsigned long long*>(ctr.data_ptr<long>()));
  return {y, mask};
}

at::Tensor dropout_bwd(at::Tensor dy, at::Tensor mask, double p,
                       bool channelwise) {
  CHECK_F32(dy);
  long n = dy.numel();
  long hw = channelwise ? (long)dy.size(2) * dy.size(3) : 1;
  auto dx = at::empty_like(dy);
  int blocks = (int)std::min<long>(2048, cdiv(n, 256));
  hipLaunchKernelGGL(dropout_bwd_k, dim3(blocks), dim3(256), 0, cur_stream(),
                     dy.data_ptr<float>(), mask.data_ptr<unsigned char>(),
                     dx.data_ptr<float>(), n, hw, (float)p);
  return dx;
}

// ---------------------------------------------------------------------------
void fused_sgd_step(at::Tensor ptrs, at::Tensor chunks, long chunk_elems,
                    double lr, double momentum) {
  CHECK_I64(ptrs);
  TORCH_CHECK(chunks.is_cuda() && chunks.scalar_type() == at::kInt);
  int nc = chunks.size(0);
  hipLaunchKernelGGL(fused_sgd_k, dim3(nc), dim3(256), 0, cur_stream(),
                     ptrs.data_ptr<long>(), chunks.data_ptr<int>(),
                     (int)chunk_elems, (float)lr, (float)momentum);
}

#include "ext_pack.inc"
#include "ext_bf16.inc"
#include "ext_resnet.inc"
#include "ext_comm.inc"

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("conv2d_fwd", &conv2d_fwd);
  m.def("conv2d_bwd_input", &conv2d_bwd_input);
  m.def("conv2d_bwd_weight", &conv2d_bwd_weight);
  m.def("linear_fwd", &linear_fwd);
  m.def("linear_bwd", &linear_bwd);
  m.def("maxpool2x2_relu_fwd", &maxpool2x2_relu_fwd);
  m.def("maxpool2x2_relu_bwd", &maxpool2x2_relu_bwd);
  m.def("relu_fwd", &relu_fwd);
  m.def("relu_bwd", &relu_bwd);
  m.def("logsoftmax_nll_fwd", &logsoftmax_nll_fwd);
  m.def("logsoftmax_nll_bwd", &logsoftmax_nll_bwd);
  m.def("logsoftmax_fwd", &logsoftmax_fwd);
  m.def("logsoftmax_bwd", &logsoftmax_bwd);
  m.def("dropout_fwd", &dropout_fwd);
  m.def("dropout_bwd", &dropout_bwd);
  m.def("fused_sgd_step", &fused_sgd_step);
  m.def("conv_pool_stem_f32", &conv_pool_stem_f32);
  m.def("conv2d_wgrad_pool_f32", &conv2d_wgrad_pool_f32);
  // bf16 NHWC engine
  m.def("mfma_probe", &mfma_probe);
  m.def("tr16_probe", &tr16_probe);
  m.def("conv2d_fwd_bf16", &conv2d_fwd_bf16);
  m.def("repack_refresh_all", &repack_refresh_all);
  m.def("repack_cache_size", &repack_cache_size);
  m.def("repack_cache_clear", &repack_cache_clear);
  m.def("conv_tiled_s1_bf16", &conv_tiled_s1_bf16);
  m.def("conv2d_dgrad_bf16", &conv2d_dgrad_bf16);
  m.def("conv2d_wgrad_bf16", &conv2d_wgrad_bf16);
  m.def("conv_pool_stem_bf16", &conv_pool_stem_bf16);
  m.def("conv2d_wgrad_pool_bf16", &conv2d_wgrad_pool_bf16);
  m.def("linear_fwd_bf16", &linear_fwd_bf16);
  m.def("linear_dx_bf16", &linear_dx_bf16);
  m.def("linear_dw_bf16", &linear_dw_bf16);
  m.def("maxpool2x2_relu_bf16_fwd", &maxpool2x2_relu_bf16_fwd);
  m.def("maxpool2x2_relu_bf16_bwd", &maxpool2x2_relu_bf16_bwd);
  m.def("dropout_bf16_fwd", &dropout_bf16_fwd);
  m.def("dropout_bf16_bwd", &dropout_bf16_bwd);
  m.def("comm_unique_id", &comm_unique_id);
  m.def("comm_id_size", &comm_id_size);
  m.def("comm_init", &comm_init);
  m.def("comm_set_timeout", &comm_set_timeout);
  m.def("comm_aborted", &comm_aborted);
  m.def("comm_all_reduce_sum", &comm_all_reduce_sum);
  m.def("comm_abort", &comm_abort);
  m.def("comm_destroy", &comm_destroy);
  m.def("comm_rank", &comm_rank);
  m.def("comm_world", &comm_world);
  m.def("bn_fwd_bf16", &bn_fwd_bf16);
  m.def("bn_bwd_bf16", &bn_bwd_bf16);
  m.def("bn_stats_bf16", &bn_stats_bf16);
  m.def("bn_bwd_fused_bf16", &bn_bwd_fused_bf16);
  m.def("add_aff_relu_bf16", &add_aff_relu_bf16);
  m.def("conv_tiled_s1_aff_bf16", &conv_tiled_s1_aff_bf16);
  m.def("maxpool3x3s2_fwd_bf16", &maxpool3x3s2_fwd_bf16);
  m.def("maxpool3x3s2_bwd_bf16", &maxpool3x3s2_bwd_bf16);
  m.def("gap_fwd_bf16", &gap_fwd_bf16);
  m.def("gap_bwd_bf16", &gap_bwd_bf16);
  m.def("add_bf16", &add_bf16);
  m.def("relu_gate_bf16", &relu_gate_bf16);
  m.def("relu_bf16", &relu_bf16);
  m.def("logsoftmax_nll_bf16_fwd", &logsoftmax_nll_bf16_fwd);
  m.def("logsoftmax_nll_bf16_bwd", &logsoftmax_nll_bf16_bwd);
}
