// Torch bindings for the flowml CDNA4 kernels. Pure C++ TU: kernels live in
// the .hip TUs and are reached through extern "C" launchers taking a void*
// hipStream_t, so this file needs only the torch headers.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <tuple>
#include <vector>

// ---- launcher decls (defined in tree_kernels.hip / mlp_kernels.hip) -------

extern "C" {
void launch_bin_codes(const float*, const float*, uint8_t*, long long, int,
                      int, void*);
void launch_tree_hist_cls(const uint8_t*, int*, const int*, const int*,
                          const int*, const uint8_t*, const float*, const int*,
                          float*, const int*, const int*, const int*,
                          const int*, const long long*, int, const float*,
                          unsigned long long, long long, long long, int, int,
                          int, int, int, int, void*);
void launch_tree_hist_reg(const uint8_t*, int*, const int*, const int*,
                          const int*, const float*, const float*, const int*,
                          float*, const int*, const int*, const int*,
                          const int*, const long long*, int, const float*,
                          unsigned long long, long long, long long, int, int,
                          int, int, int, int, void*);
void launch_node_advance(const uint8_t*, const int*, int*, const int*,
                         const int*, const int*, const int*, const long long*,
                         long long, int, int, void*);
void launch_tree_predict(const float*, const void*, const float*,
                         const long long*, float*,
                         long long, int, int, int, int, void*);
void launch_logreg_delta(const float*, const int*, const float*, const float*,
                         const float*, float*, float*, long long, int, int,
                         void*);
void launch_softmax_xent(const float*, const int*, const float*, float*,
                         float*, long long, int, void*);
int softmax_xent_grid(long long);
void launch_split_scan(const float*, float*, int, int, int, int, int, int,
                       double, void*);
void launch_split_scan_reg(const float*, float*, int, int, int, int, double,
                           void*);
void launch_gemm_bias_act_bf16(const void*, const void*, const float*, void*,
                               long long, int, int, int, void*);
void launch_gemm_bias_act_f32(const float*, const float*, const float*, float*,
                              long long, int, int, int, void*);
void launch_gemm_dgrad_act_bf16(const void*, const void*, const void*, void*,
                                long long, int, int, int, void*);
void launch_wgrad_bias_bf16(const void*, const void*, float*, long long, int,
                            int, int, long long, int, void*);
void wgrad_bias_plan(long long, int*, long long*, int*);
void launch_wgrad_f32_splitk(const float*, const float*, float*, long long,
                             int, int, int, long long, void*);
}

static void* cur_stream() {
  return (void*)at::cuda::getCurrentCUDAStream().stream();
}

#define CHECK_GPU(x) TORCH_CHECK(x.is_cuda(), #x " must be a GPU tensor")
#define CHECK_CONTIG(x) TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")
#define CHK(x) \
  CHECK_GPU(x); \
  CHECK_CONTIG(x)

// ---------------------------------------------------------------------------

torch::Tensor bin_codes(torch::Tensor X, torch::Tensor thr) {
  CHK(X);
  CHK(thr);
  TORCH_CHECK(X.scalar_type() == torch::kFloat32, "X must be f32");
  const long long N = X.size(0);
  const int F = X.size(1);
  const int TB = thr.size(1);
  auto out = torch::empty({N, F}, X.options().dtype(torch::kUInt8));
  launch_bin_codes(X.data_ptr<float>(), thr.data_ptr<float>(),
                   out.data_ptr<uint8_t>(), N, F, TB, cur_stream());
  return out;
}

torch::Tensor tree_hist_cls(torch::Tensor codes, torch::Tensor nodeid,
                            torch::Tensor lut, torch::Tensor tree_lo,
                            torch::Tensor tree_cnt, torch::Tensor labels,
                            torch::Tensor bag, int64_t G, int64_t l_max,
                            int64_t B, int64_t C, torch::Tensor fs,
                            torch::Tensor adv_feat, torch::Tensor adv_thr,
                            torch::Tensor adv_left, torch::Tensor adv_right,
                            torch::Tensor adv_cat, int64_t bag_mode,
                            torch::Tensor bag_cdf, int64_t bag_seed,
                            int64_t row_offset) {
  CHK(codes);
  CHK(nodeid);
  CHK(lut);
  CHK(tree_lo);
  CHK(tree_cnt);
  CHK(labels);
  const long long N = codes.size(0);
  const int F = codes.size(1);
  const int T = nodeid.size(0);
  const int Fp = fs.numel() ? fs.size(1) : F;
  const bool adv = adv_feat.numel() > 0;
  auto out = torch::zeros({G, Fp, B, C}, codes.options().dtype(torch::kFloat32));
  launch_tree_hist_cls(
      codes.data_ptr<uint8_t>(), nodeid.data_ptr<int>(), lut.data_ptr<int>(),
      tree_lo.data_ptr<int>(), tree_cnt.data_ptr<int>(),
      labels.data_ptr<uint8_t>(),
      bag.numel() ? bag.data_ptr<float>() : nullptr,
      fs.numel() ? fs.data_ptr<int>() : nullptr, out.data_ptr<float>(),
      adv ? adv_feat.data_ptr<int>() : nullptr,
      adv ? adv_thr.data_ptr<int>() : nullptr,
      adv ? adv_left.data_ptr<int>() : nullptr,
      adv ? adv_right.data_ptr<int>() : nullptr,
      (adv && adv_cat.numel())
          ? reinterpret_cast<const long long*>(adv_cat.data_ptr<int64_t>())
          : nullptr,
      (int)bag_mode,
      bag_cdf.numel() ? bag_cdf.data_ptr<float>() : nullptr,
      (unsigned long long)bag_seed, (long long)row_offset, N, F,
      T, Fp, (int)B, (int)C, (int)l_max, cur_stream());
  return out;
}

torch::Tensor tree_hist_reg(torch::Tensor codes, torch::Tensor nodeid,
                            torch::Tensor lut, torch::Tensor tree_lo,
                            torch::Tensor tree_cnt, torch::Tensor targets,
                            torch::Tensor bag, int64_t G, int64_t l_max,
                            int64_t B, torch::Tensor fs,
                            torch::Tensor adv_feat, torch::Tensor adv_thr,
                            torch::Tensor adv_left, torch::Tensor adv_right,
                            torch::Tensor adv_cat, int64_t bag_mode,
                            torch::Tensor bag_cdf, int64_t bag_seed,
                            int64_t row_offset) {
  CHK(codes);
  CHK(nodeid);
  CHK(lut);
  CHK(targets);
  const long long N = codes.size(0);
  const int F = codes.size(1);
  const int T = nodeid.size(0);
  const int Fp = fs.numel() ? fs.size(1) : F;
  const bool adv = adv_feat.numel() > 0;
  // flat: G*Fp*B*2 cells then G node-s2 slots (ops splits the views)
  auto out = torch::zeros({G * Fp * B * 2 + G},
                          codes.options().dtype(torch::kFloat32));
  launch_tree_hist_reg(
      codes.data_ptr<uint8_t>(), nodeid.data_ptr<int>(), lut.data_ptr<int>(),
      tree_lo.data_ptr<int>(), tree_cnt.data_ptr<int>(),
      targets.data_ptr<float>(),
      bag.numel() ? bag.data_ptr<float>() : nullptr,
      fs.numel() ? fs.data_ptr<int>() : nullptr, out.data_ptr<float>(),
      adv ? adv_feat.data_ptr<int>() : nullptr,
      adv ? adv_thr.data_ptr<int>() : nullptr,
      adv ? adv_left.data_ptr<int>() : nullptr,
      adv ? adv_right.data_ptr<int>() : nullptr,
      (adv && adv_cat.numel())
          ? reinterpret_cast<const long long*>(adv_cat.data_ptr<int64_t>())
          : nullptr,
      (int)bag_mode,
      bag_cdf.numel() ? bag_cdf.data_ptr<float>() : nullptr,
      (unsigned long long)bag_seed, (long long)row_offset, N, F,
      T, Fp, (int)B, (int)G, (int)l_max, cur_stream());
  return out;
}

torch::Tensor split_scan_reg(torch::Tensor hist_flat, int64_t G, int64_t Fp,
                             int64_t B, int64_t min_instances,
                             double min_info_gain) {
  CHK(hist_flat);
  TORCH_CHECK(hist_flat.numel() == G * Fp * B * 2 + G, "reg hist layout");
  auto packed = torch::empty({G, 10}, hist_flat.options());
  launch_split_scan_reg(hist_flat.data_ptr<float>(), packed.data_ptr<float>(),
                        (int)G, (int)Fp, (int)B, (int)min_instances,
                        min_info_gain, cur_stream());
  return packed;
}

torch::Tensor split_scan(torch::Tensor hist, int64_t imp_code,
                         int64_t min_instances, double min_info_gain) {
  CHK(hist);
  const int G = hist.size(0);
  const int Fp = hist.size(1);
  const int B = hist.size(2);
  const int S = hist.size(3);
  TORCH_CHECK(S <= 8, "split_scan supports up to 8 stats");
  auto packed = torch::empty({G, 8 + 3 * S}, hist.options());
  launch_split_scan(hist.data_ptr<float>(), packed.data_ptr<float>(), G, Fp, B,
                    S, (int)imp_code, (int)min_instances, min_info_gain,
                    cur_stream());
  return packed;
}

torch::Tensor node_advance(torch::Tensor codes, torch::Tensor nodeid,
                           torch::Tensor feat, torch::Tensor thr,
                           torch::Tensor left, torch::Tensor right,
                           torch::Tensor catmask) {
  CHK(codes);
  CHK(nodeid);
  TORCH_CHECK(catmask.scalar_type() == torch::kInt64, "catmask must be i64");
  const long long N = codes.size(0);
  const int F = codes.size(1);
  const int T = nodeid.size(0);
  auto out = torch::empty_like(nodeid);
  launch_node_advance(codes.data_ptr<uint8_t>(), nodeid.data_ptr<int>(),
                      out.data_ptr<int>(), feat.data_ptr<int>(),
                      thr.data_ptr<int>(), left.data_ptr<int>(),
                      right.data_ptr<int>(),
                      reinterpret_cast<const long long*>(catmask.data_ptr<int64_t>()),
                      N, F, T, cur_stream());
  return out;
}

torch::Tensor tree_predict(torch::Tensor X, torch::Tensor packed,
                           torch::Tensor value, torch::Tensor catmask) {
  CHK(X);
  CHK(packed);
  CHK(value);
  TORCH_CHECK(packed.scalar_type() == torch::kInt32, "packed must be i32");
  TORCH_CHECK(packed.dim() == 3 && packed.size(2) == 4,
              "packed must be T x M x 4");
  TORCH_CHECK(catmask.scalar_type() == torch::kInt64, "catmask must be i64");
  const long long N = X.size(0);
  const int F = X.size(1);
  const int T = packed.size(0);
  const int M = packed.size(1);
  const int K = value.size(2);
  TORCH_CHECK(K <= 16, "tree_predict supports up to 16 classes");
  auto out = torch::empty({N, K}, X.options().dtype(torch::kFloat32));
  launch_tree_predict(X.data_ptr<float>(), packed.data_ptr<int>(),
                      value.data_ptr<float>(),
                      reinterpret_cast<const long long*>(catmask.data_ptr<int64_t>()),
                      out.data_ptr<float>(), N, F, T, M, K, cur_stream());
  return out;
}

torch::Tensor gemm_dgrad_act(torch::Tensor D, torch::Tensor Wt,
                             torch::Tensor H, int64_t act) {
  CHK(D);
  CHK(Wt);
  CHK(H);
  TORCH_CHECK(D.scalar_type() == torch::kBFloat16, "D must be bf16");
  const long long N = D.size(0);
  const int Kp = D.size(1);
  const int M = Wt.size(0);
  TORCH_CHECK(Wt.size(1) == Kp, "Wt must be M x Kp");
  TORCH_CHECK(Kp % 32 == 0, "Kp must be a multiple of 32");
  TORCH_CHECK(M <= 128, "M <= 128");
  TORCH_CHECK(H.size(0) == N && H.size(1) == M, "H must be N x M");
  auto out = torch::empty({N, M}, D.options());
  launch_gemm_dgrad_act_bf16(D.data_ptr(), Wt.data_ptr(), H.data_ptr(),
                             out.data_ptr(), N, M, Kp, (int)act,
                             cur_stream());
  return out;
}

std::tuple<torch::Tensor, torch::Tensor> logreg_delta(torch::Tensor X,
                                                      torch::Tensor y,
                                                      torch::Tensor w,
                                                      torch::Tensor W,
                                                      torch::Tensor b) {
  CHK(X);
  CHK(y);
  CHK(W);
  CHK(b);
  const long long N = X.size(0);
  const int F = X.size(1);
  const int K = W.size(0);
  TORCH_CHECK(K <= 16, "logreg_delta supports up to 16 classes");
  auto delta = torch::empty({N, K}, X.options());
  auto loss = torch::zeros({1}, X.options());
  launch_logreg_delta(X.data_ptr<float>(), y.data_ptr<int>(),
                      w.numel() ? w.data_ptr<float>() : nullptr,
                      W.data_ptr<float>(), b.data_ptr<float>(),
                      delta.data_ptr<float>(), loss.data_ptr<float>(), N, F, K,
                      cur_stream());
  return {loss.squeeze(0), delta};
}

std::tuple<torch::Tensor, torch::Tensor> softmax_xent_grad(
    torch::Tensor logits, torch::Tensor y, torch::Tensor w) {
  CHK(logits);
  CHK(y);
  const long long N = logits.size(0);
  const int K = logits.size(1);
  TORCH_CHECK(K <= 16, "softmax_xent supports up to 16 classes");
  auto delta = torch::empty_like(logits);
  // per-block loss partials summed in fixed order -> deterministic
  auto loss = torch::zeros({softmax_xent_grid(N)}, logits.options());
  launch_softmax_xent(logits.data_ptr<float>(), y.data_ptr<int>(),
                      w.numel() ? w.data_ptr<float>() : nullptr,
                      delta.data_ptr<float>(), loss.data_ptr<float>(), N, K,
                      cur_stream());
  return {loss.sum(), delta};
}

static torch::Tensor pad_k(torch::Tensor t, int64_t mult) {
  const int64_t K = t.size(1);
  const int64_t Kp = ((K + mult - 1) / mult) * mult;
  if (Kp == K) return t;
  return torch::constant_pad_nd(t, {0, Kp - K});
}

torch::Tensor gemm_bias_act(torch::Tensor X, torch::Tensor W, torch::Tensor b,
                            int64_t act) {
  CHK(X);
  CHK(W);
  CHK(b);
  // X may come PRE-padded to the MFMA K multiple (32 bf16 / 4 f32) so the
  // big operand isn't re-padded on every call; W (small) is padded here.
  const int64_t kmult = X.scalar_type() == torch::kBFloat16 ? 32 : 4;
  const int64_t Kw = W.size(1);
  const int64_t Kw_p = ((Kw + kmult - 1) / kmult) * kmult;
  TORCH_CHECK(X.size(1) == Kw || X.size(1) == Kw_p,
              "K mismatch (X may be pre-padded to the next multiple of ", kmult, ")");
  const long long N = X.size(0);
  const int M = W.size(0);
  TORCH_CHECK(M <= 128, "gemm_bias_act supports M <= 128 (MLP-stack shapes)");
  auto bf = b.to(torch::kFloat32);
  if (X.scalar_type() == torch::kBFloat16) {
    auto Xp = pad_k(X, 32).contiguous();
    auto Wp = pad_k(W, 32).contiguous();
    const int Kp = Xp.size(1);
    auto out = torch::empty({N, M}, X.options());
    launch_gemm_bias_act_bf16(Xp.data_ptr(), Wp.data_ptr(),
                              bf.data_ptr<float>(), out.data_ptr(), N, M, Kp,
                              (int)act, cur_stream());
    return out;
  }
  TORCH_CHECK(X.scalar_type() == torch::kFloat32, "X must be bf16 or f32");
  auto Xp = pad_k(X, 4).contiguous();
  auto Wp = pad_k(W.to(torch::kFloat32), 4).contiguous();
  const int Kp = Xp.size(1);
  auto out = torch::empty({N, M}, X.options());
  launch_gemm_bias_act_f32(Xp.data_ptr<float>(), Wp.data_ptr<float>(),
                           bf.data_ptr<float>(), out.data_ptr<float>(), N, M,
                           Kp, (int)act, cur_stream());
  return out;
}

std::tuple<torch::Tensor, torch::Tensor> wgrad_bias(torch::Tensor delta,
                                                    torch::Tensor act) {
  CHK(delta);
  CHK(act);
  TORCH_CHECK(delta.scalar_type() == torch::kBFloat16 &&
                  act.scalar_type() == torch::kBFloat16,
              "bf16 only");
  const long long N = delta.size(0);
  const int K = delta.size(1);
  const int F = act.size(1);
  int grid = 0, version = 4;
  long long chunk = 0;
  wgrad_bias_plan(N, &grid, &chunk, &version);
  torch::Tensor out;
  if (version >= 4) {
    // exact per-block partial slabs, summed in fixed order -> deterministic
    auto part = torch::zeros({grid, (int64_t)K * (F + 1)},
                             delta.options().dtype(torch::kFloat32));
    launch_wgrad_bias_bf16(delta.data_ptr(), act.data_ptr(),
                           part.data_ptr<float>(), N, K, F, grid, chunk,
                           version, cur_stream());
    out = part.sum(0).reshape({K, F + 1});
  } else {
    out = torch::zeros({K, F + 1}, delta.options().dtype(torch::kFloat32));
    launch_wgrad_bias_bf16(delta.data_ptr(), act.data_ptr(),
                           out.data_ptr<float>(), N, K, F, grid, chunk,
                           version, cur_stream());
  }
  auto gW = out.narrow(1, 0, F).contiguous();
  auto gB = out.select(1, F).contiguous();
  return {gW, gB};
}



// deterministic split-K f32 wgrad: gradW = Dᵀ·X (+bias grad), for the LR
// outer product where rocBLAS picks a no-split-K tile (2 workgroups).
std::tuple<torch::Tensor, torch::Tensor> wgrad_f32(torch::Tensor X,
                                                   torch::Tensor D) {
  CHK(X);
  CHK(D);
  TORCH_CHECK(X.scalar_type() == torch::kFloat32 &&
                  D.scalar_type() == torch::kFloat32,
              "f32 only");
  const long long N = X.size(0);
  const int F = X.size(1);
  const int C = D.size(1);
  TORCH_CHECK(D.size(0) == N, "row mismatch");
  const int FC = C * (F + 1);
  TORCH_CHECK(FC <= 1024, "C*(F+1) must be <= 1024");
  long long chunk = (N + 767) / 768;
  if (chunk < 128) chunk = 128;
  const int grid = (int)((N + chunk - 1) / chunk);
  auto part = torch::empty({grid, FC}, X.options());
  launch_wgrad_f32_splitk(X.data_ptr<float>(), D.data_ptr<float>(),
                          part.data_ptr<float>(), N, F, C, grid, chunk,
                          cur_stream());
  auto full = part.sum(0);  // fixed-order reduction -> deterministic
  auto m = full.reshape({C, F + 1});
  return {m.slice(1, 0, F).contiguous(), m.select(1, F).contiguous()};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("bin_codes", &bin_codes, "K4a quantile binning");
  m.def("tree_hist_cls", &tree_hist_cls, "K2 class histograms (all trees)");
  m.def("tree_hist_reg", &tree_hist_reg, "K2b regression histograms");
  m.def("node_advance", &node_advance, "K4b node-id advance");
  m.def("split_scan", &split_scan, "K3 split-gain scan (packed)");
  m.def("split_scan_reg", &split_scan_reg, "K3b variance scan (2-wide + node s2)");
  m.def("gemm_dgrad_act", &gemm_dgrad_act);
  m.def("tree_predict", &tree_predict, "K8 ensemble inference");
  m.def("logreg_delta", &logreg_delta, "K1 fused logistic delta");
  m.def("wgrad_f32", &wgrad_f32, "split-K f32 wgrad + bias grad (LR)");
  m.def("softmax_xent_grad", &softmax_xent_grad, "K6 softmax/xent grad");
  m.def("gemm_bias_act", &gemm_bias_act, "K5 fused MFMA GEMM+bias+act");
  m.def("wgrad_bias", &wgrad_bias, "K7 split-N wgrad + fused bias grad");
}
