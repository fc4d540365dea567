// Torch extension binding for the HIP kernel library (csrc/kernels).
// Host-side only: validates layouts/dtypes, allocates outputs (torch caching
// allocator), forwards raw pointers + the current HIP stream to the C ABI.
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <c10/util/Optional.h>
#include <map>
#include <tuple>

#include "kernels/api.h"

namespace {

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

int dt_code(const at::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kFloat: return 0;
    case at::kBFloat16: return 1;
    case at::kHalf: return 2;
    default: TORCH_CHECK(false, "unsupported dtype ", t.scalar_type());
  }
}

at::Tensor empty_nhwc(int64_t n, int64_t c, int64_t h, int64_t w,
                      const at::TensorOptions& opts) {
  return at::empty({n, c, h, w}, opts.memory_format(at::MemoryFormat::ChannelsLast));
}

void check_cl(const at::Tensor& t, const char* name) {
  // Strict: a plain NCHW-contiguous activation would silently read as
  // garbage NHWC.  The ambiguous cases (C==1 or 1x1 spatial) are the only
  // ones where NCHW- and NHWC-contiguity coincide.
  TORCH_CHECK(t.is_contiguous(at::MemoryFormat::ChannelsLast) ||
                  (t.dim() == 4 &&
                   (t.size(1) == 1 || (t.size(2) == 1 && t.size(3) == 1))),
              name, " must be channels_last contiguous");
}

struct Conv {
  int N, H, W, C, K, R, S, P, Q;
};

Conv conv_geom(const at::Tensor& x, const at::Tensor& w, int sh, int sw,
               int ph, int pw) {
  Conv c;
  c.N = x.size(0); c.C = x.size(1); c.H = x.size(2); c.W = x.size(3);
  c.K = w.size(0); c.R = w.size(2); c.S = w.size(3);
  TORCH_CHECK(w.size(1) == c.C, "conv channel mismatch");
  c.P = (c.H + 2 * ph - c.R) / sh + 1;
  c.Q = (c.W + 2 * pw - c.S) / sw + 1;
  return c;
}

// ---- conv ------------------------------------------------------------------

// For C % 8 != 0 (the 7x7 C=3 stem) materialise a padded im2col buffer and
// run the tiled MFMA path as a plain GEMM — 2-3x the flat scalar-gather
// kernel's speed for ~1 GB of extra (stream-rate) traffic.
static at::Tensor& im2col_table(int R, int S, int C, int ld,
                                const at::Device& dev) {
  static std::map<std::tuple<int, int, int, int>, at::Tensor> cache;
  auto key = std::make_tuple(R, S, C, ld);
  auto it = cache.find(key);
  if (it == cache.end()) {
    auto t = at::empty({ld}, at::TensorOptions().dtype(at::kInt));
    int* p = t.data_ptr<int>();
    for (int k = 0; k < ld; ++k) {
      int r = k / (S * C), rem = k % (S * C);
      p[k] = (r << 20) | ((rem / C) << 10) | (rem % C);
    }
    it = cache.emplace(key, t.to(dev)).first;
  }
  return it->second;
}

static at::Tensor build_im2col(const at::Tensor& x, const Conv& g, int sh,
                               int sw, int ph, int pw, int ld) {
  auto col = at::empty({(int64_t)g.N * g.P * g.Q, ld}, x.options());
  auto& tbl = im2col_table(g.R, g.S, g.C, ld, x.device());
  ddl_im2col(x.data_ptr(), col.data_ptr(), tbl.data_ptr<int>(), g.N, g.H,
             g.W, g.C, g.R, g.S, g.P, g.Q, sh, sw, ph, pw, ld, dt_code(x),
             cur_stream());
  return col;
}

at::Tensor conv_fwd(at::Tensor x, at::Tensor w,
                    c10::optional<at::Tensor> bias, int64_t sh, int64_t sw,
                    int64_t ph, int64_t pw, bool relu) {
  check_cl(x, "x"); check_cl(w, "w");
  TORCH_CHECK(x.scalar_type() == w.scalar_type());
  Conv g = conv_geom(x, w, sh, sw, ph, pw);
  auto y = empty_nhwc(g.N, g.K, g.P, g.Q, x.options());
  const void* bptr = nullptr;
  if (bias.has_value() && bias->defined()) {
    TORCH_CHECK(bias->scalar_type() == x.scalar_type());
    bptr = bias->data_ptr();
  }
  int dt = dt_code(x);
  const int epu = dt == 0 ? 4 : 8;
  if (g.C % epu != 0) {
    const int kred = g.R * g.S * g.C;
    const int ld = (kred + 7) & ~7;
    auto col = build_im2col(x, g, sh, sw, ph, pw, ld);
    auto wp = at::empty({g.K, ld}, w.options());
    ddl_pad_cols(w.data_ptr(), wp.data_ptr(), g.K, kred, ld, dt, cur_stream());
    ddl_conv_fwd(col.data_ptr(), wp.data_ptr(), bptr, y.data_ptr(), g.N, g.P,
                 g.Q, ld, g.K, 1, 1, g.P, g.Q, 1, 1, 0, 0, relu, dt,
                 cur_stream());
    return y;
  }
  ddl_conv_fwd(x.data_ptr(), w.data_ptr(), bptr, y.data_ptr(), g.N, g.H, g.W,
               g.C, g.K, g.R, g.S, g.P, g.Q, sh, sw, ph, pw, relu, dt,
               cur_stream());
  return y;
}

// conv forward that also emits BN {sum,sumsq} partials from the MFMA
// epilogue ([cdiv(M,128)][2][K] fp32) so the following BatchNorm skips its
// full re-read of y.  Returns (y, partials); partials is EMPTY on the
// paths that do not support fusion (fp32, C % 8 != 0) — caller falls back
// to the standalone bn_stats pass.
std::tuple<at::Tensor, at::Tensor> conv_fwd_stats(
    at::Tensor x, at::Tensor w, c10::optional<at::Tensor> bias, int64_t sh,
    int64_t sw, int64_t ph, int64_t pw, bool relu) {
  check_cl(x, "x"); check_cl(w, "w");
  TORCH_CHECK(x.scalar_type() == w.scalar_type());
  Conv g = conv_geom(x, w, sh, sw, ph, pw);
  auto y = empty_nhwc(g.N, g.K, g.P, g.Q, x.options());
  const void* bptr = nullptr;
  if (bias.has_value() && bias->defined()) bptr = bias->data_ptr();
  int dt = dt_code(x);
  const int epu = dt == 0 ? 4 : 8;
  auto f32 = x.options().dtype(at::kFloat);
  if (g.C % epu != 0 || dt == 0) {
    auto yo = conv_fwd(x, w, bias, sh, sw, ph, pw, relu);
    return {yo, at::empty({0}, f32)};
  }
  const int64_t chunks = ddl_conv_fwd_stats_chunks(g.N, g.P, g.Q, g.C, g.K,
                                                   g.R, g.S, dt);
  auto partials = at::empty({chunks * 2 * g.K}, f32);
  ddl_conv_fwd_stats(x.data_ptr(), w.data_ptr(), bptr, y.data_ptr(),
                     partials.data_ptr<float>(), g.N, g.H, g.W, g.C, g.K,
                     g.R, g.S, g.P, g.Q, sh, sw, ph, pw, relu, dt,
                     cur_stream());
  return {y, partials};
}

// small-C (stem) variant returning the im2col buffer so backward reuses
// it, plus the BN stats partials from the (flat-GEMM) epilogue — the stem
// feeds bn1 like every other conv (empty when unsupported: fp32).
std::tuple<at::Tensor, at::Tensor, at::Tensor> conv_fwd_smallc(
    at::Tensor x, at::Tensor w, c10::optional<at::Tensor> bias, int64_t sh,
    int64_t sw, int64_t ph, int64_t pw, bool relu) {
  check_cl(x, "x"); check_cl(w, "w");
  Conv g = conv_geom(x, w, sh, sw, ph, pw);
  int dt = dt_code(x);
  const int kred = g.R * g.S * g.C;
  const int ld = (kred + 7) & ~7;
  auto col = build_im2col(x, g, sh, sw, ph, pw, ld);
  auto wp = at::empty({g.K, ld}, w.options());
  ddl_pad_cols(w.data_ptr(), wp.data_ptr(), g.K, kred, ld, dt, cur_stream());
  auto y = empty_nhwc(g.N, g.K, g.P, g.Q, x.options());
  const void* bptr = (bias.has_value() && bias->defined())
                         ? bias->data_ptr() : nullptr;
  auto f32 = x.options().dtype(at::kFloat);
  at::Tensor partials;
  float* pptr = nullptr;
  if (dt != 0) {
    const int64_t chunks = ddl_conv_fwd_stats_chunks(g.N, g.P, g.Q, ld, g.K,
                                                     1, 1, dt);
    partials = at::empty({chunks * 2 * g.K}, f32);
    pptr = partials.data_ptr<float>();
  } else {
    partials = at::empty({0}, f32);
  }
  ddl_conv_fwd_stats(col.data_ptr(), wp.data_ptr(), bptr, y.data_ptr(), pptr,
                     g.N, g.P, g.Q, ld, g.K, 1, 1, g.P, g.Q, 1, 1, 0, 0,
                     relu, dt, cur_stream());
  return {y, col, partials};
}

// stem dgrad (C % 8 != 0): dcol = dy @ wp (library GEMM, K8-class) then a
// col2im scatter-add back to dx.  Completes K2 generality for the 7x7 C=3
// stem (input grads are only needed when training on learned inputs).
at::Tensor conv_dgrad_smallc(at::Tensor dy, at::Tensor w,
                             std::vector<int64_t> x_shape, int64_t sh,
                             int64_t sw, int64_t ph, int64_t pw) {
  check_cl(dy, "dy"); check_cl(w, "w");
  int N = x_shape[0], C = x_shape[1], H = x_shape[2], W = x_shape[3];
  int K = w.size(0), R = w.size(2), S = w.size(3);
  int P = dy.size(2), Q = dy.size(3);
  const int dt = dt_code(dy);
  const int kred = R * S * C;
  const int ld = (kred + 7) & ~7;
  auto wp = at::empty({K, ld}, w.options());
  ddl_pad_cols(w.data_ptr(), wp.data_ptr(), K, kred, ld, dt, cur_stream());
  // dy rows are [N*P*Q][K] in channels_last memory
  auto dy2d = dy.permute({0, 2, 3, 1}).reshape({(int64_t)N * P * Q, K});
  auto dcol = at::matmul(dy2d, wp);  // [M][ld]
  auto ws = at::zeros({(int64_t)N * H * W * C},
                      dy.options().dtype(at::kFloat));
  auto& tbl = im2col_table(R, S, C, ld, dy.device());
  ddl_col2im_acc(dcol.data_ptr(), ws.data_ptr<float>(), tbl.data_ptr<int>(),
                 N, H, W, C, P, Q, sh, sw, ph, pw, kred, ld, dt,
                 cur_stream());
  auto dx = empty_nhwc(N, C, H, W, dy.options());
  ddl_cast_f32_to(ws.data_ptr<float>(), dx.data_ptr(),
                  (int64_t)N * H * W * C, dt, cur_stream());
  return dx;
}

at::Tensor conv_wgrad_smallc(at::Tensor dy, at::Tensor col,
                             std::vector<int64_t> w_shape,
                             std::vector<int64_t> x_shape) {
  check_cl(dy, "dy");
  int N = x_shape[0], C = x_shape[1];
  int K = w_shape[0], R = w_shape[2], S = w_shape[3];
  int P = dy.size(2), Q = dy.size(3);
  int dt = dt_code(dy);
  const int kred = R * S * C;
  const int ld = (int)col.size(1);
  auto ws = at::empty({ddl_conv_wgrad_ws_elems(N, ld, K, 1, 1, P, Q)},
                      dy.options().dtype(at::kFloat));
  auto dwp = at::empty({K, ld}, dy.options());
  ddl_conv_wgrad(dy.data_ptr(), col.data_ptr(), ws.data_ptr<float>(),
                 dwp.data_ptr(), N, P, Q, ld, K, 1, 1, P, Q, 1, 1, 0, 0,
                 dt, cur_stream());
  auto dw = empty_nhwc(K, C, R, S, dy.options());
  ddl_narrow_cols(dwp.data_ptr(), dw.data_ptr(), K, kred, ld, dt,
                  cur_stream());
  return dw;
}

at::Tensor conv_dgrad(at::Tensor dy, at::Tensor w,
                      std::vector<int64_t> x_shape, int64_t sh, int64_t sw,
                      int64_t ph, int64_t pw,
                      c10::optional<at::Tensor> accum = c10::nullopt) {
  check_cl(dy, "dy"); check_cl(w, "w");
  int N = x_shape[0], C = x_shape[1], H = x_shape[2], W = x_shape[3];
  int K = w.size(0), R = w.size(2), S = w.size(3);
  int P = dy.size(2), Q = dy.size(3);
  TORCH_CHECK(C % (dy.scalar_type() == at::kFloat ? 4 : 8) == 0,
              "dgrad requires channel alignment (skip first conv)");
  // build CRSK-transposed weight for contiguous dgrad B-tiles
  auto w_t = at::empty({(int64_t)C * R * S * K}, w.options());
  ddl_transpose_krsc_crsk(w.data_ptr(), w_t.data_ptr(), K, R * S, C,
                          dt_code(w), cur_stream());
  auto dx = empty_nhwc(N, C, H, W, dy.options());
  const void* aptr = nullptr;
  if (accum.has_value() && accum->defined()) {
    check_cl(*accum, "accum");
    TORCH_CHECK(accum->scalar_type() == dy.scalar_type() &&
                    accum->numel() == dx.numel(),
                "dgrad accum shape/dtype mismatch");
    aptr = accum->data_ptr();
  }
  ddl_conv_dgrad_accum(dy.data_ptr(), w_t.data_ptr(), dx.data_ptr(), aptr,
                       N, H, W, C, K, R, S, P, Q, sh, sw, ph, pw,
                       dt_code(dy), cur_stream());
  return dx;
}

// dgrad that also emits the PRECEDING BatchNorm's backward stats partials
// from its epilogue (the bwd analog of conv_fwd_stats): returns
// (dx, partials[chunks*2*C]).  2-byte dtypes, channel-aligned only —
// caller checks via the same gate as conv_dgrad.
std::tuple<at::Tensor, at::Tensor> conv_dgrad_bnstats(
    at::Tensor dy, at::Tensor w, std::vector<int64_t> x_shape, int64_t sh,
    int64_t sw, int64_t ph, int64_t pw, at::Tensor bnx, at::Tensor bn_mean,
    at::Tensor bn_rstd, at::Tensor bn_w, at::Tensor bn_b,
    c10::optional<at::Tensor> accum, c10::optional<at::Tensor> bn_mask) {
  check_cl(dy, "dy"); check_cl(w, "w"); check_cl(bnx, "bnx");
  int N = x_shape[0], C = x_shape[1], H = x_shape[2], W = x_shape[3];
  int K = w.size(0), R = w.size(2), S = w.size(3);
  int P = dy.size(2), Q = dy.size(3);
  const int dt = dt_code(dy);
  TORCH_CHECK(dt != 0 && C % 8 == 0,
              "dgrad+bnstats fusion needs 2-byte dtype and C % 8 == 0");
  TORCH_CHECK(bnx.sizes() == at::IntArrayRef(x_shape) &&
                  bnx.scalar_type() == dy.scalar_type(),
              "bnx must match the conv input shape/dtype");
  TORCH_CHECK(bn_mean.scalar_type() == at::kFloat &&
              bn_rstd.scalar_type() == at::kFloat);
  TORCH_CHECK(bn_w.scalar_type() == dy.scalar_type() &&
              bn_b.scalar_type() == dy.scalar_type());
  const void* aptr = nullptr;
  if (accum.has_value() && accum->defined()) {
    check_cl(*accum, "accum");
    TORCH_CHECK(accum->scalar_type() == dy.scalar_type() &&
                    accum->numel() == (int64_t)N * C * H * W,
                "bnstats accum shape/dtype mismatch");
    aptr = accum->data_ptr();
  }
  const uint8_t* mptr = nullptr;
  if (bn_mask.has_value() && bn_mask->defined() && bn_mask->numel() > 0) {
    TORCH_CHECK(bn_mask->numel() == (int64_t)N * H * W * (C / 8),
                "bnstats mask size mismatch");
    mptr = bn_mask->data_ptr<uint8_t>();
  }
  auto w_t = at::empty({(int64_t)C * R * S * K}, w.options());
  ddl_transpose_krsc_crsk(w.data_ptr(), w_t.data_ptr(), K, R * S, C,
                          dt_code(w), cur_stream());
  auto dx = empty_nhwc(N, C, H, W, dy.options());
  const int64_t chunks = ddl_conv_dgrad_bnstats_chunks(N, H, W, C, K, sh, sw);
  auto partials = at::empty({chunks * 2 * C}, dy.options().dtype(at::kFloat));
  ddl_conv_dgrad_bnstats(dy.data_ptr(), w_t.data_ptr(), dx.data_ptr(),
                         bnx.data_ptr(), bn_mean.data_ptr<float>(),
                         bn_rstd.data_ptr<float>(), bn_w.data_ptr(),
                         bn_b.data_ptr(), partials.data_ptr<float>(), aptr,
                         mptr, N, H, W, C, K, R, S, P, Q, sh, sw, ph, pw,
                         dt, cur_stream());
  return {dx, partials};
}

// out (optional): write dw into this channels_last tensor (the flat-bucket
// grad view — eliminates the AccumulateGrad add) instead of allocating.
at::Tensor conv_wgrad(at::Tensor dy, at::Tensor x,
                      std::vector<int64_t> w_shape, int64_t sh, int64_t sw,
                      int64_t ph, int64_t pw,
                      c10::optional<at::Tensor> out = c10::nullopt) {
  check_cl(dy, "dy"); check_cl(x, "x");
  int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  int K = w_shape[0], R = w_shape[2], S = w_shape[3];
  int P = dy.size(2), Q = dy.size(3);
  int dt = dt_code(x);
  at::Tensor dw;
  if (out.has_value() && out->defined()) {
    check_cl(*out, "out");
    TORCH_CHECK(out->sizes() == at::IntArrayRef(w_shape) &&
                    out->scalar_type() == x.scalar_type(),
                "wgrad out tensor shape/dtype mismatch");
    dw = *out;
  } else {
    dw = empty_nhwc(K, C, R, S, x.options());
  }
  if (C % (dt == 0 ? 4 : 8) != 0) {
    const int kred = R * S * C;
    const int ld = (kred + 7) & ~7;
    Conv g; g.N = N; g.H = H; g.W = W; g.C = C; g.K = K; g.R = R; g.S = S;
    g.P = P; g.Q = Q;
    auto col = build_im2col(x, g, sh, sw, ph, pw, ld);
    auto ws = at::empty({ddl_conv_wgrad_ws_elems(N, ld, K, 1, 1, P, Q)},
                        x.options().dtype(at::kFloat));
    auto dwp = at::empty({K, ld}, x.options());
    ddl_conv_wgrad(dy.data_ptr(), col.data_ptr(), ws.data_ptr<float>(),
                   dwp.data_ptr(), N, P, Q, ld, K, 1, 1, P, Q, 1, 1, 0, 0,
                   dt, cur_stream());
    ddl_narrow_cols(dwp.data_ptr(), dw.data_ptr(), K, kred, ld, dt,
                    cur_stream());
    return dw;
  }
  auto ws = at::empty({ddl_conv_wgrad_ws_elems(N, C, K, R, S, P, Q)},
                      x.options().dtype(at::kFloat));
  ddl_conv_wgrad(dy.data_ptr(), x.data_ptr(), ws.data_ptr<float>(),
                 dw.data_ptr(), N, H, W, C, K, R, S, P, Q, sh, sw, ph, pw,
                 dt, cur_stream());
  return dw;
}

// ---- batchnorm -------------------------------------------------------------

// returns (y, mean, rstd, relu_mask); relu_mask is the bit-packed [M][C/8]
// ReLU activity, non-empty only on the training relu+residual path where
// the backward would otherwise re-read y for the mask (mode 1 -> mode 3).
std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor> bn_relu_fwd(
    at::Tensor x, at::Tensor weight, at::Tensor bias, at::Tensor rmean,
    at::Tensor rvar, double momentum, double eps, bool relu,
    c10::optional<at::Tensor> residual,
    c10::optional<at::Tensor> conv_partials) {
  check_cl(x, "x");
  int64_t C = x.size(1);
  int64_t M = x.numel() / C;
  TORCH_CHECK(C % 8 == 0, "BN requires C % 8 == 0");
  TORCH_CHECK(rmean.scalar_type() == at::kFloat && rvar.scalar_type() == at::kFloat,
              "running stats must stay fp32");
  TORCH_CHECK(weight.scalar_type() == x.scalar_type());
  auto f32 = x.options().dtype(at::kFloat);
  auto mean = at::empty({C}, f32);
  auto rstd = at::empty({C}, f32);
  auto y = at::empty_like(x);
  auto s = cur_stream();
  int dt = dt_code(x);
  if (conv_partials.has_value() && conv_partials->numel() > 0) {
    // partials from the producing conv's epilogue: [chunks][2][C]
    const int64_t pchunks = conv_partials->numel() / (2 * C);
    TORCH_CHECK(pchunks * 2 * C == conv_partials->numel(),
                "conv partials shape mismatch");
    auto fold_ws = at::empty({(2 + 512) * C}, f32);
    ddl_bn_reduce_finalize(conv_partials->data_ptr<float>(),
                           fold_ws.data_ptr<float>(), (int)pchunks, C, M,
                           momentum, eps, rmean.data_ptr<float>(),
                           rvar.data_ptr<float>(), mean.data_ptr<float>(),
                           rstd.data_ptr<float>(), 1, s);
  } else {
    auto sums = at::empty({2 * C}, f32);
    int chunks = ddl_bn_stats_chunks(M, C);
    auto partials = at::empty({(int64_t)chunks * 2 * C}, f32);
    ddl_bn_stats(x.data_ptr(), M, C, dt, sums.data_ptr<float>(),
                 partials.data_ptr<float>(), chunks, s);
    ddl_bn_finalize(sums.data_ptr<float>(), C, M, momentum, eps,
                    rmean.data_ptr<float>(), rvar.data_ptr<float>(),
                    mean.data_ptr<float>(), rstd.data_ptr<float>(), 1, s);
  }
  const void* res = residual.has_value() && residual->defined()
                        ? residual->data_ptr() : nullptr;
  at::Tensor mask = at::empty({0}, x.options().dtype(at::kByte));
  uint8_t* mptr = nullptr;
  if (relu && res != nullptr) {
    mask = at::empty({M * (C / 8)}, x.options().dtype(at::kByte));
    mptr = mask.data_ptr<uint8_t>();
  }
  ddl_bn_apply_mask(x.data_ptr(), y.data_ptr(), res, mean.data_ptr<float>(),
                    rstd.data_ptr<float>(), weight.data_ptr(),
                    bias.data_ptr(), M, C, relu, dt, mptr, s);
  return {y, mean, rstd, mask};
}

at::Tensor bn_relu_infer(at::Tensor x, at::Tensor weight, at::Tensor bias,
                         at::Tensor rmean, at::Tensor rvar, double eps,
                         bool relu, c10::optional<at::Tensor> residual) {
  check_cl(x, "x");
  int64_t C = x.size(1);
  int64_t M = x.numel() / C;
  auto y = at::empty_like(x);
  const void* res = residual.has_value() && residual->defined()
                        ? residual->data_ptr() : nullptr;
  ddl_bn_infer(x.data_ptr(), y.data_ptr(), res, rmean.data_ptr<float>(),
               rvar.data_ptr<float>(), eps, weight.data_ptr(),
               bias.data_ptr(), M, C, relu, dt_code(x), cur_stream());
  return y;
}

std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor> bn_relu_bwd(
    at::Tensor dy, at::Tensor x, at::Tensor weight, at::Tensor bias,
    at::Tensor mean, at::Tensor rstd, at::Tensor y, bool relu,
    bool has_residual, c10::optional<at::Tensor> pre_partials,
    c10::optional<at::Tensor> relu_mask) {
  check_cl(dy, "dy"); check_cl(x, "x");
  int64_t C = x.size(1);
  int64_t M = x.numel() / C;
  auto s = cur_stream();
  int dt = dt_code(x);
  // relu mask source: with a residual, y != bn(x) so the mask comes from
  // the forward's bit-packed mask (mode 3; 1/16 the bytes of re-reading
  // y) or from y itself (mode 1); without one, recompute (w*xhat+b > 0).
  const uint8_t* mptr = nullptr;
  if (relu_mask.has_value() && relu_mask->numel() > 0) {
    TORCH_CHECK(relu_mask->numel() == M * (C / 8), "relu mask size");
    mptr = relu_mask->data_ptr<uint8_t>();
  }
  const int relu_mode =
      relu ? (has_residual ? (mptr ? 3 : 1) : 2) : 0;
  at::Tensor sums;
  if (pre_partials.has_value() && pre_partials->numel() > 0) {
    // stats partials already emitted by the consuming conv's dgrad
    // epilogue (conv_dgrad_bnstats): reduce them, skip the stats pass
    TORCH_CHECK(relu_mode == 2 || relu_mode == 3,
                "pre-computed bwd partials need a recomputable (mode 2) or "
                "saved-mask (mode 3) ReLU");
    const int64_t pchunks = pre_partials->numel() / (2 * C);
    TORCH_CHECK(pchunks * 2 * C == pre_partials->numel(),
                "bwd partials shape mismatch");
    auto fold_ws = at::empty({(2 + 512) * C}, x.options().dtype(at::kFloat));
    ddl_bn_reduce_sums(pre_partials->data_ptr<float>(),
                       fold_ws.data_ptr<float>(), (int)pchunks, C, s);
    sums = fold_ws.narrow(0, 0, 2 * C);
  } else {
    sums = at::empty({2 * C}, x.options().dtype(at::kFloat));
    int chunks = ddl_bn_stats_chunks(M, C);
    auto partials = at::empty({(int64_t)chunks * 2 * C},
                              x.options().dtype(at::kFloat));
    ddl_bn_bwd_stats_mask(dy.data_ptr(), x.data_ptr(), y.data_ptr(),
                          relu_mode, mean.data_ptr<float>(),
                          rstd.data_ptr<float>(), weight.data_ptr(),
                          bias.data_ptr(), M, C, dt, sums.data_ptr<float>(),
                          partials.data_ptr<float>(), chunks, mptr, s);
  }
  auto dx = at::empty_like(x);
  at::Tensor dres;
  void* dres_p = nullptr;
  if (has_residual) {
    dres = at::empty_like(x);
    dres_p = dres.data_ptr();
  }
  ddl_bn_bwd_apply_mask(dy.data_ptr(), x.data_ptr(), y.data_ptr(),
                        dx.data_ptr(), dres_p, mean.data_ptr<float>(),
                        rstd.data_ptr<float>(), weight.data_ptr(),
                        bias.data_ptr(), sums.data_ptr<float>(), M, C,
                        relu_mode, dt, mptr, s);
  auto dw = at::empty_like(weight);
  auto db = at::empty_like(weight);
  ddl_bn_grads_from_sums(sums.data_ptr<float>(), dw.data_ptr(), db.data_ptr(),
                         C, dt_code(weight), s);
  return {dx, dw, db, dres};
}

// ---- pooling ---------------------------------------------------------------

std::tuple<at::Tensor, at::Tensor> maxpool_fwd(at::Tensor x, int64_t kh,
                                               int64_t kw, int64_t sh,
                                               int64_t sw, int64_t ph,
                                               int64_t pw) {
  check_cl(x, "x");
  int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  int P = (H + 2 * ph - kh) / sh + 1;
  int Q = (W + 2 * pw - kw) / sw + 1;
  TORCH_CHECK(C % 8 == 0);
  auto y = empty_nhwc(N, C, P, Q, x.options());
  auto idx = empty_nhwc(N, C, P, Q, x.options().dtype(at::kByte));
  ddl_maxpool_fwd(x.data_ptr(), y.data_ptr(), (uint8_t*)idx.data_ptr(), N, H,
                  W, C, P, Q, kh, kw, sh, sw, ph, pw, dt_code(x),
                  cur_stream());
  return {y, idx};
}

at::Tensor maxpool_bwd(at::Tensor dy, at::Tensor idx,
                       std::vector<int64_t> x_shape, int64_t kh, int64_t kw,
                       int64_t sh, int64_t sw, int64_t ph, int64_t pw) {
  check_cl(dy, "dy");
  int N = x_shape[0], C = x_shape[1], H = x_shape[2], W = x_shape[3];
  int P = dy.size(2), Q = dy.size(3);
  auto dx = empty_nhwc(N, C, H, W, dy.options());
  ddl_maxpool_bwd(dy.data_ptr(), (const uint8_t*)idx.data_ptr(),
                  dx.data_ptr(), N, H, W, C, P, Q, kh, kw, sh, sw, ph, pw,
                  dt_code(dy), cur_stream());
  return dx;
}

at::Tensor gap_fwd(at::Tensor x) {
  check_cl(x, "x");
  int N = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
  TORCH_CHECK(C % 8 == 0);
  auto y = at::empty({N, C}, x.options());
  ddl_gap_fwd(x.data_ptr(), y.data_ptr(), N, HW, C, dt_code(x), cur_stream());
  return y;
}

at::Tensor gap_bwd(at::Tensor dy, std::vector<int64_t> x_shape) {
  int N = x_shape[0], C = x_shape[1], HW = x_shape[2] * x_shape[3];
  auto dx = empty_nhwc(N, C, x_shape[2], x_shape[3], dy.options());
  ddl_gap_bwd(dy.contiguous().data_ptr(), dx.data_ptr(), N, HW, C,
              dt_code(dy), cur_stream());
  return dx;
}

// ---- softmax-xent ----------------------------------------------------------

std::tuple<at::Tensor, at::Tensor> softmax_xent_fwd(at::Tensor logits,
                                                    at::Tensor target,
                                                    double ls) {
  TORCH_CHECK(logits.dim() == 2 && logits.is_contiguous());
  TORCH_CHECK(target.scalar_type() == at::kLong);
  int N = logits.size(0), C = logits.size(1);
  auto f32 = logits.options().dtype(at::kFloat);
  auto losses = at::empty({N}, f32);
  auto probs = at::empty({N, C}, f32);
  ddl_softmax_xent_fwd(logits.data_ptr(), target.data_ptr<int64_t>(),
                       losses.data_ptr<float>(), probs.data_ptr<float>(), N,
                       C, dt_code(logits), ls, cur_stream());
  return {losses, probs};
}

at::Tensor softmax_xent_bwd(at::Tensor probs, at::Tensor target, double ls,
                            at::Tensor dloss, int64_t out_dtype) {
  int N = probs.size(0), C = probs.size(1);
  auto dtype = out_dtype == 0 ? at::kFloat
             : out_dtype == 1 ? at::kBFloat16 : at::kHalf;
  auto dlogits = at::empty({N, C}, probs.options().dtype(dtype));
  TORCH_CHECK(dloss.scalar_type() == at::kFloat && dloss.is_cuda());
  ddl_softmax_xent_bwd(probs.data_ptr<float>(), target.data_ptr<int64_t>(),
                       dlogits.data_ptr(), N, C, (int)out_dtype, ls,
                       dloss.data_ptr<float>(), 1.0f / N, cur_stream());
  return dlogits;
}

// ---- elementwise / dropout / scaling ---------------------------------------

at::Tensor relu_bwd(at::Tensor dy, at::Tensor y) {
  auto out = at::empty_like(dy);
  ddl_relu_bwd(dy.data_ptr(), y.data_ptr(), out.data_ptr(), dy.numel(),
               dt_code(dy), cur_stream());
  return out;
}

std::tuple<at::Tensor, at::Tensor> dropout_fwd(at::Tensor x, double p,
                                               int64_t seed) {
  auto y = at::empty_like(x);
  auto mask = at::empty({x.numel()}, x.options().dtype(at::kByte));
  ddl_dropout_fwd(x.data_ptr(), y.data_ptr(), (uint8_t*)mask.data_ptr(),
                  x.numel(), dt_code(x), p, (uint64_t)seed, cur_stream());
  return {y, mask};
}

at::Tensor dropout_bwd(at::Tensor dy, at::Tensor mask, double p) {
  auto dx = at::empty_like(dy);
  ddl_dropout_bwd(dy.data_ptr(), dx.data_ptr(),
                  (const uint8_t*)mask.data_ptr(), dy.numel(), dt_code(dy), p,
                  cur_stream());
  return dx;
}

void check_finite(at::Tensor t, at::Tensor found_inf) {
  ddl_check_finite(t.data_ptr(), t.numel(), dt_code(t),
                   found_inf.data_ptr<int>(), cur_stream());
}

void unscale_check(at::Tensor t, double inv_scale, at::Tensor found_inf) {
  ddl_unscale_check(t.data_ptr(), t.numel(), dt_code(t), inv_scale,
                    found_inf.data_ptr<int>(), cur_stream());
}

// ---- fused optimizer -------------------------------------------------------

void fused_sgd(at::Tensor table, c10::optional<at::Tensor> norms,
               int64_t dtype, double lr, double momentum, double wd,
               bool nesterov, double lars_eta, double lars_eps,
               double inv_scale) {
  int nrows = table.size(0);
  const float* nptr = nullptr;
  int ntensors = 0;
  if (norms.has_value() && norms->defined()) {
    nptr = norms->data_ptr<float>();
    ntensors = norms->size(1);
  }
  ddl_fused_sgd(table.data_ptr<int64_t>(), nrows, nptr, ntensors, dtype, lr,
                momentum, wd, nesterov, lars_eta, lars_eps, inv_scale,
                cur_stream());
}

void lars_norms(at::Tensor table, at::Tensor norms, int64_t dtype, double wd,
                double inv_scale) {
  // deterministic two-stage: per-chunk partials then a fixed-order fold
  auto partials = at::empty({2 * table.size(0)},
                            norms.options().dtype(at::kFloat));
  ddl_lars_norms(table.data_ptr<int64_t>(), table.size(0),
                 norms.data_ptr<float>(), partials.data_ptr<float>(),
                 norms.size(1), dtype, wd, inv_scale, cur_stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("conv_fwd", &conv_fwd);
  m.def("conv_fwd_stats", &conv_fwd_stats);
  m.def("conv_dgrad", &conv_dgrad);
  m.def("conv_dgrad_bnstats", &conv_dgrad_bnstats);
  m.def("conv_dgrad_bnstats",
        [](at::Tensor dy, at::Tensor w, std::vector<int64_t> xs, int64_t sh,
           int64_t sw, int64_t ph, int64_t pw, at::Tensor bnx,
           at::Tensor bn_mean, at::Tensor bn_rstd, at::Tensor bn_w,
           at::Tensor bn_b) {
          return conv_dgrad_bnstats(dy, w, xs, sh, sw, ph, pw, bnx, bn_mean,
                                    bn_rstd, bn_w, bn_b, c10::nullopt,
                                    c10::nullopt);
        });
  m.def("conv_dgrad_smallc", &conv_dgrad_smallc);
  m.def("conv_dgrad",
        [](at::Tensor dy, at::Tensor w, std::vector<int64_t> xs, int64_t sh,
           int64_t sw, int64_t ph, int64_t pw) {
          return conv_dgrad(dy, w, xs, sh, sw, ph, pw, c10::nullopt);
        });
  m.def("conv_wgrad", &conv_wgrad);
  m.def("conv_wgrad",
        [](at::Tensor dy, at::Tensor x, std::vector<int64_t> ws, int64_t sh,
           int64_t sw, int64_t ph, int64_t pw) {
          return conv_wgrad(dy, x, ws, sh, sw, ph, pw, c10::nullopt);
        });
  m.def("conv_fwd_smallc", &conv_fwd_smallc);
  m.def("conv_wgrad_smallc", &conv_wgrad_smallc);
  m.def("bn_relu_fwd", &bn_relu_fwd);
  // 9-arg overload (no conv partials) for existing callers
  m.def("bn_relu_fwd",
        [](at::Tensor x, at::Tensor weight, at::Tensor bias, at::Tensor rmean,
           at::Tensor rvar, double momentum, double eps, bool relu,
           c10::optional<at::Tensor> residual) {
          return bn_relu_fwd(x, weight, bias, rmean, rvar, momentum, eps,
                             relu, residual, c10::nullopt);
        });
  m.def("bn_relu_infer", &bn_relu_infer);
  m.def("bn_relu_bwd", &bn_relu_bwd);
  m.def("bn_relu_bwd",
        [](at::Tensor dy, at::Tensor x, at::Tensor weight, at::Tensor bias,
           at::Tensor mean, at::Tensor rstd, at::Tensor y, bool relu,
           bool has_residual) {
          return bn_relu_bwd(dy, x, weight, bias, mean, rstd, y, relu,
                             has_residual, c10::nullopt, c10::nullopt);
        });
  m.def("bn_relu_bwd",
        [](at::Tensor dy, at::Tensor x, at::Tensor weight, at::Tensor bias,
           at::Tensor mean, at::Tensor rstd, at::Tensor y, bool relu,
           bool has_residual, c10::optional<at::Tensor> pre_partials) {
          return bn_relu_bwd(dy, x, weight, bias, mean, rstd, y, relu,
                             has_residual, pre_partials, c10::nullopt);
        });
  m.def("maxpool_fwd", &maxpool_fwd);
  m.def("maxpool_bwd", &maxpool_bwd);
  m.def("gap_fwd", &gap_fwd);
  m.def("gap_bwd", &gap_bwd);
  m.def("softmax_xent_fwd", &softmax_xent_fwd);
  m.def("softmax_xent_bwd", &softmax_xent_bwd);
  m.def("relu_bwd", &relu_bwd);
  m.def("dropout_fwd", &dropout_fwd);
  m.def("dropout_bwd", &dropout_bwd);
  m.def("check_finite", &check_finite);
  m.def("unscale_check", &unscale_check);
  m.def("fused_sgd", &fused_sgd);
  m.def("lars_norms", &lars_norms);
}
