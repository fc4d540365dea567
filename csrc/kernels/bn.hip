// K4/K5/K14: fused BatchNorm + residual-add + ReLU over NHWC (M rows x C).
// Training forward = 3 launches: per-channel {sum, sumsq} partials with
// atomic accumulation (stats), a tiny finalize (mean/rstd + running-stat
// update, fp32), and one fused normalize+add+relu pass.  The reference
// issues separate cuDNN BN, add and ReLU kernels (2-3 full activation
// round-trips more traffic — SURVEY.md K4/K14).
// Constraint: C % 8 == 0 (vectorized channel groups); every BN in
// ResNet/VGG has C in {64..2048}.
#include "common.h"
#include "api.h"

namespace {

constexpr int ROWS = 32;     // rows per block pass
constexpr int CG = 8;        // channel strip = CG vec8-groups = 64 channels

// ---- stats: partials[chunk][{0,1}][C] = per-chunk {sum, sumsq} ------------
// Two-stage (no atomics): 2048 chunks all adding to the same channel would
// serialize ~2000-deep per address at L2; per-chunk partials + a finalize
// reduction run at streaming bandwidth AND make BN stats deterministic.
template <typename T>
__global__ void bn_stats_k(const T* __restrict__ x, int64_t M, int C,
                           float* __restrict__ partials) {
  // block: 256 threads = CG vec-groups x ROWS rows; grid.x = C/64 strips,
  // grid.y = row chunks.
  __shared__ float red[2][ROWS][CG * 8];
  const int vg = threadIdx.x % CG;          // which vec8 group in the strip
  const int trow = threadIdx.x / CG;        // 0..31
  const int c0 = blockIdx.x * CG * 8 + vg * 8;
  if (c0 >= C) return;
  float s[8] = {0}, q[8] = {0};
  for (int64_t m = (int64_t)blockIdx.y * ROWS + trow; m < M;
       m += (int64_t)gridDim.y * ROWS) {
    Vec8<T> v = load8(x + m * C + c0);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = to_f32(v.v[j]);
      s[j] += f;
      q[j] += f * f;
    }
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    red[0][trow][vg * 8 + j] = s[j];
    red[1][trow][vg * 8 + j] = q[j];
  }
  __syncthreads();
#pragma unroll
  for (int step = ROWS / 2; step > 0; step >>= 1) {
    if (trow < step) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        red[0][trow][vg * 8 + j] += red[0][trow + step][vg * 8 + j];
        red[1][trow][vg * 8 + j] += red[1][trow + step][vg * 8 + j];
      }
    }
    __syncthreads();
  }
  if (trow == 0) {
    float* dst = partials + ((int64_t)blockIdx.y * 2) * C;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      dst[c0 + j] = red[0][0][vg * 8 + j];
      dst[C + c0 + j] = red[1][0][vg * 8 + j];
    }
  }
}

// reduce partials[chunks][2][C] into sums[2][C]: block = 256 threads =
// 8 channel-vec-groups x 32 chunk-rows (same shape as the stats kernel so
// the reads stay coalesced and the reduction is parallel over chunks).
__global__ void bn_reduce_partials_k(const float* __restrict__ partials,
                                     float* __restrict__ sums, int C,
                                     int chunks) {
  __shared__ float red[2][ROWS][CG * 8];
  const int vg = threadIdx.x % CG;
  const int trow = threadIdx.x / CG;
  const int c0 = blockIdx.x * CG * 8 + vg * 8;
  if (c0 >= C) return;
  float s0[8] = {0}, s1[8] = {0};
  for (int i = trow; i < chunks; i += ROWS) {
    const float* p0 = partials + ((int64_t)i * 2) * C + c0;
    const float* p1 = p0 + C;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      s0[j] += p0[j];
      s1[j] += p1[j];
    }
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    red[0][trow][vg * 8 + j] = s0[j];
    red[1][trow][vg * 8 + j] = s1[j];
  }
  __syncthreads();
#pragma unroll
  for (int step = ROWS / 2; step > 0; step >>= 1) {
    if (trow < step) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        red[0][trow][vg * 8 + j] += red[0][trow + step][vg * 8 + j];
        red[1][trow][vg * 8 + j] += red[1][trow + step][vg * 8 + j];
      }
    }
    __syncthreads();
  }
  if (trow == 0) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      sums[c0 + j] = red[0][0][vg * 8 + j];
      sums[C + c0 + j] = red[1][0][vg * 8 + j];
    }
  }
}

// fold partials[chunks][2][C] -> fold_ws[FOLD][2][C]: group g sums chunks
// g, g+FOLD, ... (fixed order, deterministic).  The conv-epilogue partials
// have one chunk PER 128-ROW M-TILE (tens of thousands for 56^2 layers);
// bn_reduce_partials_k alone has only C/64 blocks of parallelism there.
__global__ void bn_fold_partials_k(const float* __restrict__ partials,
                                   float* __restrict__ fold_ws, int C,
                                   int chunks, int fold) {
  const int vg = threadIdx.x % CG;
  const int trow = threadIdx.x / CG;
  const int c0 = blockIdx.x * CG * 8 + vg * 8;
  const int g = blockIdx.y;
  if (c0 >= C) return;
  __shared__ float red[2][ROWS][CG * 8];
  float s0[8] = {0}, s1[8] = {0};
  for (int i = g + trow * fold; i < chunks; i += fold * ROWS) {
    const float* p0 = partials + ((int64_t)i * 2) * C + c0;
    const float* p1 = p0 + C;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      s0[j] += p0[j];
      s1[j] += p1[j];
    }
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    red[0][trow][vg * 8 + j] = s0[j];
    red[1][trow][vg * 8 + j] = s1[j];
  }
  __syncthreads();
#pragma unroll
  for (int step = ROWS / 2; step > 0; step >>= 1) {
    if (trow < step) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        red[0][trow][vg * 8 + j] += red[0][trow + step][vg * 8 + j];
        red[1][trow][vg * 8 + j] += red[1][trow + step][vg * 8 + j];
      }
    }
    __syncthreads();
  }
  if (trow == 0) {
    float* dst = fold_ws + ((int64_t)g * 2) * C;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      dst[c0 + j] = red[0][0][vg * 8 + j];
      dst[C + c0 + j] = red[1][0][vg * 8 + j];
    }
  }
}

__global__ void bn_finalize_k(const float* __restrict__ sums, int C,
                              float inv_M, float unbias, float momentum,
                              float eps, float* running_mean,
                              float* running_var, float* mean_out,
                              float* rstd_out, int update_running) {
  for (int c = threadIdx.x + blockIdx.x * blockDim.x; c < C;
       c += blockDim.x * gridDim.x) {
    float mean = sums[c] * inv_M;
    float var = fmaxf(sums[C + c] * inv_M - mean * mean, 0.f);
    mean_out[c] = mean;
    rstd_out[c] = rsqrtf(var + eps);
    if (update_running) {
      running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
      running_var[c] = (1.f - momentum) * running_var[c] + momentum * var * unbias;
    }
  }
}

// ---- apply: y = relu?((x-mean)*rstd*w + b + res?) -------------------------
// mask (training relu+residual path): bit-packed ReLU activity, one byte
// per 8-channel group (MC/8 bytes = 1/16 of an activation pass) — the
// backward then reads the mask instead of re-reading y (mode 3), saving a
// full y pass in BOTH bn_bwd_stats and bn_bwd_apply for the residual-join
// BNs (mode 1), the widest-channel layers of the net.
template <typename T, bool RELU, bool RES, bool MASK = false>
__global__ void bn_apply_k(const T* __restrict__ x, T* __restrict__ y,
                           const T* __restrict__ res,
                           const float* __restrict__ mean,
                           const float* __restrict__ rstd,
                           const T* __restrict__ weight,
                           const T* __restrict__ bias, int64_t M, int C,
                           uint8_t* __restrict__ mask = nullptr) {
  const int ncg = C / 8;
  const int64_t total = M * ncg;
  for (int64_t i = global_tid(); i < total; i += grid_stride()) {
    const int cg = (int)(i % ncg);
    const int64_t m = i / ncg;
    const int c0 = cg * 8;
    Vec8<T> v = load8(x + m * C + c0);
    Vec8<T> w8 = load8(weight + c0);
    Vec8<T> b8 = load8(bias + c0);
    Vec8<T> r8;
    if (RES) r8 = load8(res + m * C + c0);
    uint8_t mb = 0;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = (to_f32(v.v[j]) - mean[c0 + j]) * rstd[c0 + j];
      f = f * to_f32(w8.v[j]) + to_f32(b8.v[j]);
      if (RES) f += to_f32(r8.v[j]);
      if (RELU) {
        if (MASK && f > 0.f) mb |= (uint8_t)(1u << j);
        f = fmaxf(f, 0.f);
      }
      v.v[j] = from_f32<T>(f);
    }
    store8(y + m * C + c0, v);
    if (MASK) mask[m * ncg + cg] = mb;
  }
}

// inference: same but scale/shift precomputed from running stats on the fly
template <typename T, bool RELU, bool RES>
__global__ void bn_infer_k(const T* __restrict__ x, T* __restrict__ y,
                           const T* __restrict__ res,
                           const float* __restrict__ rmean,
                           const float* __restrict__ rvar, float eps,
                           const T* __restrict__ weight,
                           const T* __restrict__ bias, int64_t M, int C) {
  const int ncg = C / 8;
  const int64_t total = M * ncg;
  for (int64_t i = global_tid(); i < total; i += grid_stride()) {
    const int cg = (int)(i % ncg);
    const int64_t m = i / ncg;
    const int c0 = cg * 8;
    Vec8<T> v = load8(x + m * C + c0);
    Vec8<T> w8 = load8(weight + c0);
    Vec8<T> b8 = load8(bias + c0);
    Vec8<T> r8;
    if (RES) r8 = load8(res + m * C + c0);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float rs = rsqrtf(rvar[c0 + j] + eps);
      float f = (to_f32(v.v[j]) - rmean[c0 + j]) * rs;
      f = f * to_f32(w8.v[j]) + to_f32(b8.v[j]);
      if (RES) f += to_f32(r8.v[j]);
      if (RELU) f = fmaxf(f, 0.f);
      v.v[j] = from_f32<T>(f);
    }
    store8(y + m * C + c0, v);
  }
}

// ---- backward stats: sums[c] = sum(dy_m * xhat), sums[C+c] = sum(dy_m) ----
// RELU modes: 0 = no relu; 1 = mask from y (residual case — y != bn(x));
// 2 = recompute mask as (w*xhat + b > 0), skipping the y read entirely
// (one full activation pass saved; exact for relu-without-residual);
// 3 = bit-packed mask saved by the forward (residual case, 1/16 the
// bytes of the y read it replaces).
template <typename T, int RELU>
__global__ void bn_bwd_stats_k(const T* __restrict__ dy,
                               const T* __restrict__ x,
                               const T* __restrict__ y,
                               const float* __restrict__ mean,
                               const float* __restrict__ rstd,
                               const T* __restrict__ weight,
                               const T* __restrict__ bias, int64_t M,
                               int C, float* __restrict__ sums,
                               const uint8_t* __restrict__ mask = nullptr) {
  __shared__ float red[2][ROWS][CG * 8];
  const int vg = threadIdx.x % CG;
  const int trow = threadIdx.x / CG;
  const int c0 = blockIdx.x * CG * 8 + vg * 8;
  if (c0 >= C) return;
  float s0[8] = {0}, s1[8] = {0};
  Vec8<T> w8, b8;
  if (RELU == 2) {
    w8 = load8(weight + c0);
    b8 = load8(bias + c0);
  }
  for (int64_t m = (int64_t)blockIdx.y * ROWS + trow; m < M;
       m += (int64_t)gridDim.y * ROWS) {
    Vec8<T> vdy = load8(dy + m * C + c0);
    Vec8<T> vx = load8(x + m * C + c0);
    Vec8<T> vy;
    if (RELU == 1) vy = load8(y + m * C + c0);
    uint8_t mb = 0xff;
    if (RELU == 3) mb = mask[m * (C / 8) + c0 / 8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = to_f32(vdy.v[j]);
      float xh = (to_f32(vx.v[j]) - mean[c0 + j]) * rstd[c0 + j];
      if (RELU == 1) g = to_f32(vy.v[j]) > 0.f ? g : 0.f;
      if (RELU == 2)
        g = (xh * to_f32(w8.v[j]) + to_f32(b8.v[j])) > 0.f ? g : 0.f;
      if (RELU == 3) g = (mb >> j) & 1 ? g : 0.f;
      s0[j] += g * xh;
      s1[j] += g;
    }
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    red[0][trow][vg * 8 + j] = s0[j];
    red[1][trow][vg * 8 + j] = s1[j];
  }
  __syncthreads();
#pragma unroll
  for (int step = ROWS / 2; step > 0; step >>= 1) {
    if (trow < step) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        red[0][trow][vg * 8 + j] += red[0][trow + step][vg * 8 + j];
        red[1][trow][vg * 8 + j] += red[1][trow + step][vg * 8 + j];
      }
    }
    __syncthreads();
  }
  if (trow == 0) {
    float* dst = sums + ((int64_t)blockIdx.y * 2) * C;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      dst[c0 + j] = red[0][0][vg * 8 + j];
      dst[C + c0 + j] = red[1][0][vg * 8 + j];
    }
  }
}

// ---- backward apply: dx = w*rstd*(dy_m - s1/M - xhat*s0/M); dres = dy_m ---
// RELU: 0 none / 1 mask from y (residual) / 2 recomputed (w*xhat+b > 0).
template <typename T, int RELU, bool RES>
__global__ void bn_bwd_apply_k(const T* __restrict__ dy,
                               const T* __restrict__ x,
                               const T* __restrict__ y, T* __restrict__ dx,
                               T* __restrict__ dres,
                               const float* __restrict__ mean,
                               const float* __restrict__ rstd,
                               const T* __restrict__ weight,
                               const T* __restrict__ bias,
                               const float* __restrict__ sums, int64_t M,
                               int C,
                               const uint8_t* __restrict__ mask = nullptr) {
  const int ncg = C / 8;
  const int64_t total = M * ncg;
  const float invM = 1.f / (float)M;
  for (int64_t i = global_tid(); i < total; i += grid_stride()) {
    const int cg = (int)(i % ncg);
    const int64_t m = i / ncg;
    const int c0 = cg * 8;
    Vec8<T> vdy = load8(dy + m * C + c0);
    Vec8<T> vx = load8(x + m * C + c0);
    Vec8<T> w8 = load8(weight + c0);
    Vec8<T> b8;
    if (RELU == 2) b8 = load8(bias + c0);
    Vec8<T> vy, out, rout;
    if (RELU == 1) vy = load8(y + m * C + c0);
    uint8_t mb = 0xff;
    if (RELU == 3) mb = mask[m * ncg + cg];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int c = c0 + j;
      float g = to_f32(vdy.v[j]);
      float xh = (to_f32(vx.v[j]) - mean[c]) * rstd[c];
      if (RELU == 1) g = to_f32(vy.v[j]) > 0.f ? g : 0.f;
      if (RELU == 2)
        g = (xh * to_f32(w8.v[j]) + to_f32(b8.v[j])) > 0.f ? g : 0.f;
      if (RELU == 3) g = (mb >> j) & 1 ? g : 0.f;
      if (RES) rout.v[j] = from_f32<T>(g);
      float t = g - sums[C + c] * invM - xh * sums[c] * invM;
      out.v[j] = from_f32<T>(to_f32(w8.v[j]) * rstd[c] * t);
    }
    store8(dx + m * C + c0, out);
    if (RES) store8(dres + m * C + c0, rout);
  }
}

template <typename T>
__global__ void bn_grads_k(const float* sums, T* dw, T* db, int C) {
  for (int c = threadIdx.x + blockIdx.x * blockDim.x; c < C;
       c += blockDim.x * gridDim.x) {
    dw[c] = from_f32<T>(sums[c]);
    db[c] = from_f32<T>(sums[C + c]);
  }
}

inline dim3 stats_grid(int64_t M, int C) {
  int strips = (C + 63) / 64;
  int64_t want = 2048 / strips;
  int64_t chunks = std::min<int64_t>(cdiv(M, ROWS), std::max<int64_t>(want, 1));
  return dim3(strips, (unsigned)chunks);
}

inline int apply_grid(int64_t M, int C) {
  return (int)std::min<int64_t>(cdiv(M * (C / 8), 256), 2048);
}

}  // namespace

extern "C" {

int ddl_bn_stats_chunks(int64_t M, int C) { return (int)stats_grid(M, C).y; }

void ddl_bn_stats(const void* x, int64_t M, int C, int dtype, float* sums,
                  float* partials, int max_chunks, hipStream_t s) {
  dim3 g = stats_grid(M, C);
  if ((int)g.y > max_chunks) g.y = max_chunks;
#define RUN(T) bn_stats_k<T><<<g, 256, 0, s>>>((const T*)x, M, C, partials)
  switch (dtype) { case 0: RUN(float); break; case 1: RUN(bf16); break; case 2: RUN(f16); break; }
#undef RUN
  bn_reduce_partials_k<<<cdiv(C, 64), 256, 0, s>>>(partials, sums, C, g.y);
}

void ddl_bn_finalize(const float* sums, int C, int64_t M, float momentum,
                     float eps, float* running_mean, float* running_var,
                     float* mean_out, float* rstd_out, int update_running,
                     hipStream_t s) {
  float unbias = M > 1 ? (float)M / (float)(M - 1) : 1.f;
  bn_finalize_k<<<cdiv(C, 256), 256, 0, s>>>(
      sums, C, 1.f / (float)M, unbias, momentum, eps, running_mean,
      running_var, mean_out, rstd_out, update_running);
}

void ddl_bn_reduce_finalize(const float* partials, float* fold_ws, int chunks,
                            int C, int64_t M, float momentum, float eps,
                            float* running_mean, float* running_var,
                            float* mean_out, float* rstd_out,
                            int update_running, hipStream_t s) {
  // sums land in the first 2*C of fold_ws after the final reduce
  float* sums = fold_ws;
  if (chunks > 512) {
    const int fold = 256;
    float* stage = fold_ws + 2 * (int64_t)C;  // [fold][2][C]
    dim3 g(cdiv(C, 64), fold);
    bn_fold_partials_k<<<g, 256, 0, s>>>(partials, stage, C, chunks, fold);
    bn_reduce_partials_k<<<cdiv(C, 64), 256, 0, s>>>(stage, sums, C, fold);
  } else {
    bn_reduce_partials_k<<<cdiv(C, 64), 256, 0, s>>>(partials, sums, C,
                                                     chunks);
  }
  float unbias = M > 1 ? (float)M / (float)(M - 1) : 1.f;
  bn_finalize_k<<<cdiv(C, 256), 256, 0, s>>>(
      sums, C, 1.f / (float)M, unbias, momentum, eps, running_mean,
      running_var, mean_out, rstd_out, update_running);
}

// reduce-only (no finalize): bwd partials [chunks][2][C] -> sums in
// fold_ws[0..2C).  fold_ws must hold (2 + 512) * C floats for the
// two-stage path (stage slab [256][2][C]).
void ddl_bn_reduce_sums(const float* partials, float* fold_ws, int chunks,
                        int C, hipStream_t s) {
  float* sums = fold_ws;
  if (chunks > 512) {
    const int fold = 256;
    float* stage = fold_ws + 2 * (int64_t)C;
    dim3 g(cdiv(C, 64), fold);
    bn_fold_partials_k<<<g, 256, 0, s>>>(partials, stage, C, chunks, fold);
    bn_reduce_partials_k<<<cdiv(C, 64), 256, 0, s>>>(stage, sums, C, fold);
  } else {
    bn_reduce_partials_k<<<cdiv(C, 64), 256, 0, s>>>(partials, sums, C,
                                                     chunks);
  }
}

void ddl_bn_apply(const void* x, void* y, const void* residual,
                  const float* mean, const float* rstd, const void* weight,
                  const void* bias, int64_t M, int C, int relu, int dtype,
                  hipStream_t s) {
  ddl_bn_apply_mask(x, y, residual, mean, rstd, weight, bias, M, C, relu,
                    dtype, nullptr, s);
}

// relu_mask != nullptr (training relu paths): also write the bit-packed
// ReLU mask [M][C/8] consumed by the mode-3 backward.
void ddl_bn_apply_mask(const void* x, void* y, const void* residual,
                       const float* mean, const float* rstd,
                       const void* weight, const void* bias, int64_t M,
                       int C, int relu, int dtype, uint8_t* relu_mask,
                       hipStream_t s) {
  int g = apply_grid(M, C);
#define RUN(T, RELU, RES, MASK)                                                \
  bn_apply_k<T, RELU, RES, MASK><<<g, 256, 0, s>>>(                            \
      (const T*)x, (T*)y, (const T*)residual, mean, rstd, (const T*)weight,    \
      (const T*)bias, M, C, relu_mask)
#define SEL(T)                                                                 \
  if (relu && residual && relu_mask) RUN(T, true, true, true);                 \
  else if (relu && residual) RUN(T, true, true, false);                        \
  else if (relu && relu_mask) RUN(T, true, false, true);                       \
  else if (relu) RUN(T, true, false, false);                                   \
  else if (residual) RUN(T, false, true, false);                               \
  else RUN(T, false, false, false)
  switch (dtype) { case 0: { SEL(float); } break; case 1: { SEL(bf16); } break; case 2: { SEL(f16); } break; }
#undef SEL
#undef RUN
}

void ddl_bn_infer(const void* x, void* y, const void* residual,
                  const float* running_mean, const float* running_var,
                  float eps, const void* weight, const void* bias, int64_t M,
                  int C, int relu, int dtype, hipStream_t s) {
  int g = apply_grid(M, C);
#define RUN(T, RELU, RES)                                                      \
  bn_infer_k<T, RELU, RES><<<g, 256, 0, s>>>(                                  \
      (const T*)x, (T*)y, (const T*)residual, running_mean, running_var, eps,  \
      (const T*)weight, (const T*)bias, M, C)
#define SEL(T)                                                                 \
  if (relu && residual) RUN(T, true, true);                                    \
  else if (relu) RUN(T, true, false);                                          \
  else if (residual) RUN(T, false, true);                                      \
  else RUN(T, false, false)
  switch (dtype) { case 0: { SEL(float); } break; case 1: { SEL(bf16); } break; case 2: { SEL(f16); } break; }
#undef SEL
#undef RUN
}

void ddl_bn_bwd_stats(const void* dy, const void* x, const void* y,
                      int relu_mode, const float* mean, const float* rstd,
                      const void* weight, const void* bias, int64_t M, int C,
                      int dtype, float* sums, float* partials, int max_chunks,
                      hipStream_t s) {
  ddl_bn_bwd_stats_mask(dy, x, y, relu_mode, mean, rstd, weight, bias, M, C,
                        dtype, sums, partials, max_chunks, nullptr, s);
}

void ddl_bn_bwd_stats_mask(const void* dy, const void* x, const void* y,
                           int relu_mode, const float* mean,
                           const float* rstd, const void* weight,
                           const void* bias, int64_t M, int C, int dtype,
                           float* sums, float* partials, int max_chunks,
                           const uint8_t* relu_mask, hipStream_t s) {
  dim3 g = stats_grid(M, C);
  if ((int)g.y > max_chunks) g.y = max_chunks;
#define RUN(T, RELU)                                                           \
  bn_bwd_stats_k<T, RELU><<<g, 256, 0, s>>>(                                   \
      (const T*)dy, (const T*)x, (const T*)y, mean, rstd, (const T*)weight,    \
      (const T*)bias, M, C, partials, relu_mask)
#define SEL(T)                                                                 \
  if (relu_mode == 3) RUN(T, 3); else if (relu_mode == 2) RUN(T, 2);           \
  else if (relu_mode == 1) RUN(T, 1); else RUN(T, 0)
  switch (dtype) { case 0: { SEL(float); } break; case 1: { SEL(bf16); } break; case 2: { SEL(f16); } break; }
#undef SEL
#undef RUN
  bn_reduce_partials_k<<<cdiv(C, 64), 256, 0, s>>>(partials, sums, C, g.y);
}

void ddl_bn_bwd_apply(const void* dy, const void* x, const void* y, void* dx,
                      void* dres, const float* mean, const float* rstd,
                      const void* weight, const void* bias, const float* sums,
                      int64_t M, int C, int relu_mode, int dtype,
                      hipStream_t s) {
  ddl_bn_bwd_apply_mask(dy, x, y, dx, dres, mean, rstd, weight, bias, sums,
                        M, C, relu_mode, dtype, nullptr, s);
}

void ddl_bn_bwd_apply_mask(const void* dy, const void* x, const void* y,
                           void* dx, void* dres, const float* mean,
                           const float* rstd, const void* weight,
                           const void* bias, const float* sums, int64_t M,
                           int C, int relu_mode, int dtype,
                           const uint8_t* relu_mask, hipStream_t s) {
  int g = apply_grid(M, C);
#define RUN(T, RELU, RES)                                                      \
  bn_bwd_apply_k<T, RELU, RES><<<g, 256, 0, s>>>(                              \
      (const T*)dy, (const T*)x, (const T*)y, (T*)dx, (T*)dres, mean, rstd,    \
      (const T*)weight, (const T*)bias, sums, M, C, relu_mask)
#define SEL(T)                                                                 \
  if (relu_mode == 3 && dres) RUN(T, 3, true);                                 \
  else if (relu_mode == 3) RUN(T, 3, false);                                   \
  else if (relu_mode == 1 && dres) RUN(T, 1, true);                            \
  else if (relu_mode == 2 && dres) RUN(T, 2, true);                            \
  else if (relu_mode == 2) RUN(T, 2, false);                                   \
  else if (relu_mode == 1) RUN(T, 1, false);                                   \
  else if (dres) RUN(T, 0, true);                                              \
  else RUN(T, 0, false)
  switch (dtype) { case 0: { SEL(float); } break; case 1: { SEL(bf16); } break; case 2: { SEL(f16); } break; }
#undef SEL
#undef RUN
}

void ddl_bn_grads_from_sums(const float* sums, void* dweight, void* dbias,
                            int C, int dtype, hipStream_t s) {
#define RUN(T) bn_grads_k<T><<<cdiv(C, 256), 256, 0, s>>>(sums, (T*)dweight, (T*)dbias, C)
  switch (dtype) { case 0: RUN(float); break; case 1: RUN(bf16); break; case 2: RUN(f16); break; }
#undef RUN
}

}  // extern "C"
