// Elementwise utilities: ReLU backward mask, finite checks / unscale (K13),
// dropout (K15, Philox-lite counter RNG), f32 casts, weight transpose (K12).
// All memory-bound: vectorized 8-wide loads (guide G13), grid-stride.
#include "common.h"
#include "api.h"

namespace {

template <typename T>
__global__ void relu_bwd_k(const T* dy, const T* y, T* out, int64_t n) {
  int64_t i0 = global_tid() * 8;
  for (int64_t i = i0; i < n; i += grid_stride() * 8) {
    if (i + 8 <= n) {
      Vec8<T> vdy = load8(dy + i), vy = load8(y + i), r;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        r.v[j] = to_f32(vy.v[j]) > 0.f ? vdy.v[j] : from_f32<T>(0.f);
      store8(out + i, r);
    } else {
      for (int64_t k = i; k < n; ++k)
        out[k] = to_f32(y[k]) > 0.f ? dy[k] : from_f32<T>(0.f);
    }
  }
}

// Bit-level non-finite test (exponent all-ones) — immune to -ffast-math.
__device__ __forceinline__ bool is_bad(float v) {
  uint32_t b = __float_as_uint(v);
  return ((b >> 23) & 0xFFu) == 0xFFu;
}
__device__ __forceinline__ bool is_bad(bf16 v) {
  uint16_t b = *reinterpret_cast<uint16_t*>(&v);
  return ((b >> 7) & 0xFFu) == 0xFFu;
}
__device__ __forceinline__ bool is_bad(f16 v) {
  uint16_t b = *reinterpret_cast<uint16_t*>(&v);
  return ((b >> 10) & 0x1Fu) == 0x1Fu;
}

template <typename T>
__global__ void check_finite_k(const T* g, int64_t n, int* found_inf) {
  int bad = 0;
  int64_t i0 = global_tid() * 8;
  for (int64_t i = i0; i < n; i += grid_stride() * 8) {
    if (i + 8 <= n) {
      Vec8<T> v = load8(g + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) bad |= is_bad(v.v[j]);
    } else {
      for (int64_t k = i; k < n; ++k) bad |= is_bad(g[k]);
    }
  }
  if (bad) atomicOr(found_inf, 1);
}

template <typename T>
__global__ void unscale_check_k(T* g, int64_t n, float inv_scale,
                                int* found_inf) {
  int bad = 0;
  int64_t i0 = global_tid() * 8;
  for (int64_t i = i0; i < n; i += grid_stride() * 8) {
    if (i + 8 <= n) {
      Vec8<T> v = load8(g + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        bad |= is_bad(v.v[j]);
        v.v[j] = from_f32<T>(to_f32(v.v[j]) * inv_scale);
      }
      store8(g + i, v);
    } else {
      for (int64_t k = i; k < n; ++k) {
        bad |= is_bad(g[k]);
        g[k] = from_f32<T>(to_f32(g[k]) * inv_scale);
      }
    }
  }
  if (bad) atomicOr(found_inf, 1);
}

// Philox-lite 32-bit counter RNG: 4 rounds of the Philox-2x32 round
// function — statistically fine for dropout masks (not crypto).
__device__ __forceinline__ uint32_t philox2x32(uint64_t seed, uint64_t ctr) {
  uint32_t k = (uint32_t)(seed ^ (seed >> 32));
  uint32_t c0 = (uint32_t)ctr, c1 = (uint32_t)(ctr >> 32);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    uint64_t prod = (uint64_t)0xD256D193u * c0;
    uint32_t hi = (uint32_t)(prod >> 32), lo = (uint32_t)prod;
    c0 = hi ^ c1 ^ k;
    c1 = lo;
    k += 0x9E3779B9u;
  }
  return c0;
}

template <typename T>
__global__ void dropout_fwd_k(const T* x, T* y, uint8_t* mask, int64_t n,
                              float p, uint64_t seed) {
  float keep_scale = 1.f / (1.f - p);
  uint32_t thresh = (uint32_t)(p * 4294967296.0f);
  for (int64_t i = global_tid(); i < n; i += grid_stride()) {
    uint32_t r = philox2x32(seed, (uint64_t)i);
    uint8_t keep = r >= thresh;
    mask[i] = keep;
    y[i] = keep ? from_f32<T>(to_f32(x[i]) * keep_scale) : from_f32<T>(0.f);
  }
}

template <typename T>
__global__ void dropout_bwd_k(const T* dy, T* dx, const uint8_t* mask,
                              int64_t n, float p) {
  float keep_scale = 1.f / (1.f - p);
  for (int64_t i = global_tid(); i < n; i += grid_stride()) {
    dx[i] = mask[i] ? from_f32<T>(to_f32(dy[i]) * keep_scale)
                    : from_f32<T>(0.f);
  }
}

template <typename T>
__global__ void cast_f32_k(const float* src, T* dst, int64_t n) {
  for (int64_t i = global_tid(); i < n; i += grid_stride())
    dst[i] = from_f32<T>(src[i]);
}

// [K, RS, C] -> [C, RS, K]: tiled LDS transpose of the (K, C) planes for
// each rs.  Grid: (tilesK x tilesC, RS).  32x32 tiles, padded LDS row.
template <typename T>
__global__ void transpose_kc_k(const T* w, T* w_t, int K, int RS, int C) {
  __shared__ T tile[32][33];
  int rs = blockIdx.y;
  int ntc = (C + 31) / 32;
  int tk = (blockIdx.x / ntc) * 32, tc = (blockIdx.x % ntc) * 32;
  int lx = threadIdx.x % 32, ly = threadIdx.x / 32;  // 32x8 threads
#pragma unroll
  for (int yy = 0; yy < 32; yy += 8) {
    int k = tk + ly + yy, c = tc + lx;
    tile[ly + yy][lx] = (k < K && c < C)
        ? w[((int64_t)k * RS + rs) * C + c] : from_f32<T>(0.f);
  }
  __syncthreads();
#pragma unroll
  for (int yy = 0; yy < 32; yy += 8) {
    int c = tc + ly + yy, k = tk + lx;
    if (c < C && k < K)
      w_t[((int64_t)c * RS + rs) * K + k] = tile[lx][ly + yy];
  }
}

// im2col for small-C stems (conv1: 7x7 s2 C=3): rows m=(n,p,q), cols
// kred=(r,s,c) zero-padded to `ldout` (multiple of 8) so the tiled MFMA
// conv/wgrad path can consume it as a plain GEMM operand.
// tbl[kred] = (r<<20)|(s<<10)|c precomputed host-side (integer divides in
// the inner gather cost ~8 divides/16B and dominated the first version).
template <typename T>
__global__ void im2col_k(const T* __restrict__ x, T* __restrict__ out,
                         const int* __restrict__ tbl, int N, int H, int W,
                         int C, int P, int Q, int sh, int sw, int ph, int pw,
                         int kred_total, int ldout) {
  const int64_t M = (int64_t)N * P * Q;
  constexpr int EPU0 = 16 / (int)sizeof(T);
  const int64_t total = M * (ldout / EPU0);
  for (int64_t i = global_tid(); i < total; i += grid_stride()) {
    const int ug = (int)(i % (ldout / EPU0));
    const int64_t m = i / (ldout / EPU0);
    const int q = (int)(m % Q);
    const int p = (int)((m / Q) % P);
    const int n = (int)(m / ((int64_t)Q * P));
    const int h0 = p * sh - ph, w0 = q * sw - pw;
    constexpr int EPU = 16 / (int)sizeof(T);
    T e[EPU] = {};
#pragma unroll
    for (int j = 0; j < EPU; ++j) {
      const int kred = ug * EPU + j;
      if (kred < kred_total) {
        const int t = tbl[kred];
        const int hi = h0 + (t >> 20);
        const int wi = w0 + ((t >> 10) & 1023);
        if (hi >= 0 && hi < H && wi >= 0 && wi < W)
          e[j] = x[(((int64_t)n * H + hi) * W + wi) * C + (t & 1023)];
      }
    }
    *reinterpret_cast<uint4*>(out + m * ldout + ug * EPU) =
        *reinterpret_cast<const uint4*>(e);
  }
}

// Row-staged im2col: one block per (n, p) output row-group stages the R
// input rows it needs into LDS (vectorized 16 B loads; W*C is a multiple
// of 8 for the stem's 224*3 so rows stay 16 B aligned) and gathers the
// (r,s,c) taps from LDS.  The per-element global-gather kernel above is
// instruction/L1-transaction bound (8 scalar 2 B loads per 16 B unit,
// VALUBusy ~56%) at ~5x its bandwidth roofline; staging turns the global
// traffic into clean 16 B streams.
template <typename T>
__global__ __launch_bounds__(256) void im2col_rows_k(
    const T* __restrict__ x, T* __restrict__ out,
    const int* __restrict__ tbl, int N, int H, int W, int C, int R, int P,
    int Q, int sh, int sw, int ph, int pw, int kred_total, int ldout) {
  __shared__ __align__(16) char smem[48 * 1024];
  T* slab = reinterpret_cast<T*>(smem);
  const int n = blockIdx.x / P, p = blockIdx.x % P;
  const int h0 = p * sh - ph;
  const int rowlen = W * C;  // elements per input row
  constexpr int EPU = 16 / (int)sizeof(T);
  // stage R rows (out-of-image rows as zeros)
  const int units_per_row = rowlen / EPU;
  for (int u = threadIdx.x; u < R * units_per_row; u += 256) {
    const int r = u / units_per_row;
    const int uu = u % units_per_row;
    const int hi = h0 + r;
    uint4 v = uint4{0, 0, 0, 0};
    if (hi >= 0 && hi < H)
      v = *reinterpret_cast<const uint4*>(
          x + ((int64_t)n * H + hi) * rowlen + uu * EPU);
    *reinterpret_cast<uint4*>(slab + r * rowlen + uu * EPU) = v;
  }
  __syncthreads();
  const int nunits = ldout / EPU;
  const int64_t m_base = ((int64_t)n * P + p) * Q;
  for (int i = threadIdx.x; i < Q * nunits; i += 256) {
    const int q = i / nunits;
    const int ug = i % nunits;
    const int w0 = q * sw - pw;
    T e[EPU] = {};
#pragma unroll
    for (int j = 0; j < EPU; ++j) {
      const int kred = ug * EPU + j;
      if (kred < kred_total) {
        const int t = tbl[kred];
        const int wi = w0 + ((t >> 10) & 1023);
        if (wi >= 0 && wi < W)
          e[j] = slab[(t >> 20) * rowlen + wi * C + (t & 1023)];
      }
    }
    *reinterpret_cast<uint4*>(out + (m_base + q) * ldout + ug * EPU) =
        *reinterpret_cast<const uint4*>(e);
  }
}

// zero-pad the K x kred weight matrix into K x ldout
template <typename T>
__global__ void pad_cols_k(const T* __restrict__ w, T* __restrict__ out,
                           int K, int kred, int ldout) {
  const int64_t total = (int64_t)K * ldout;
  for (int64_t i = global_tid(); i < total; i += grid_stride()) {
    const int col = (int)(i % ldout);
    const int64_t k = i / ldout;
    out[i] = col < kred ? w[k * kred + col] : from_f32<T>(0.f);
  }
}

// narrow K x ldout back to K x kred (wgrad output)
template <typename T>
__global__ void narrow_cols_k(const T* __restrict__ w, T* __restrict__ out,
                              int K, int kred, int ldout) {
  const int64_t total = (int64_t)K * kred;
  for (int64_t i = global_tid(); i < total; i += grid_stride()) {
    const int col = (int)(i % kred);
    const int64_t k = i / kred;
    out[i] = w[k * ldout + col];
  }
}

inline int ew_grid(int64_t n, int per_thread = 8, int block = 256) {
  int64_t blocks = cdiv(n, (int64_t)block * per_thread);
  return (int)std::min<int64_t>(blocks, 2048);
}


// col2im scatter-add (stem dgrad, C % 8 != 0): dcol [N*P*Q][ldout] rows are
// the per-output-pixel im2col gradients; each (r,s,c) column element adds
// into dx_ws[n, p*sh+r-ph, q*sw+s-pw, c] (fp32 accumulation via atomics --
// each dx element receives ~R*S/(sh*sw) contributions from distinct rows,
// negligible contention).  tbl packs (r<<20 | s<<10 | c) like im2col.
template <typename T>
__global__ void col2im_acc_k(const T* __restrict__ dcol,
                             float* __restrict__ dx_ws,
                             const int* __restrict__ tbl, int N, int H,
                             int W, int C, int P, int Q, int sh, int sw,
                             int ph, int pw, int kred_total, int ldout) {
  const int64_t total = (int64_t)N * P * Q * kred_total;
  for (int64_t i = global_tid(); i < total; i += grid_stride()) {
    const int kred = (int)(i % kred_total);
    const int64_t m = i / kred_total;
    const float v = to_f32(dcol[m * ldout + kred]);
    if (v == 0.f) continue;
    const int q = (int)(m % Q);
    const int64_t np = m / Q;
    const int p = (int)(np % P);
    const int n = (int)(np / P);
    const int t = tbl[kred];
    const int hi = p * sh + (t >> 20) - ph;
    const int wi = q * sw + ((t >> 10) & 1023) - pw;
    if (hi >= 0 && hi < H && wi >= 0 && wi < W)
      atomicAdd(dx_ws + (((int64_t)n * H + hi) * W + wi) * C + (t & 1023), v);
  }
}

}  // namespace

#define DISPATCH(dtype, F, ...)                                \
  switch (dtype) {                                             \
    case 0: F<float>(__VA_ARGS__); break;                      \
    case 1: F<bf16>(__VA_ARGS__); break;                       \
    case 2: F<f16>(__VA_ARGS__); break;                        \
  }

extern "C" {

void ddl_relu_bwd(const void* dy, const void* y, void* out, int64_t n,
                  int dtype, hipStream_t s) {
  int g = ew_grid(n);
#define RUN(T) relu_bwd_k<T><<<g, 256, 0, s>>>((const T*)dy, (const T*)y, (T*)out, n)
  switch (dtype) { case 0: RUN(float); break; case 1: RUN(bf16); break; case 2: RUN(f16); break; }
#undef RUN
}

void ddl_check_finite(const void* g_, int64_t n, int dtype, int* found_inf,
                      hipStream_t s) {
  int g = ew_grid(n);
#define RUN(T) check_finite_k<T><<<g, 256, 0, s>>>((const T*)g_, n, found_inf)
  switch (dtype) { case 0: RUN(float); break; case 1: RUN(bf16); break; case 2: RUN(f16); break; }
#undef RUN
}

void ddl_unscale_check(void* g_, int64_t n, int dtype, float inv_scale,
                       int* found_inf, hipStream_t s) {
  int g = ew_grid(n);
#define RUN(T) unscale_check_k<T><<<g, 256, 0, s>>>((T*)g_, n, inv_scale, found_inf)
  switch (dtype) { case 0: RUN(float); break; case 1: RUN(bf16); break; case 2: RUN(f16); break; }
#undef RUN
}

void ddl_dropout_fwd(const void* x, void* y, uint8_t* mask, int64_t n,
                     int dtype, float p, uint64_t seed, hipStream_t s) {
  int g = ew_grid(n, 1);
#define RUN(T) dropout_fwd_k<T><<<g, 256, 0, s>>>((const T*)x, (T*)y, mask, n, p, seed)
  switch (dtype) { case 0: RUN(float); break; case 1: RUN(bf16); break; case 2: RUN(f16); break; }
#undef RUN
}

void ddl_dropout_bwd(const void* dy, void* dx, const uint8_t* mask, int64_t n,
                     int dtype, float p, hipStream_t s) {
  int g = ew_grid(n, 1);
#define RUN(T) dropout_bwd_k<T><<<g, 256, 0, s>>>((const T*)dy, (T*)dx, mask, n, p)
  switch (dtype) { case 0: RUN(float); break; case 1: RUN(bf16); break; case 2: RUN(f16); break; }
#undef RUN
}

void ddl_cast_f32_to(const float* src, void* dst, int64_t n, int dtype,
                     hipStream_t s) {
  int g = ew_grid(n, 4);
#define RUN(T) cast_f32_k<T><<<g, 256, 0, s>>>(src, (T*)dst, n)
  switch (dtype) { case 0: RUN(float); break; case 1: RUN(bf16); break; case 2: RUN(f16); break; }
#undef RUN
}

void ddl_im2col(const void* x, void* out, const int* tbl, int N, int H,
                int W, int C, int R, int S, int P, int Q, int sh, int sw,
                int ph, int pw, int ldout, int dtype, hipStream_t s) {
  int kred_total = R * S * C;
  const int esz = dtype == 0 ? 4 : 2;
  const int epu = 16 / esz;
  // row-staged variant: the block's R input rows must fit 48 KiB LDS and
  // stay 16 B aligned (rowlen % EPU == 0)
  const bool rows_ok = (int64_t)R * W * C * esz <= 48 * 1024 &&
                       (W * C) % epu == 0 && (int64_t)N * P < (1u << 31);
  if (rows_ok) {
    int g = N * P;
#define RUN(T)                                                                 \
  im2col_rows_k<T><<<g, 256, 0, s>>>((const T*)x, (T*)out, tbl, N, H, W, C,   \
                                     R, P, Q, sh, sw, ph, pw, kred_total,     \
                                     ldout)
    switch (dtype) { case 0: RUN(float); break; case 1: RUN(bf16); break; case 2: RUN(f16); break; }
#undef RUN
    return;
  }
  int64_t total = (int64_t)N * P * Q * (ldout / 8);
  int g = (int)std::min<int64_t>(cdiv(total, 256), 8192);
#define RUN(T)                                                                 \
  im2col_k<T><<<g, 256, 0, s>>>((const T*)x, (T*)out, tbl, N, H, W, C, P,     \
                                Q, sh, sw, ph, pw, kred_total, ldout)
  switch (dtype) { case 0: RUN(float); break; case 1: RUN(bf16); break; case 2: RUN(f16); break; }
#undef RUN
}

void ddl_pad_cols(const void* w, void* out, int K, int kred, int ldout,
                  int dtype, hipStream_t s) {
  int g = (int)std::min<int64_t>(cdiv((int64_t)K * ldout, 256), 1024);
#define RUN(T) pad_cols_k<T><<<g, 256, 0, s>>>((const T*)w, (T*)out, K, kred, ldout)
  switch (dtype) { case 0: RUN(float); break; case 1: RUN(bf16); break; case 2: RUN(f16); break; }
#undef RUN
}

void ddl_narrow_cols(const void* w, void* out, int K, int kred, int ldout,
                     int dtype, hipStream_t s) {
  int g = (int)std::min<int64_t>(cdiv((int64_t)K * kred, 256), 1024);
#define RUN(T) narrow_cols_k<T><<<g, 256, 0, s>>>((const T*)w, (T*)out, K, kred, ldout)
  switch (dtype) { case 0: RUN(float); break; case 1: RUN(bf16); break; case 2: RUN(f16); break; }
#undef RUN
}

void ddl_transpose_krsc_crsk(const void* w, void* w_t, int K, int RS, int C,
                             int dtype, hipStream_t s) {
  dim3 grid(cdiv(K, 32) * cdiv(C, 32), RS);
#define RUN(T) transpose_kc_k<T><<<grid, 256, 0, s>>>((const T*)w, (T*)w_t, K, RS, C)
  switch (dtype) { case 0: RUN(float); break; case 1: RUN(bf16); break; case 2: RUN(f16); break; }
#undef RUN
}


void ddl_col2im_acc(const void* dcol, float* dx_ws, const int* tbl, int N,
                    int H, int W, int C, int P, int Q, int sh, int sw,
                    int ph, int pw, int kred_total, int ldout, int dtype,
                    hipStream_t s) {
  const int64_t total = (int64_t)N * P * Q * kred_total;
  const int g = (int)std::min<int64_t>(cdiv(total, 256), 4096);
#define RUN(T)                                                                 \
  col2im_acc_k<T><<<g, 256, 0, s>>>((const T*)dcol, dx_ws, tbl, N, H, W, C,   \
                                    P, Q, sh, sw, ph, pw, kred_total, ldout)
  switch (dtype) { case 0: RUN(float); break; case 1: RUN(bf16); break; case 2: RUN(f16); break; }
#undef RUN
}

}  // extern "C"
