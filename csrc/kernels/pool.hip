// K6/K7: MaxPool (argmax-index fwd, deterministic gather bwd) and global
// average pool over NHWC.  Memory-bound; vectorized over 8-channel groups.
#include "common.h"
#include "api.h"

namespace {

template <typename T>
__global__ void maxpool_fwd_k(const T* __restrict__ x, T* __restrict__ y,
                              uint8_t* __restrict__ idx, int N, int H, int W,
                              int C, int P, int Q, int kh, int kw, int sh,
                              int sw, int ph, int pw) {
  const int ncg = C / 8;
  const int64_t total = (int64_t)N * P * Q * ncg;
  for (int64_t i = global_tid(); i < total; i += grid_stride()) {
    const int cg = (int)(i % ncg);
    int64_t t = i / ncg;
    const int q = (int)(t % Q);
    t /= Q;
    const int p = (int)(t % P);
    const int n = (int)(t / P);
    const int c0 = cg * 8;
    float best[8];
    uint8_t bidx[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) { best[j] = -3.4e38f; bidx[j] = 0; }
    const int h0 = p * sh - ph, w0 = q * sw - pw;
    for (int r = 0; r < kh; ++r) {
      const int h = h0 + r;
      if (h < 0 || h >= H) continue;
      for (int s = 0; s < kw; ++s) {
        const int w = w0 + s;
        if (w < 0 || w >= W) continue;
        Vec8<T> v = load8(x + (((int64_t)n * H + h) * W + w) * C + c0);
        const uint8_t pos = (uint8_t)(r * kw + s);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = to_f32(v.v[j]);
          if (f > best[j]) { best[j] = f; bidx[j] = pos; }
        }
      }
    }
    Vec8<T> out;
#pragma unroll
    for (int j = 0; j < 8; ++j) out.v[j] = from_f32<T>(best[j]);
    const int64_t o = (((int64_t)n * P + p) * Q + q) * C + c0;
    store8(y + o, out);
    *reinterpret_cast<uint2*>(idx + o) = *reinterpret_cast<uint2*>(bidx);
  }
}

// Deterministic gather backward: for each input element, scan the (few)
// output windows that cover it and pick up dy where the argmax matches.
template <typename T>
__global__ void maxpool_bwd_k(const T* __restrict__ dy,
                              const uint8_t* __restrict__ idx,
                              T* __restrict__ dx, int N, int H, int W, int C,
                              int P, int Q, int kh, int kw, int sh, int sw,
                              int ph, int pw) {
  const int ncg = C / 8;
  const int64_t total = (int64_t)N * H * W * ncg;
  for (int64_t i = global_tid(); i < total; i += grid_stride()) {
    const int cg = (int)(i % ncg);
    int64_t t = i / ncg;
    const int w = (int)(t % W);
    t /= W;
    const int h = (int)(t % H);
    const int n = (int)(t / H);
    const int c0 = cg * 8;
    float acc[8] = {0};
    // output rows p with p*sh - ph <= h <= p*sh - ph + kh-1
    const int plo = max(0, (h + ph - kh + sh) / sh);  // ceil((h+ph-kh+1)/sh)
    const int phi = min(P - 1, (h + ph) / sh);
    const int qlo = max(0, (w + pw - kw + sw) / sw);
    const int qhi = min(Q - 1, (w + pw) / sw);
    for (int p = plo; p <= phi; ++p) {
      const int r = h - (p * sh - ph);
      for (int q = qlo; q <= qhi; ++q) {
        const int s = w - (q * sw - pw);
        const uint8_t pos = (uint8_t)(r * kw + s);
        const int64_t o = (((int64_t)n * P + p) * Q + q) * C + c0;
        uint2 raw = *reinterpret_cast<const uint2*>(idx + o);
        const uint8_t* bidx = reinterpret_cast<const uint8_t*>(&raw);
        Vec8<T> v = load8(dy + o);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (bidx[j] == pos) acc[j] += to_f32(v.v[j]);
      }
    }
    Vec8<T> out;
#pragma unroll
    for (int j = 0; j < 8; ++j) out.v[j] = from_f32<T>(acc[j]);
    store8(dx + (((int64_t)n * H + h) * W + w) * C + c0, out);
  }
}

// Global average pool: y[n,c] = mean_hw x[n,hw,c].  Block = 256 threads =
// 8 vec-groups x 32 rows over one (n, 64-channel strip).
template <typename T>
__global__ void gap_fwd_k(const T* __restrict__ x, T* __restrict__ y, int HW,
                          int C, float inv_hw) {
  __shared__ float red[32][64];
  const int vg = threadIdx.x % 8;
  const int trow = threadIdx.x / 8;
  const int n = blockIdx.y;
  const int c0 = blockIdx.x * 64 + vg * 8;
  if (c0 >= C) return;
  const T* xp = x + (int64_t)n * HW * C + c0;
  float s[8] = {0};
  for (int m = trow; m < HW; m += 32) {
    Vec8<T> v = load8(xp + (int64_t)m * C);
#pragma unroll
    for (int j = 0; j < 8; ++j) s[j] += to_f32(v.v[j]);
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) red[trow][vg * 8 + j] = s[j];
  __syncthreads();
#pragma unroll
  for (int step = 16; step > 0; step >>= 1) {
    if (trow < step) {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        red[trow][vg * 8 + j] += red[trow + step][vg * 8 + j];
    }
    __syncthreads();
  }
  if (trow == 0) {
    Vec8<T> out;
#pragma unroll
    for (int j = 0; j < 8; ++j) out.v[j] = from_f32<T>(red[0][vg * 8 + j] * inv_hw);
    store8(y + (int64_t)n * C + c0, out);
  }
}

template <typename T>
__global__ void gap_bwd_k(const T* __restrict__ dy, T* __restrict__ dx,
                          int64_t N, int HW, int C, float inv_hw) {
  const int ncg = C / 8;
  const int64_t total = N * HW * ncg;
  for (int64_t i = global_tid(); i < total; i += grid_stride()) {
    const int cg = (int)(i % ncg);
    const int64_t nm = i / ncg;
    const int64_t n = nm / HW;
    Vec8<T> v = load8(dy + n * C + cg * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) v.v[j] = from_f32<T>(to_f32(v.v[j]) * inv_hw);
    store8(dx + nm * C + cg * 8, v);
  }
}

}  // namespace

extern "C" {

void ddl_maxpool_fwd(const void* x, void* y, uint8_t* idx, int N, int H,
                     int W, int C, int P, int Q, int kh, int kw, int sh,
                     int sw, int ph, int pw, int dtype, hipStream_t s) {
  int64_t total = (int64_t)N * P * Q * (C / 8);
  int g = (int)std::min<int64_t>(cdiv(total, 256), 4096);
#define RUN(T)                                                                 \
  maxpool_fwd_k<T><<<g, 256, 0, s>>>((const T*)x, (T*)y, idx, N, H, W, C, P,  \
                                     Q, kh, kw, sh, sw, ph, pw)
  switch (dtype) { case 0: RUN(float); break; case 1: RUN(bf16); break; case 2: RUN(f16); break; }
#undef RUN
}

void ddl_maxpool_bwd(const void* dy, const uint8_t* idx, void* dx, int N,
                     int H, int W, int C, int P, int Q, int kh, int kw,
                     int sh, int sw, int ph, int pw, int dtype, hipStream_t s) {
  int64_t total = (int64_t)N * H * W * (C / 8);
  int g = (int)std::min<int64_t>(cdiv(total, 256), 4096);
#define RUN(T)                                                                 \
  maxpool_bwd_k<T><<<g, 256, 0, s>>>((const T*)dy, idx, (T*)dx, N, H, W, C,   \
                                     P, Q, kh, kw, sh, sw, ph, pw)
  switch (dtype) { case 0: RUN(float); break; case 1: RUN(bf16); break; case 2: RUN(f16); break; }
#undef RUN
}

void ddl_gap_fwd(const void* x, void* y, int N, int HW, int C, int dtype,
                 hipStream_t s) {
  dim3 g(cdiv(C, 64), N);
#define RUN(T) gap_fwd_k<T><<<g, 256, 0, s>>>((const T*)x, (T*)y, HW, C, 1.f / HW)
  switch (dtype) { case 0: RUN(float); break; case 1: RUN(bf16); break; case 2: RUN(f16); break; }
#undef RUN
}

void ddl_gap_bwd(const void* dy, void* dx, int N, int HW, int C, int dtype,
                 hipStream_t s) {
  int64_t total = (int64_t)N * HW * (C / 8);
  int g = (int)std::min<int64_t>(cdiv(total, 256), 4096);
#define RUN(T) gap_bwd_k<T><<<g, 256, 0, s>>>((const T*)dy, (T*)dx, N, HW, C, 1.f / HW)
  switch (dtype) { case 0: RUN(float); break; case 1: RUN(bf16); break; case 2: RUN(f16); break; }
#undef RUN
}

}  // extern "C"
