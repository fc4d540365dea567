// K9: fused softmax + cross-entropy (forward: per-row loss + fp32 probs;
// backward: dlogits from probs), with optional label smoothing.
// One 256-thread block per row (C ~ 1000): single pass of loads into
// registers + two block reductions.
#include "common.h"
#include "api.h"

namespace {

template <typename T>
__global__ void softmax_xent_fwd_k(const T* __restrict__ logits,
                                   const int64_t* __restrict__ target,
                                   float* __restrict__ losses,
                                   float* __restrict__ probs, int C,
                                   float ls) {
  __shared__ float scratch[8];
  __shared__ float s_max, s_sum;
  const int n = blockIdx.x;
  const T* row = logits + (int64_t)n * C;
  float* prow = probs + (int64_t)n * C;

  float lmax = -3.4e38f, lsum_logit = 0.f;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    float v = to_f32(row[c]);
    lmax = fmaxf(lmax, v);
    lsum_logit += v;
  }
  // block max
  {
    int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
    float m = wave_reduce_max(lmax);
    if (lane == 0) scratch[wid] = m;
    __syncthreads();
    if (threadIdx.x == 0) {
      float mm = scratch[0];
      for (int i = 1; i < (int)(blockDim.x >> 6); ++i) mm = fmaxf(mm, scratch[i]);
      s_max = mm;
    }
    __syncthreads();
  }
  const float mx = s_max;
  float lsum = 0.f;
  for (int c = threadIdx.x; c < C; c += blockDim.x)
    lsum += __expf(to_f32(row[c]) - mx);
  {
    __syncthreads();
    float t = block_reduce_sum(lsum, scratch);
    if (threadIdx.x == 0) s_sum = t;
    __syncthreads();
  }
  const float inv_sum = 1.f / s_sum;
  const float lse = __logf(s_sum) + mx;
  for (int c = threadIdx.x; c < C; c += blockDim.x)
    prow[c] = __expf(to_f32(row[c]) - mx) * inv_sum;
  // loss = lse - (1-ls)*logit[target] - ls*mean(logits)
  if (threadIdx.x == 0)
    losses[n] = lse - (1.f - ls) * to_f32(row[target[n]]);
  if (ls > 0.f) {
    __syncthreads();
    float t = block_reduce_sum(lsum_logit, scratch);
    if (threadIdx.x == 0) losses[n] -= ls * (t / C);
  }
}

template <typename T>
__global__ void softmax_xent_bwd_k(const float* __restrict__ probs,
                                   const int64_t* __restrict__ target,
                                   T* __restrict__ dlogits, int64_t N, int C,
                                   float ls, const float* __restrict__ scale_p,
                                   float scale_c) {
  const float scale = scale_c * (scale_p ? *scale_p : 1.f);
  const int64_t total = N * C;
  for (int64_t i = global_tid(); i < total; i += grid_stride()) {
    const int64_t n = i / C;
    const int c = (int)(i % C);
    float g = probs[i] - ls / C - (c == (int)target[n] ? (1.f - ls) : 0.f);
    dlogits[i] = from_f32<T>(g * scale);
  }
}

}  // namespace

extern "C" {

void ddl_softmax_xent_fwd(const void* logits, const int64_t* target,
                          float* losses, float* probs, int N, int C,
                          int dtype, float label_smoothing, hipStream_t s) {
#define RUN(T)                                                                 \
  softmax_xent_fwd_k<T><<<N, 256, 0, s>>>((const T*)logits, target, losses,    \
                                          probs, C, label_smoothing)
  switch (dtype) { case 0: RUN(float); break; case 1: RUN(bf16); break; case 2: RUN(f16); break; }
#undef RUN
}

void ddl_softmax_xent_bwd(const float* probs, const int64_t* target,
                          void* dlogits, int N, int C, int dtype,
                          float label_smoothing, const float* scale_ptr,
                          float scale_const, hipStream_t s) {
  int64_t total = (int64_t)N * C;
  int g = (int)std::min<int64_t>(cdiv(total, 256), 2048);
#define RUN(T)                                                                 \
  softmax_xent_bwd_k<T><<<g, 256, 0, s>>>(probs, target, (T*)dlogits,          \
                                          (int64_t)N, C, label_smoothing,      \
                                          scale_ptr, scale_const)
  switch (dtype) { case 0: RUN(float); break; case 1: RUN(bf16); break; case 2: RUN(f16); break; }
#undef RUN
}

}  // extern "C"
