// K10/K11: fused multi-tensor SGD+momentum / LARS update.
// One launch updates the whole model: the chunk table (built once on the
// Python side — gradient pointers are stable views into the persistent flat
// buckets) has one row per block: [param_ptr, grad_ptr, master_ptr, mom_ptr,
// numel, use_lars, tensor_idx].  fp32 master + momentum under bf16/fp16
// params; folds the 1/loss_scale unscale and the LARS trust ratio in.
// Reference: Horovod leaves this to framework foreach ops; a single pass
// here is HBM-bound at ~22 B/element.
#include "common.h"
#include "api.h"

namespace {

struct Row {
  int64_t p, g, m, v, n, lars, ti;
};

template <typename T>
__global__ void fused_sgd_k(const int64_t* table, const float* norms,
                            int ntensors, float lr, float momentum, float wd,
                            int nesterov, float lars_eta, float lars_eps,
                            float inv_scale) {
  const Row row = *reinterpret_cast<const Row*>(table + (int64_t)blockIdx.x * 7);
  T* param = reinterpret_cast<T*>(row.p);
  const T* grad = reinterpret_cast<const T*>(row.g);
  float* master = reinterpret_cast<float*>(row.m);
  float* mom = reinterpret_cast<float*>(row.v);
  const int n = (int)row.n;

  float ratio = 1.f;
  if (row.lars) {
    float wn = sqrtf(norms[row.ti]);
    float gn = sqrtf(norms[ntensors + row.ti]);
    if (wn > 0.f && gn > 0.f) ratio = lars_eta * wn / (gn + lars_eps);
  }

  for (int i = threadIdx.x * 4; i < n; i += blockDim.x * 4) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int k = i + j;
      if (k >= n) break;
      float w = master[k];
      float g = to_f32(grad[k]) * inv_scale + wd * w;
      g *= ratio;
      if (momentum != 0.f) {
        float m = mom[k] * momentum + g;
        mom[k] = m;
        g = nesterov ? g + momentum * m : m;
      }
      w -= lr * g;
      master[k] = w;
      param[k] = from_f32<T>(w);
    }
  }
}

// Per-chunk partial squared norms -> partials[2][nrows] (deterministic,
// no atomics): sw = sum(master^2), sg = sum(g'^2) with
// g' = grad*inv_scale + wd*master (the effective gradient LARS scales).
template <typename T>
__global__ void lars_norms_k(const int64_t* table, float* partials, int nrows,
                             float wd, float inv_scale) {
  __shared__ float scratch[8];
  const Row row = *reinterpret_cast<const Row*>(table + (int64_t)blockIdx.x * 7);
  float sw = 0.f, sg = 0.f;
  if (row.lars) {
    const T* grad = reinterpret_cast<const T*>(row.g);
    const float* master = reinterpret_cast<const float*>(row.m);
    const int n = (int)row.n;
    for (int i = threadIdx.x; i < n; i += blockDim.x) {
      float w = master[i];
      float g = to_f32(grad[i]) * inv_scale + wd * w;
      sw += w * w;
      sg += g * g;
    }
  }
  sw = block_reduce_sum(sw, scratch);
  __syncthreads();
  sg = block_reduce_sum(sg, scratch);
  if (threadIdx.x == 0) {
    partials[blockIdx.x] = sw;
    partials[nrows + blockIdx.x] = sg;
  }
}

// Fold the per-chunk partials into per-tensor norms in fixed chunk order:
// one block per tensor scans the (tensor-sorted) rows.
__global__ void lars_fold_norms_k(const int64_t* table,
                                  const float* partials, float* norms,
                                  int nrows, int ntensors) {
  const int ti = blockIdx.x;
  if (threadIdx.x != 0) return;
  float sw = 0.f, sg = 0.f;
  for (int r = 0; r < nrows; ++r) {
    const Row row = *reinterpret_cast<const Row*>(table + (int64_t)r * 7);
    if ((int)row.ti == ti && row.lars) {
      sw += partials[r];
      sg += partials[nrows + r];
    }
  }
  norms[ti] = sw;
  norms[ntensors + ti] = sg;
}

}  // namespace

extern "C" {

void ddl_fused_sgd(const int64_t* table, int nrows, const float* norms,
                   int ntensors, int dtype, float lr, float momentum,
                   float wd, int nesterov, float lars_eta, float lars_eps,
                   float inv_scale, hipStream_t s) {
#define RUN(T)                                                                 \
  fused_sgd_k<T><<<nrows, 256, 0, s>>>(table, norms, ntensors, lr, momentum,   \
                                       wd, nesterov, lars_eta, lars_eps,       \
                                       inv_scale)
  switch (dtype) { case 0: RUN(float); break; case 1: RUN(bf16); break; case 2: RUN(f16); break; }
#undef RUN
}

void ddl_lars_norms(const int64_t* table, int nrows, float* norms,
                    float* partials, int ntensors, int dtype, float wd,
                    float inv_scale, hipStream_t s) {
#define RUN(T)                                                                 \
  lars_norms_k<T><<<nrows, 256, 0, s>>>(table, partials, nrows, wd, inv_scale)
  switch (dtype) { case 0: RUN(float); break; case 1: RUN(bf16); break; case 2: RUN(f16); break; }
#undef RUN
  lars_fold_norms_k<<<ntensors, 32, 0, s>>>(table, partials, norms, nrows,
                                            ntensors);
}

}  // extern "C"
