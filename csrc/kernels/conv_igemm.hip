// K1 (forward) and K2 (dgrad) implicit-GEMM NHWC convolution for gfx950.
//
// Forward:  y[m=(n,p,q)][k] = sum_{r,s,c} x[n, p*sh+r-ph, q*sw+s-pw, c] *
//                                         w[k,r,s,c]            (+bias,relu)
// Dgrad:    dx[m=(n,h,w)][c] = sum_{r,s,k} dy[n,(h+ph-r)/sh,(w+pw-s)/sw,k] *
//                                          w[k,r,s,c]   (terms where /sh,/sw
//                                          divide exactly)
// Both are the same GEMM pipeline over the linearized (r,s,c-tile) reduce
// sequence: double-buffered swizzled LDS tiles with register staging split
// issue-early/write-late (guide §6 G15 T14: next tile's global loads issue
// before the MFMA section, the LDS write lands after it, one barrier per
// tile), 128x128x64 block tile, 4 waves, v_mfma_f32_16x16x32_{bf16,f16}.
// Padding is handled by per-row validity masks (zero-fill); dgrad consumes
// the CRSK-transposed weight (ddl_transpose_krsc_crsk) so its B-tile loads
// are contiguous.  MODE_FLAT flattens the reduce dim to R*S*C with scalar
// gather for C % 8 != 0 (the 7x7 C=3 stem).
#include "conv_tile.h"
#include "api.h"

namespace {

enum Mode { FWD = 0, DGRAD = 1, FWD_FLAT = 2 };

// DBUF=true: double-buffered LDS (64 KiB, 2 blocks/CU) with loads issued
// under the MFMAs — best for deep reduce loops.  DBUF=false: single buffer
// (32 KiB, ~4 blocks/CU) with issue-early loads and two barriers per tile —
// the shallow-reduce 56x56/64-channel layers are LATENCY-bound (their whole
// K-loop is 1-9 tiles of ~600 MFMA cycles), so occupancy beats pipelining.
template <typename T, int MODE, bool DBUF>
__global__ __launch_bounds__(256) void conv_igemm_k(
    const T* __restrict__ Asrc, const T* __restrict__ Bsrc,
    const T* __restrict__ bias, T* __restrict__ out, ConvDims d, int64_t M,
    int Nc, int relu) {
  __shared__ __align__(16) char smem[DBUF ? 4 * TILE_BYTES : 2 * TILE_BYTES];
  char* const lA[2] = {smem, DBUF ? smem + 2 * TILE_BYTES : smem};
  char* const lB[2] = {smem + TILE_BYTES,
                       DBUF ? smem + 3 * TILE_BYTES : smem + TILE_BYTES};

  const int nt_n = (Nc + BN - 1) / BN;
  const unsigned bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int64_t tile_m = (int64_t)(bid / nt_n) * BM;
  const int tile_n = (int)(bid % nt_n) * BN;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = (wid >> 1) * 64;
  const int wc = (wid & 1) * 64;
  const int trow = threadIdx.x >> 3;
  const int tcol = threadIdx.x & 7;

  f32x4_t acc[4][4] = {};

  // per-thread A-row decode (4 staged rows)
  int a_n[4], a_i1[4], a_i2[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int64_t m = tile_m + trow + 32 * i;
    if (m < M) {
      if (MODE == DGRAD) {
        a_i2[i] = (int)(m % d.W);
        a_i1[i] = (int)((m / d.W) % d.H);
        a_n[i] = (int)(m / ((int64_t)d.W * d.H));
      } else {
        a_i2[i] = (int)(m % d.Q);
        a_i1[i] = (int)((m / d.Q) % d.P);
        a_n[i] = (int)(m / ((int64_t)d.Q * d.P));
      }
    } else {
      a_n[i] = -1;
    }
  }

  const int Cred = (MODE == DGRAD) ? d.K : d.C;
  const int64_t a_ld = (MODE == DGRAD) ? d.K : d.C;
  const int b_ld = (MODE == DGRAD) ? d.K : d.C;
  const int rs_total = (MODE == FWD_FLAT) ? 1 : d.R * d.S;
  const int kred_total = (MODE == FWD_FLAT) ? d.R * d.S * d.C : Cred;
  const int nct = (kred_total + BK - 1) / BK;
  const int total_tiles = rs_total * nct;

  int cur_rs = -1;
  const T* a_ptr[4];
  const T* b_ptr[4];

  // ---- per-tile register staging (issue-early half) ----------------------
  uint4 ra[4], rb[4];
  auto load_tile = [&](int t) {
    const int rs = t / nct;
    const int c0 = (t % nct) * BK;
    const int r = rs / d.S, s = rs % d.S;
    if (rs != cur_rs) {
      cur_rs = rs;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        a_ptr[i] = nullptr;
        if (a_n[i] >= 0) {
          if (MODE == FWD) {
            const int hi = a_i1[i] * d.sh + r - d.ph;
            const int wi = a_i2[i] * d.sw + s - d.pw;
            if (hi >= 0 && hi < d.H && wi >= 0 && wi < d.W)
              a_ptr[i] = Asrc + (((int64_t)a_n[i] * d.H + hi) * d.W + wi) * a_ld;
          } else if (MODE == DGRAD) {
            const int hn = a_i1[i] + d.ph - r;
            const int wn = a_i2[i] + d.pw - s;
            if (hn >= 0 && wn >= 0 && hn % d.sh == 0 && wn % d.sw == 0) {
              const int ho = hn / d.sh, wo = wn / d.sw;
              if (ho < d.P && wo < d.Q)
                a_ptr[i] = Asrc + (((int64_t)a_n[i] * d.P + ho) * d.Q + wo) * a_ld;
            }
          } else {
            a_ptr[i] = Asrc;
          }
        }
      }
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int row_n = tile_n + trow + 32 * i;
        b_ptr[i] = nullptr;
        if (row_n < Nc) {
          if (MODE == FWD_FLAT)
            b_ptr[i] = Bsrc + (int64_t)row_n * kred_total;
          else
            b_ptr[i] = Bsrc + ((int64_t)row_n * d.R + r) * d.S * b_ld +
                       (int64_t)s * b_ld;
        }
      }
    }
    const int cel = c0 + tcol * 8;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      ra[i] = zero16();
      if (MODE == FWD_FLAT) {
        if (a_n[i] >= 0) {
          uint16_t* e = reinterpret_cast<uint16_t*>(&ra[i]);
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            const int kred = cel + j;
            if (kred < kred_total) {
              const int rr = kred / (d.S * d.C);
              const int rem = kred % (d.S * d.C);
              const int ss = rem / d.C;
              const int cc = rem % d.C;
              const int hi = a_i1[i] * d.sh + rr - d.ph;
              const int wi = a_i2[i] * d.sw + ss - d.pw;
              if (hi >= 0 && hi < d.H && wi >= 0 && wi < d.W)
                e[j] = *reinterpret_cast<const uint16_t*>(
                    Asrc + (((int64_t)a_n[i] * d.H + hi) * d.W + wi) * d.C + cc);
            }
          }
        }
      } else if (a_ptr[i] != nullptr && cel + 8 <= Cred) {
        ra[i] = *reinterpret_cast<const uint4*>(a_ptr[i] + cel);
      }
    }
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      rb[i] = zero16();
      if (b_ptr[i] != nullptr) {
        if (MODE == FWD_FLAT) {
          uint16_t* e = reinterpret_cast<uint16_t*>(&rb[i]);
#pragma unroll
          for (int j = 0; j < 8; ++j)
            if (cel + j < kred_total)
              e[j] = *reinterpret_cast<const uint16_t*>(b_ptr[i] + cel + j);
        } else if (cel + 8 <= Cred) {
          rb[i] = *reinterpret_cast<const uint4*>(b_ptr[i] + cel);
        }
      }
    }
  };

  auto write_tile = [&](char* la, char* lb) {
#pragma unroll
    for (int i = 0; i < 4; ++i)
      lds_write16(la, trow + 32 * i, tcol * 16, ra[i]);
#pragma unroll
    for (int i = 0; i < 4; ++i)
      lds_write16(lb, trow + 32 * i, tcol * 16, rb[i]);
  };

  const int frow = lane & 15;
  const int fcolb = (lane >> 4) * 16;
  auto mfma_tile = [&](const char* la, const char* lb) {
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      typename MfmaFrag<T>::type a[4], b[4];
#pragma unroll
      for (int mr = 0; mr < 4; ++mr) {
        uint4 v = lds_read16(la, wr + mr * 16 + frow, kk * 64 + fcolb);
        a[mr] = *reinterpret_cast<typename MfmaFrag<T>::type*>(&v);
      }
#pragma unroll
      for (int nr = 0; nr < 4; ++nr) {
        uint4 v = lds_read16(lb, wc + nr * 16 + frow, kk * 64 + fcolb);
        b[nr] = *reinterpret_cast<typename MfmaFrag<T>::type*>(&v);
      }
#pragma unroll
      for (int mr = 0; mr < 4; ++mr)
#pragma unroll
        for (int nr = 0; nr < 4; ++nr)
          acc[mr][nr] = mfma_16x16x32<T>(a[mr], b[nr], acc[mr][nr]);
    }
  };

  if (DBUF) {
    // ---- pipelined main loop: one barrier per tile -----------------------
    load_tile(0);
    write_tile(lA[0], lB[0]);
    __syncthreads();
    for (int t = 0; t < total_tiles; ++t) {
      const int cur = t & 1;
      const bool more = (t + 1) < total_tiles;
      if (more) load_tile(t + 1);               // global loads in flight...
      char* la = cur ? lA[1] : lA[0];
      char* lb = cur ? lB[1] : lB[0];
      mfma_tile(la, lb);                        // ...hidden under the MFMAs
      if (more) write_tile(cur ? lA[0] : lA[1], cur ? lB[0] : lB[1]);
      __syncthreads();
    }
  } else {
    // ---- single buffer, issue-early loads, two barriers per tile ---------
    load_tile(0);
    for (int t = 0; t < total_tiles; ++t) {
      write_tile(lA[0], lB[0]);
      __syncthreads();
      if (t + 1 < total_tiles) load_tile(t + 1);  // overlap with MFMAs
      mfma_tile(lA[0], lB[0]);
      __syncthreads();
    }
  }

  // ---- epilogue through LDS: repack D fragments ([4 consecutive m] x
  // [1 n] per lane) into a [128][128] element tile, then store coalesced
  // 16B rows.  Direct fragment stores are 64 scalar 2B globals per lane —
  // issue-bound (guide T21 diagnostic) and dominant for shallow layers.
  __syncthreads();  // done with the operand tiles; reuse smem
  T* etile = reinterpret_cast<T*>(smem);  // [BM][BN] elements (32 KiB)
  const int en = lane & 15;
  const int em = (lane >> 4) * 4;
#pragma unroll
  for (int nr = 0; nr < 4; ++nr) {
    const int n_loc = wc + nr * 16 + en;
    const float bv = (MODE != DGRAD && bias != nullptr && tile_n + n_loc < Nc)
                         ? to_f32(bias[tile_n + n_loc]) : 0.f;
#pragma unroll
    for (int mr = 0; mr < 4; ++mr) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        float f = acc[mr][nr][reg] + bv;
        if (relu) f = fmaxf(f, 0.f);
        etile[(wr + mr * 16 + em + reg) * BN + n_loc] = from_f32<T>(f);
      }
    }
  }
  __syncthreads();
  // 256 threads x 8 iterations of 16B: row-major sweep of the 128x128 tile
  const int nvalid = Nc - tile_n;  // columns of this tile in range
#pragma unroll
  for (int it = 0; it < 8; ++it) {
    const int unit = threadIdx.x + it * 256;   // 16B unit index (2048 total)
    const int row = unit >> 4;                 // BN*2/16 = 16 units per row
    const int ub = (unit & 15) * 8;            // first element in row
    const int64_t m = tile_m + row;
    if (m < M && ub < nvalid) {
      *reinterpret_cast<uint4*>(out + m * Nc + tile_n + ub) =
          *reinterpret_cast<const uint4*>(etile + row * BN + ub);
    }
  }
}

bool conv_use_dbuf(int cred, int rs_total) {
  const char* env = getenv("DDL_CONV_DBUF");
  if (env) return env[0] == '1';
  // measured (bench ladder): the 32 KiB single-buffer variant's extra
  // occupancy (4 blocks/CU vs 2) beats the double-buffer pipeline at every
  // ResNet/VGG shape on gfx950 — these kernels are latency-bound, not
  // bandwidth-bound (PMC: MfmaUtil 19%, VALUBusy 31%).
  (void)cred; (void)rs_total;
  return false;
}

}  // namespace

extern "C" {

void ddl_conv_fwd(const void* x, const void* w, const void* bias, void* y,
                  int N, int H, int W, int C, int K, int R, int S, int P,
                  int Q, int sh, int sw, int ph, int pw, int relu, int dtype,
                  hipStream_t s) {
  ConvDims d{N, H, W, C, K, R, S, P, Q, sh, sw, ph, pw};
  const int64_t M = (int64_t)N * P * Q;
  const int Nc = K;
  const int64_t grid = cdiv(M, BM) * cdiv(Nc, BN);
  const bool flat = (C % 8) != 0;
  const bool dbuf = conv_use_dbuf(flat ? R * S * C : C, flat ? 1 : R * S);
#define RUN(T, MODE, DB)                                                       \
  conv_igemm_k<T, MODE, DB><<<(unsigned)grid, 256, 0, s>>>(                    \
      (const T*)x, (const T*)w, (const T*)bias, (T*)y, d, M, Nc, relu)
#define SEL(T, MODE)                                                           \
  if (dbuf) RUN(T, MODE, true); else RUN(T, MODE, false)
  if (dtype == 1) { if (flat) { SEL(bf16, FWD_FLAT); } else { SEL(bf16, FWD); } }
  else if (dtype == 2) { if (flat) { SEL(f16, FWD_FLAT); } else { SEL(f16, FWD); } }
#undef SEL
#undef RUN
}

void ddl_conv_dgrad(const void* dy, const void* w_t, void* dx, int N, int H,
                    int W, int C, int K, int R, int S, int P, int Q, int sh,
                    int sw, int ph, int pw, int dtype, hipStream_t s) {
  ConvDims d{N, H, W, C, K, R, S, P, Q, sh, sw, ph, pw};
  const int64_t M = (int64_t)N * H * W;
  const int Nc = C;
  const int64_t grid = cdiv(M, BM) * cdiv(Nc, BN);
  const bool dbuf = conv_use_dbuf(K, R * S);
#define RUN(T, DB)                                                             \
  conv_igemm_k<T, DGRAD, DB><<<(unsigned)grid, 256, 0, s>>>(                   \
      (const T*)dy, (const T*)w_t, nullptr, (T*)dx, d, M, Nc, 0)
  if (dtype == 1) { if (dbuf) RUN(bf16, true); else RUN(bf16, false); }
  else if (dtype == 2) { if (dbuf) RUN(f16, true); else RUN(f16, false); }
#undef RUN
}

}  // extern "C"
