// K1 (forward) and K2 (dgrad) implicit-GEMM NHWC convolution for gfx950.
//
// Forward:  y[m=(n,p,q)][k] = sum_{r,s,c} x[n, p*sh+r-ph, q*sw+s-pw, c] *
//                                         w[k,r,s,c]            (+bias,relu)
// Dgrad:    dx[m=(n,h,w)][c] = sum_{r,s,k} dy[n,(h+ph-r)/sh,(w+pw-s)/sw,k] *
//                                          w[k,r,s,c]   (terms where /sh,/sw
//                                          divide exactly)
// Both are the same GEMM pipeline over the linearized (r,s,c-tile) reduce
// sequence: 128x128x64 block tile, 4 waves, v_mfma_f32_16x16x32_{bf16,f16},
// operand tiles staged by global_load_lds (async HBM->LDS DMA, no VGPR
// round-trip — guide §5 step 3; +67% over register staging on the plain
// GEMM ladder).  glds writes lane-linear, so the bank swizzle lives on the
// per-lane SOURCE address (rule 21): column unit cu -> cu ^ (row & 7),
// giving <=2-way b128 read conflicts (guide G4).  Padding rows read a
// device zero page (glds has no predication).  MODE_FLAT (C % 8 != 0, the
// 7x7 C=3 stem) keeps register staging with per-element scalar gather.
#include "conv_tile.h"
#include "api.h"

namespace {

enum Mode { FWD = 0, DGRAD = 1, FWD_FLAT = 2, DGRAD_S2 = 3 };

// Strided-dgrad phase parameters: output rows h = hh*sh + ph2 (w alike)
// only receive taps r = r0 + i*sh, making every gather row dense (a
// non-phased stride-2 dgrad wastes 3/4 of its MFMA work on masked rows —
// profiles/conv_shape_tflops.md).
struct DgradPhase {
  int ph2, pw2;      // this phase's (h % sh, w % sw)
  int r0, s0;        // first valid tap per axis
  int Sp;            // valid taps along s
  int Hp, Wp;        // phase subgrid dims
  FastDiv fdWp, fdHpWp;
};

__device__ __align__(256) const unsigned char ddl_zero_page[256] = {0};

typedef __attribute__((address_space(1))) const unsigned int as1_uint;
typedef __attribute__((address_space(3))) unsigned int as3_uint;

__device__ __forceinline__ void glds16(const void* gsrc, void* lds_base) {
  __builtin_amdgcn_global_load_lds((as1_uint*)gsrc, (as3_uint*)lds_base, 16,
                                   0, 0);
}

// stats_ws (FWD, 2-byte dtypes only): per-m-tile BN partials
// [cdiv(M,BM)][2][Nc] {sum, sumsq} of the fp32 accumulators — the following
// BatchNorm then skips its full re-read of y (bn_stats_k) and just folds
// these (stats of the pre-bf16-rounding values; the difference is bf16
// rounding noise, orders below BN's eps).
// TBM/TBN: block tile.  (128,128) = the standard 2x2-wave tile; (256,64)
// puts the 4 waves in a 4x1 m-column over a 64-wide n-tile for Nc <= 64
// outputs (ResNet layer1 / the im2col'd stem: a 128-wide n-tile there runs
// 50% dead MFMA work).  SBUF only (A tile is 32 KiB at TBM=256).
// bn_bwd fusion (DGRAD modes, 2-byte dtypes): when bstats_ws != nullptr the
// epilogue also emits the PRECEDING BatchNorm's backward partials
// [chunk][2][Nc] {sum(g*xhat), sum(g)} with g = this dgrad's output masked
// by the BN's recomputed ReLU (w*xhat+b > 0, relu_mode 2) and
// xhat = (bnx - mean)*rstd from the BN's own input — that BN then skips
// its bn_bwd_stats full (dy, x) read pass (the backward analog of the
// conv_fwd_stats fusion; docs/ROADMAP.md #8).
template <typename T, int MODE, bool DBUF, int TBM = 128, int TBN = 128,
          bool BSTATS = false>
__global__ __launch_bounds__(256) void conv_igemm_k(
    const T* __restrict__ Asrc, const T* __restrict__ Bsrc,
    const T* __restrict__ bias, T* __restrict__ out, ConvDims d, int64_t M,
    int Nc, int relu, DgradPhase gp, float* __restrict__ stats_ws,
    const T* __restrict__ accum, const T* __restrict__ bnx = nullptr,
    const float* __restrict__ bn_mean = nullptr,
    const float* __restrict__ bn_rstd = nullptr,
    const T* __restrict__ bn_w = nullptr,
    const T* __restrict__ bn_b = nullptr,
    float* __restrict__ bstats_ws = nullptr, int bchunk_base = 0,
    const uint8_t* __restrict__ bn_mask = nullptr) {
  constexpr int AROWS = TBM / 32;     // staged A rows per thread
  constexpr int BROWS = TBN / 32;     // staged B rows per thread
  constexpr int ATB = TBM * 128;      // A tile bytes (128 B rows)
  constexpr int BTB = TBN * 128;      // B tile bytes
  static_assert(!(DBUF && TBM != 128), "DBUF only for the 128^2 tile");
  __shared__ __align__(16) char smem[DBUF ? 2 * (ATB + BTB) : ATB + BTB];
  char* const lA[2] = {smem, DBUF ? smem + ATB + BTB : smem};
  char* const lB[2] = {smem + ATB,
                       DBUF ? smem + 2 * ATB + BTB : smem + ATB};

  const int nt_n = (Nc + TBN - 1) / TBN;
  const unsigned bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int64_t tile_m = (int64_t)(bid / nt_n) * TBM;
  const int tile_n = (int)(bid % nt_n) * TBN;

  const int lane = threadIdx.x & 63;
  const int wid = __builtin_amdgcn_readfirstlane(threadIdx.x >> 6);
  const int wr = (TBN == 64) ? (threadIdx.x >> 6) * 64
                             : ((threadIdx.x >> 6) >> 1) * 64;
  const int wc = (TBN == 64) ? 0 : ((threadIdx.x >> 6) & 1) * 64;
  const int trow = threadIdx.x >> 3;   // this thread's staging row base
  const int tcol = threadIdx.x & 7;    // 16B unit within the row

  f32x4_t acc[4][4] = {};

  // per-thread A-row decode (AROWS staged rows: trow + 32*i)
  int a_n[AROWS], a_i1[AROWS], a_i2[AROWS];
#pragma unroll
  for (int i = 0; i < AROWS; ++i) {
    int64_t m = tile_m + trow + 32 * i;
    if (m < M) {
      if (MODE == DGRAD_S2) {
        // m indexes the phase subgrid; store ACTUAL (h, w)
        const uint32_t mw = fd_div((uint32_t)m, gp.fdWp);
        a_i2[i] = (int)((uint32_t)m - mw * gp.Wp) * d.sw + gp.pw2;
        const uint32_t n = fd_div((uint32_t)m, gp.fdHpWp);
        a_i1[i] = (int)(mw - n * gp.Hp) * d.sh + gp.ph2;
        a_n[i] = (int)n;
      } else if (MODE == DGRAD) {
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

  const bool isdg = (MODE == DGRAD || MODE == DGRAD_S2);
  const int Cred = isdg ? d.K : d.C;
  const int64_t a_ld = isdg ? d.K : d.C;
  const int b_ld = isdg ? d.K : d.C;
  constexpr int BKT = ConvElems<T>::BKT;  // reduce elements per LDS tile
  const int Rp = (MODE == DGRAD_S2) ? (d.R - gp.r0 + d.sh - 1) / d.sh : d.R;
  const int Sp = (MODE == DGRAD_S2) ? gp.Sp : d.S;
  const int rs_total = (MODE == FWD_FLAT) ? 1 : Rp * Sp;
  const int kred_total = (MODE == FWD_FLAT) ? d.R * d.S * d.C : Cred;
  const int nct = (kred_total + BKT - 1) / BKT;
  const int total_tiles = rs_total * nct;

  int cur_rs = -1;
  const T* a_ptr[AROWS];
  const T* b_ptr[BROWS];
  const T* zpad = reinterpret_cast<const T*>(ddl_zero_page);

  auto update_rs_ptrs = [&](int rs) {
    int r, s;
    if (MODE == DGRAD_S2) {
      r = gp.r0 + (rs / Sp) * d.sh;
      s = gp.s0 + (rs % Sp) * d.sw;
    } else {
      r = rs / d.S;
      s = rs % d.S;
    }
    cur_rs = rs;
#pragma unroll
    for (int i = 0; i < AROWS; ++i) {
      a_ptr[i] = nullptr;
      if (a_n[i] >= 0) {
        if (MODE == FWD) {
          const int hi = a_i1[i] * d.sh + r - d.ph;
          const int wi = a_i2[i] * d.sw + s - d.pw;
          if (hi >= 0 && hi < d.H && wi >= 0 && wi < d.W)
            a_ptr[i] = Asrc + (((int64_t)a_n[i] * d.H + hi) * d.W + wi) * a_ld;
        } else if (MODE == DGRAD_S2) {
          // phase construction guarantees hn % sh == 0 and wn % sw == 0
          const int hn = a_i1[i] + d.ph - r;
          const int wn = a_i2[i] + d.pw - s;
          if (hn >= 0 && wn >= 0) {
            const int ho = hn / d.sh, wo = wn / d.sw;
            if (ho < d.P && wo < d.Q)
              a_ptr[i] = Asrc + (((int64_t)a_n[i] * d.P + ho) * d.Q + wo) * a_ld;
          }
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
    for (int i = 0; i < BROWS; ++i) {
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
  };

  // ---- glds staging (FWD/DGRAD): one async 16B DMA per thread per row ----
  // Lane-linear LDS image; source column unit XOR'd with (row&7) so the
  // b128 fragment reads see the swizzled layout (rule 21).
  constexpr int EPU = ConvElems<T>::PER_UNIT;  // elements per 16 B unit
  const int scu = tcol ^ (trow & 7);   // this thread's swizzled source unit
  // (rs, c0) are carried by the caller as counters — a per-tile t/nct
  // integer division costs ~40 VALU cycles/tile and showed up in PMC.
  auto stage_glds = [&](int rs, int c0, char* la, char* lb) {
    if (rs != cur_rs) update_rs_ptrs(rs);
    const int cel = c0 + scu * EPU;
    const bool cok = cel + EPU <= Cred;
#pragma unroll
    for (int i = 0; i < AROWS; ++i) {
      const T* src = (a_ptr[i] != nullptr && cok) ? a_ptr[i] + cel : zpad;
      glds16(src, la + wid * 1024 + i * 4096);
    }
#pragma unroll
    for (int i = 0; i < BROWS; ++i) {
      const T* src = (b_ptr[i] != nullptr && cok) ? b_ptr[i] + cel : zpad;
      glds16(src, lb + wid * 1024 + i * 4096);
    }
  };

  // ---- register staging (FWD_FLAT only: scalar gather) -------------------
  uint4 ra[4], rb[4];
  auto load_tile_flat = [&](int c0) {
    if (cur_rs != 0) update_rs_ptrs(0);
    const int cel = c0 + tcol * EPU;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      ra[i] = zero16();
      if (a_n[i] >= 0) {
        T* e = reinterpret_cast<T*>(&ra[i]);
#pragma unroll
        for (int j = 0; j < EPU; ++j) {
          const int kred = cel + j;
          if (kred < kred_total) {
            const int rr = kred / (d.S * d.C);
            const int rem = kred % (d.S * d.C);
            const int ss = rem / d.C;
            const int cc = rem % d.C;
            const int hi = a_i1[i] * d.sh + rr - d.ph;
            const int wi = a_i2[i] * d.sw + ss - d.pw;
            if (hi >= 0 && hi < d.H && wi >= 0 && wi < d.W)
              e[j] = Asrc[(((int64_t)a_n[i] * d.H + hi) * d.W + wi) * d.C + cc];
          }
        }
      }
    }
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      rb[i] = zero16();
      if (b_ptr[i] != nullptr) {
        T* e = reinterpret_cast<T*>(&rb[i]);
#pragma unroll
        for (int j = 0; j < EPU; ++j)
          if (cel + j < kred_total)
            e[j] = b_ptr[i][cel + j];
      }
    }
  };
  auto write_tile_flat = [&](char* la, char* lb) {
#pragma unroll
    for (int i = 0; i < 4; ++i)
      lds_write16(la, trow + 32 * i, tcol * 16, ra[i]);
#pragma unroll
    for (int i = 0; i < 4; ++i)
      lds_write16(lb, trow + 32 * i, tcol * 16, rb[i]);
  };

  const int frow = lane & 15;
  // 2-byte dtypes: 2 K=32 mfma steps per tile reading 16B fragments;
  // fp32: 8 K=4 v_mfma_f32_16x16x4_f32 steps reading one scalar per lane
  // (exact f32 at the vector rate — no TF32 on gfx950, guide §3).
  auto lds_off = [&](int row, int colb) {
    return (MODE == FWD_FLAT) ? swz_off(row, colb) : swz7_off(row, colb);
  };
  auto mfma_tile = [&](const char* la, const char* lb) {
#pragma unroll
    for (int kk = 0; kk < ConvElems<T>::KSTEPS; ++kk) {
      if constexpr (sizeof(T) == 2) {
        const int fcolb = kk * 64 + (lane >> 4) * 16;
        typename MfmaFrag<T>::type a[4], b[4];
#pragma unroll
        for (int mr = 0; mr < 4; ++mr) {
          uint4 v = *reinterpret_cast<const uint4*>(
              la + lds_off(wr + mr * 16 + frow, fcolb));
          a[mr] = *reinterpret_cast<typename MfmaFrag<T>::type*>(&v);
        }
#pragma unroll
        for (int nr = 0; nr < 4; ++nr) {
          uint4 v = *reinterpret_cast<const uint4*>(
              lb + lds_off(wc + nr * 16 + frow, fcolb));
          b[nr] = *reinterpret_cast<typename MfmaFrag<T>::type*>(&v);
        }
#pragma unroll
        for (int mr = 0; mr < 4; ++mr)
#pragma unroll
          for (int nr = 0; nr < 4; ++nr)
            acc[mr][nr] = mfma_16x16x32<T>(a[mr], b[nr], acc[mr][nr]);
      } else {
        const int fcolb = kk * 16 + (lane >> 4) * 4;
        float a[4], b[4];
#pragma unroll
        for (int mr = 0; mr < 4; ++mr)
          a[mr] = *reinterpret_cast<const float*>(
              la + lds_off(wr + mr * 16 + frow, fcolb));
#pragma unroll
        for (int nr = 0; nr < 4; ++nr)
          b[nr] = *reinterpret_cast<const float*>(
              lb + lds_off(wc + nr * 16 + frow, fcolb));
#pragma unroll
        for (int mr = 0; mr < 4; ++mr)
#pragma unroll
          for (int nr = 0; nr < 4; ++nr)
            acc[mr][nr] = __builtin_amdgcn_mfma_f32_16x16x4f32(
                a[mr], b[nr], acc[mr][nr], 0, 0, 0);
      }
    }
  };

  if (MODE == FWD_FLAT) {
    load_tile_flat(0);
    int c0n = 0;
    for (int t = 0; t < total_tiles; ++t) {
      write_tile_flat(lA[0], lB[0]);
      __syncthreads();
      if (t + 1 < total_tiles) {
        c0n += BKT;
        load_tile_flat(c0n);
      }
      mfma_tile(lA[0], lB[0]);
      __syncthreads();
    }
  } else if (total_tiles == 0) {
    // empty phase (e.g. stride-2 1x1 odd rows): output is all zeros
  } else if (DBUF) {
    // Counted-vmcnt pipeline (guide §5 T3/T4 minimum form): raw s_barrier
    // (no implicit vmcnt(0) drain) + `s_waitcnt vmcnt(8)` lets tile t+1's
    // 8 glds DMAs stay in flight across the barrier while tile t computes.
    // Per iteration: barrier #1 = everyone done READING buf[cur^1] (it is
    // about to be overwritten); issue glds(t+1); vmcnt(8) = own glds(t)
    // landed (FIFO; only t+1's 8 remain); barrier #2 = every wave's
    // glds(t) writes visible; then MFMA on buf[cur].
    stage_glds(0, 0, lA[0], lB[0]);
    int rsn = 0, c0n = 0;  // counters for tile t+1
    for (int t = 0; t < total_tiles; ++t) {
      const int cur = t & 1;
      const bool more = (t + 1) < total_tiles;
      __builtin_amdgcn_s_barrier();
      if (more) {
        c0n += BKT;
        if (c0n >= kred_total) { c0n = 0; ++rsn; }
        stage_glds(rsn, c0n, cur ? lA[0] : lA[1], cur ? lB[0] : lB[1]);
        asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      } else {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      __builtin_amdgcn_s_barrier();
      // raw s_barrier has no compiler memory-ordering semantics: pin the
      // fragment ds_reads below it (guide §5.4 rule 18 analog)
      __builtin_amdgcn_sched_barrier(0);
      mfma_tile(cur ? lA[1] : lA[0], cur ? lB[1] : lB[0]);
    }
  } else {
    int rsn = 0, c0n = 0;
    for (int t = 0; t < total_tiles; ++t) {
      stage_glds(rsn, c0n, lA[0], lB[0]);
      c0n += BKT;
      if (c0n >= kred_total) { c0n = 0; ++rsn; }
      __syncthreads();  // drains the DMA (vmcnt(0) + barrier)
      mfma_tile(lA[0], lB[0]);
      __syncthreads();
    }
  }

  // ---- epilogue -----------------------------------------------------------
  // For DGRAD_S2 the kernel's m-space is the phase subgrid: map each row
  // back to its actual (n, h, w) output offset.
  auto out_row_elem = [&](int64_t m) -> int64_t {
    if (MODE == DGRAD_S2) {
      const uint32_t mw = fd_div((uint32_t)m, gp.fdWp);
      const int w_ = (int)((uint32_t)m - mw * gp.Wp) * d.sw + gp.pw2;
      const uint32_t n_ = fd_div((uint32_t)m, gp.fdHpWp);
      const int h_ = (int)(mw - n_ * gp.Hp) * d.sh + gp.ph2;
      return (((int64_t)n_ * d.H + h_) * d.W + w_) * (int64_t)Nc;
    }
    return m * Nc;
  };
  const int en = lane & 15;
  const int em = (lane >> 4) * 4;
  if constexpr (sizeof(T) == 4) {
    // fp32: direct fragment stores (the repack tile would need 64 KiB)
#pragma unroll
    for (int nr = 0; nr < 4; ++nr) {
      const int n = tile_n + wc + nr * 16 + en;
      if (n >= Nc) continue;
      const float bv = (!isdg && bias != nullptr) ? to_f32(bias[n]) : 0.f;
#pragma unroll
      for (int mr = 0; mr < 4; ++mr) {
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const int64_t m = tile_m + wr + mr * 16 + em + reg;
          if (m >= M) continue;
          float f = acc[mr][nr][reg] + bv;
          if (relu) f = fmaxf(f, 0.f);
          if (accum != nullptr) f += to_f32(accum[out_row_elem(m) + n]);
          out[out_row_elem(m) + n] = f;
        }
      }
    }
    return;
  }
  // 2-byte dtypes: repack through LDS, store coalesced 16 B rows
  __syncthreads();
  T* etile = reinterpret_cast<T*>(smem);  // [TBM][TBN] elements (32 KiB)
  const bool emit_stats = (MODE == FWD) && stats_ws != nullptr;
  float cs[4] = {}, cq[4] = {};  // per-nr column {sum, sumsq} (BN fusion)
#pragma unroll
  for (int nr = 0; nr < 4; ++nr) {
    const int n_loc = wc + nr * 16 + en;
    const float bv = (!isdg && bias != nullptr && tile_n + n_loc < Nc)
                         ? to_f32(bias[tile_n + n_loc]) : 0.f;
#pragma unroll
    for (int mr = 0; mr < 4; ++mr) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        float f = acc[mr][nr][reg] + bv;
        if (relu) f = fmaxf(f, 0.f);
        if (emit_stats) {
          cs[nr] += f;
          cq[nr] += f * f;
        }
        etile[(wr + mr * 16 + em + reg) * TBN + n_loc] = from_f32<T>(f);
      }
    }
  }
  __syncthreads();
  const int nvalid = Nc - tile_n;
  constexpr int UPR = TBN / 8;  // 16 B units per etile row
  // bn_bwd fusion per-thread state: this thread's column group is FIXED
  // across store iterations (256 % UPR == 0), so the BN per-channel
  // constants hoist out of the loop and the partial sums live in 16 VGPRs.
  float bs0[8] = {}, bs1[8] = {};
  float bmean[8], brstd[8], bw8[8], bb8[8];
  if constexpr (BSTATS) {
    const int cbase = tile_n + (threadIdx.x % UPR) * 8;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int c = cbase + j;
      const bool ok = c < Nc;
      bmean[j] = ok ? bn_mean[c] : 0.f;
      brstd[j] = ok ? bn_rstd[c] : 0.f;
      // with a saved bit-mask (residual-join BNs, bn.hip mode 3) the
      // ReLU activity comes from the mask, not from w*xhat+b
      bw8[j] = (ok && bn_mask == nullptr) ? to_f32(bn_w[c]) : 0.f;
      bb8[j] = (ok && bn_mask == nullptr) ? to_f32(bn_b[c]) : 0.f;
    }
  }
#pragma unroll
  for (int it = 0; it < TBM * TBN / 8 / 256; ++it) {
    const int unit = threadIdx.x + it * 256;
    const int row = unit / UPR;
    const int ub = (unit % UPR) * 8;
    const int64_t m = tile_m + row;
    if (m < M && ub < nvalid) {
      uint4 u = *reinterpret_cast<const uint4*>(etile + row * TBN + ub);
      const int64_t off = out_row_elem(m) + tile_n + ub;
      if (accum != nullptr) {
        // dgrad residual-join fusion: dx += the identity-path gradient
        // here instead of a separate full-tensor autograd add
        uint4 a = *reinterpret_cast<const uint4*>(accum + off);
        T* ue = reinterpret_cast<T*>(&u);
        const T* ae = reinterpret_cast<const T*>(&a);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          ue[j] = from_f32<T>(to_f32(ue[j]) + to_f32(ae[j]));
      }
      if constexpr (BSTATS) {
        // g = this dgrad row (the BN's dy, residual grad already
        // accumulated above), masked by the recomputed or saved ReLU
        uint4 bx = *reinterpret_cast<const uint4*>(bnx + off);
        const T* be = reinterpret_cast<const T*>(&bx);
        const T* ue = reinterpret_cast<const T*>(&u);
        uint8_t mb = 0;
        if (bn_mask != nullptr) mb = bn_mask[off >> 3];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float xh = (to_f32(be[j]) - bmean[j]) * brstd[j];
          const bool on = bn_mask != nullptr ? ((mb >> j) & 1)
                                             : (xh * bw8[j] + bb8[j] > 0.f);
          if (on) {
            const float g = to_f32(ue[j]);
            bs0[j] += g * xh;
            bs1[j] += g;
          }
        }
      }
      *reinterpret_cast<uint4*>(out + off) = u;
    }
  }
  if constexpr (BSTATS) {
    __syncthreads();
    // fold the 256/UPR threads sharing each column group through LDS
    float* bred = reinterpret_cast<float*>(smem);  // [256][8][2] = 16 KiB
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      bred[(threadIdx.x * 8 + j) * 2 + 0] = bs0[j];
      bred[(threadIdx.x * 8 + j) * 2 + 1] = bs1[j];
    }
    __syncthreads();
    const int col = threadIdx.x;
    if (col < TBN && tile_n + col < Nc) {
      const int ccg = col >> 3, cj = col & 7;
      float s0 = 0.f, s1 = 0.f;
#pragma unroll 4
      for (int r = 0; r < 256 / UPR; ++r) {
        s0 += bred[((r * UPR + ccg) * 8 + cj) * 2 + 0];
        s1 += bred[((r * UPR + ccg) * 8 + cj) * 2 + 1];
      }
      float* dst = bstats_ws +
                   (int64_t)(bchunk_base + (int)(bid / nt_n)) * 2 * Nc +
                   tile_n + col;
      dst[0] = s0;
      dst[Nc] = s1;
    }
  }
  if (!emit_stats) return;
  // fold the per-thread column sums: lanes l, l+16, l+32, l+48 share a
  // column (em groups) -> xor-shuffle; then the two m-half waves combine
  // through LDS (reused as scratch after the store loop).
#pragma unroll
  for (int nr = 0; nr < 4; ++nr) {
#pragma unroll
    for (int off = 16; off <= 32; off <<= 1) {
      cs[nr] += __shfl_xor(cs[nr], off, 64);
      cq[nr] += __shfl_xor(cq[nr], off, 64);
    }
  }
  __syncthreads();
  float* sred = reinterpret_cast<float*>(smem);  // [4 wid][4 nr][16 en][2]
  if (lane < 16) {
#pragma unroll
    for (int nr = 0; nr < 4; ++nr) {
      sred[((wid * 4 + nr) * 16 + lane) * 2 + 0] = cs[nr];
      sred[((wid * 4 + nr) * 16 + lane) * 2 + 1] = cq[nr];
    }
  }
  __syncthreads();
  const int col = threadIdx.x;
  if (col < TBN && tile_n + col < Nc) {
    const int nr = (col >> 4) & 3, e = col & 15;
    float* dst = stats_ws + (int64_t)(bid / nt_n) * 2 * Nc + tile_n + col;
    if (TBN == 64) {
      // 4x1 wave column: every wave covers every column
      float s = 0.f, q = 0.f;
#pragma unroll
      for (int w = 0; w < 4; ++w) {
        s += sred[((w * 4 + nr) * 16 + e) * 2];
        q += sred[((w * 4 + nr) * 16 + e) * 2 + 1];
      }
      dst[0] = s;
      dst[Nc] = q;
    } else {
      // col = wcg*64 + nr*16 + en is covered by waves {wcg, wcg+2}
      const int wcg = col >> 6;
      const int i0 = ((wcg * 4 + nr) * 16 + e) * 2;
      const int i1 = (((wcg + 2) * 4 + nr) * 16 + e) * 2;
      dst[0] = sred[i0] + sred[i1];
      dst[Nc] = sred[i0 + 1] + sred[i1 + 1];
    }
  }
}

// (256,64) n-narrow tile: Nc <= 64 outputs with enough m-tiles to fill
// the chip (ResNet layer1 64-out convs, their dgrads, the im2col'd stem).
bool conv_use_n64(int Nc, int64_t M) {
  const char* env = getenv("DDL_CONV_N64");
  if (env && env[0] == '0') return false;
  return Nc <= 64 && M / 256 >= 512;
}

bool conv_use_dbuf(int cred, int rs_total) {
  const char* env = getenv("DDL_CONV_DBUF");
  if (env) return env[0] == '1';
  (void)cred; (void)rs_total;
  return false;
}

}  // namespace

// 8-phase 256^2 deep-pipeline variant (conv_igemm8.hip)
extern "C" int ddl_conv8_applicable(int dtype, int64_t M, int Nc, int rs_total, int Cred);
extern "C" void ddl_conv8_fwd(const void* x, const void* w, const void* bias,
                              void* y, float* stats_ws, int N, int H, int W,
                              int C, int K, int R, int S, int P, int Q,
                              int sh, int sw, int ph, int pw, int relu,
                              int dtype, hipStream_t s);
extern "C" void ddl_conv8_dgrad(const void* dy, const void* w_t, void* dx,
                                const void* accum, int N, int H, int W, int C,
                                int K, int R, int S, int P, int Q, int sh,
                                int sw, int ph, int pw, int dtype,
                                hipStream_t s);

extern "C" {

void ddl_conv_fwd(const void* x, const void* w, const void* bias, void* y,
                  int N, int H, int W, int C, int K, int R, int S, int P,
                  int Q, int sh, int sw, int ph, int pw, int relu, int dtype,
                  hipStream_t s) {
  ddl_conv_fwd_stats(x, w, bias, y, nullptr, N, H, W, C, K, R, S, P, Q, sh,
                     sw, ph, pw, relu, dtype, s);
}

// stats_ws != nullptr (2-byte dtypes, vectorizable C): also emit the
// [cdiv(M,128)][2][K] fp32 BN {sum,sumsq} partials from the epilogue.
void ddl_conv_fwd_stats(const void* x, const void* w, const void* bias,
                        void* y, float* stats_ws, int N, int H, int W, int C,
                        int K, int R, int S, int P, int Q, int sh, int sw,
                        int ph, int pw, int relu, int dtype, hipStream_t s) {
  ConvDims d{N, H, W, C, K, R, S, P, Q, sh, sw, ph, pw};
  const int64_t M = (int64_t)N * P * Q;
  const int Nc = K;
  const bool flat = (C % (dtype == 0 ? 4 : 8)) != 0;
  if (!flat && ddl_conv8_applicable(dtype, M, Nc, R * S, C)) {
    ddl_conv8_fwd(x, w, bias, y, stats_ws, N, H, W, C, K, R, S, P, Q, sh,
                  sw, ph, pw, relu, dtype, s);
    return;
  }
  const bool n64 = !flat && conv_use_n64(Nc, M);
  const int64_t grid =
      n64 ? cdiv(M, 256) * cdiv(Nc, 64) : cdiv(M, BM) * cdiv(Nc, BN);
  const bool dbuf =
      !n64 && conv_use_dbuf(flat ? R * S * C : C, flat ? 1 : R * S);
  if (flat || dtype == 0) stats_ws = nullptr;  // unsupported paths
  DgradPhase gp{};  // unused by FWD modes
#define RUN(T, MODE, DB)                                                       \
  conv_igemm_k<T, MODE, DB><<<(unsigned)grid, 256, 0, s>>>(                    \
      (const T*)x, (const T*)w, (const T*)bias, (T*)y, d, M, Nc, relu, gp,    \
      stats_ws, nullptr)
#define RUN64(T, MODE)                                                         \
  conv_igemm_k<T, MODE, false, 256, 64><<<(unsigned)grid, 256, 0, s>>>(        \
      (const T*)x, (const T*)w, (const T*)bias, (T*)y, d, M, Nc, relu, gp,    \
      stats_ws, nullptr)
#define SEL(T, MODE)                                                           \
  if (n64) RUN64(T, MODE);                                                     \
  else if (dbuf) RUN(T, MODE, true);                                           \
  else RUN(T, MODE, false)
  if (dtype == 0) { if (flat) { SEL(float, FWD_FLAT); } else { SEL(float, FWD); } }
  else if (dtype == 1) { if (flat) { SEL(bf16, FWD_FLAT); } else { SEL(bf16, FWD); } }
  else if (dtype == 2) { if (flat) { SEL(f16, FWD_FLAT); } else { SEL(f16, FWD); } }
#undef SEL
#undef RUN64
#undef RUN
}

// chunk count of the BN partials a fused conv forward will emit
// (must mirror the tile selection above)
int64_t ddl_conv_fwd_stats_chunks(int N, int P, int Q, int C, int K, int R,
                                  int S, int dtype) {
  const int64_t M = (int64_t)N * P * Q;
  const bool flat = (C % (dtype == 0 ? 4 : 8)) != 0;
  if (!flat && ddl_conv8_applicable(dtype, M, K, R * S, C)) return cdiv(M, 256);
  const bool n64 = !flat && conv_use_n64(K, M);
  return n64 ? cdiv(M, 256) : cdiv(M, BM);
}

void ddl_conv_dgrad(const void* dy, const void* w_t, void* dx, int N, int H,
                    int W, int C, int K, int R, int S, int P, int Q, int sh,
                    int sw, int ph, int pw, int dtype, hipStream_t s) {
  ddl_conv_dgrad_accum(dy, w_t, dx, nullptr, N, H, W, C, K, R, S, P, Q, sh,
                       sw, ph, pw, dtype, s);
}

// dx = conv_dgrad(dy, w) + accum  (accum nullable; the residual-join
// gradient add fused into the epilogue)
void ddl_conv_dgrad_accum(const void* dy, const void* w_t, void* dx,
                          const void* accum, int N, int H, int W, int C,
                          int K, int R, int S, int P, int Q, int sh, int sw,
                          int ph, int pw, int dtype, hipStream_t s) {
  ConvDims d{N, H, W, C, K, R, S, P, Q, sh, sw, ph, pw};
  const int Nc = C;
  {
    // per-phase M for strided dgrad (one launch per (h%sh, w%sw) phase)
    const int64_t Mck = (int64_t)N * cdiv(H, sh) * cdiv(W, sw);
    const int rsp = (sh == 1 && sw == 1) ? R * S : cdiv(R, sh) * cdiv(S, sw);
    if (ddl_conv8_applicable(dtype, Mck, Nc, rsp, K)) {
      ddl_conv8_dgrad(dy, w_t, dx, accum, N, H, W, C, K, R, S, P, Q, sh, sw,
                      ph, pw, dtype, s);
      return;
    }
  }
  const bool dbuf = conv_use_dbuf(K, R * S);
  if (sh == 1 && sw == 1) {
    const int64_t M = (int64_t)N * H * W;
    const bool n64 = conv_use_n64(Nc, M);
    const int64_t grid =
        n64 ? cdiv(M, 256) * cdiv(Nc, 64) : cdiv(M, BM) * cdiv(Nc, BN);
    DgradPhase gp{};
#define RUN(T, DB)                                                             \
  conv_igemm_k<T, DGRAD, DB><<<(unsigned)grid, 256, 0, s>>>(                   \
      (const T*)dy, (const T*)w_t, nullptr, (T*)dx, d, M, Nc, 0, gp, nullptr, \
      (const T*)accum)
#define RUN64(T)                                                               \
  conv_igemm_k<T, DGRAD, false, 256, 64><<<(unsigned)grid, 256, 0, s>>>(       \
      (const T*)dy, (const T*)w_t, nullptr, (T*)dx, d, M, Nc, 0, gp, nullptr, \
      (const T*)accum)
    if (dtype == 0) { if (n64) RUN64(float); else if (dbuf) RUN(float, true); else RUN(float, false); }
    else if (dtype == 1) { if (n64) RUN64(bf16); else if (dbuf) RUN(bf16, true); else RUN(bf16, false); }
    else if (dtype == 2) { if (n64) RUN64(f16); else if (dbuf) RUN(f16, true); else RUN(f16, false); }
#undef RUN64
#undef RUN
    return;
  }
  // strided dgrad: one dense launch per (h%sh, w%sw) phase
  for (int ph2 = 0; ph2 < sh; ++ph2) {
    // (plain path — bn_bwd-stats fusion goes through ddl_conv_dgrad_bnstats)
    for (int pw2 = 0; pw2 < sw; ++pw2) {
      DgradPhase gp;
      gp.ph2 = ph2; gp.pw2 = pw2;
      gp.r0 = ((ph2 + ph) % sh + sh) % sh;
      gp.s0 = ((pw2 + pw) % sw + sw) % sw;
      gp.Sp = gp.s0 < S ? (S - gp.s0 + sw - 1) / sw : 0;
      gp.Hp = ph2 < H ? (H - ph2 + sh - 1) / sh : 0;
      gp.Wp = pw2 < W ? (W - pw2 + sw - 1) / sw : 0;
      if (gp.Hp == 0 || gp.Wp == 0) continue;
      gp.fdWp = make_fastdiv((uint32_t)gp.Wp);
      gp.fdHpWp = make_fastdiv((uint32_t)gp.Hp * gp.Wp);
      const int64_t M = (int64_t)N * gp.Hp * gp.Wp;
      const int64_t grid = cdiv(M, BM) * cdiv(Nc, BN);
#define RUN(T, DB)                                                             \
  conv_igemm_k<T, DGRAD_S2, DB><<<(unsigned)grid, 256, 0, s>>>(                \
      (const T*)dy, (const T*)w_t, nullptr, (T*)dx, d, M, Nc, 0, gp, nullptr, \
      (const T*)accum)
      if (dtype == 0) { if (dbuf) RUN(float, true); else RUN(float, false); }
      else if (dtype == 1) { if (dbuf) RUN(bf16, true); else RUN(bf16, false); }
      else if (dtype == 2) { if (dbuf) RUN(f16, true); else RUN(f16, false); }
#undef RUN
    }
  }
}


// dgrad + BN-backward-stats fusion: dx = conv_dgrad(dy, w) and
// bstats_ws[chunk][2][C] = the preceding BatchNorm's {sum(g*xhat), sum(g)}
// partials (relu_mode-2 mask recomputed from its own input bnx).
// 2-byte dtypes only; no accum (the fused residual-join path and this
// fusion never coincide: residual-join BNs take the mode-1 mask).
int64_t ddl_conv_dgrad_bnstats_chunks(int N, int H, int W, int C, int K,
                                      int sh, int sw) {
  if (sh == 1 && sw == 1) {
    const int64_t M = (int64_t)N * H * W;
    return conv_use_n64(C, M) ? cdiv(M, 256) : cdiv(M, BM);
  }
  int64_t chunks = 0;
  for (int ph2 = 0; ph2 < sh; ++ph2)
    for (int pw2 = 0; pw2 < sw; ++pw2) {
      const int Hp = ph2 < H ? (H - ph2 + sh - 1) / sh : 0;
      const int Wp = pw2 < W ? (W - pw2 + sw - 1) / sw : 0;
      if (Hp && Wp) chunks += cdiv((int64_t)N * Hp * Wp, BM);
    }
  return chunks;
}

void ddl_conv_dgrad_bnstats(const void* dy, const void* w_t, void* dx,
                            const void* bnx, const float* bn_mean,
                            const float* bn_rstd, const void* bn_w,
                            const void* bn_b, float* bstats_ws,
                            const void* accum, const uint8_t* bn_mask,
                            int N, int H, int W, int C, int K, int R, int S,
                            int P, int Q, int sh, int sw, int ph, int pw,
                            int dtype, hipStream_t s) {
  ConvDims d{N, H, W, C, K, R, S, P, Q, sh, sw, ph, pw};
  const int Nc = C;
  if (sh == 1 && sw == 1) {
    const int64_t M = (int64_t)N * H * W;
    const bool n64 = conv_use_n64(Nc, M);
    const int64_t grid =
        n64 ? cdiv(M, 256) * cdiv(Nc, 64) : cdiv(M, BM) * cdiv(Nc, BN);
    DgradPhase gp{};
#define RUNB(T)                                                                  conv_igemm_k<T, DGRAD, false, 128, 128, true>                                      <<<(unsigned)grid, 256, 0, s>>>(                                                   (const T*)dy, (const T*)w_t, nullptr, (T*)dx, d, M, Nc, 0, gp,                 nullptr, (const T*)accum, (const T*)bnx, bn_mean, bn_rstd, (const T*)bn_w,             (const T*)bn_b, bstats_ws, 0, bn_mask)
#define RUNB64(T)                                                                conv_igemm_k<T, DGRAD, false, 256, 64, true>                                       <<<(unsigned)grid, 256, 0, s>>>(                                                   (const T*)dy, (const T*)w_t, nullptr, (T*)dx, d, M, Nc, 0, gp,                 nullptr, (const T*)accum, (const T*)bnx, bn_mean, bn_rstd, (const T*)bn_w,             (const T*)bn_b, bstats_ws, 0, bn_mask)
    if (dtype == 1) { if (n64) RUNB64(bf16); else RUNB(bf16); }
    else { if (n64) RUNB64(f16); else RUNB(f16); }
#undef RUNB64
#undef RUNB
    return;
  }
  int chunk_base = 0;
  for (int ph2 = 0; ph2 < sh; ++ph2) {
    for (int pw2 = 0; pw2 < sw; ++pw2) {
      DgradPhase gp;
      gp.ph2 = ph2; gp.pw2 = pw2;
      gp.r0 = ((ph2 + ph) % sh + sh) % sh;
      gp.s0 = ((pw2 + pw) % sw + sw) % sw;
      gp.Sp = gp.s0 < S ? (S - gp.s0 + sw - 1) / sw : 0;
      gp.Hp = ph2 < H ? (H - ph2 + sh - 1) / sh : 0;
      gp.Wp = pw2 < W ? (W - pw2 + sw - 1) / sw : 0;
      if (gp.Hp == 0 || gp.Wp == 0) continue;
      gp.fdWp = make_fastdiv((uint32_t)gp.Wp);
      gp.fdHpWp = make_fastdiv((uint32_t)gp.Hp * gp.Wp);
      const int64_t M = (int64_t)N * gp.Hp * gp.Wp;
      const int64_t grid = cdiv(M, BM) * cdiv(Nc, BN);
#define RUNB(T)                                                                  conv_igemm_k<T, DGRAD_S2, false, 128, 128, true>                                   <<<(unsigned)grid, 256, 0, s>>>(                                                   (const T*)dy, (const T*)w_t, nullptr, (T*)dx, d, M, Nc, 0, gp,                 nullptr, (const T*)accum, (const T*)bnx, bn_mean, bn_rstd, (const T*)bn_w,             (const T*)bn_b, bstats_ws, chunk_base, bn_mask)
      if (dtype == 1) RUNB(bf16); else RUNB(f16);
#undef RUNB
      chunk_base += (int)cdiv(M, BM);
    }
  }
}

}  // extern "C"
