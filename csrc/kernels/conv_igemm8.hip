// K1/K2 implicit-GEMM convolution, 256x256 8-phase schedule for gfx950.
//
// The guide's deep-pipeline GEMM template (cdna_hip_programming.md §5
// "256² 8-phase", T3/T4/T5) adapted to the im2col gather: 512 threads
// (8 waves, 2 waves/SIMD), 256x256 output tile, BK=64 reduce step,
// 128 KiB LDS as a ring of four 16 KiB HALF-tiles per operand
// (A = gathered input rows, B = weight rows).  Each K-tile runs four
// phases, one 128x128 output quadrant per phase; all 8 waves work the
// same quadrant (wave piece 64x32 = 4x2 fragments x K=64 = 16 MFMAs per
// phase).  Staging is global_load_lds (2 per thread per phase, one
// half-tile of tile t+1 per phase) with counted `s_waitcnt vmcnt(4)`
// at three of the four phase barriers — loads stay in flight ACROSS
// raw s_barriers, never drained to 0 in the loop (T4).  s_setprio(1)
// brackets each MFMA cluster (T5: pays only on phase-split schedules).
//
// Why: the 4-wave 128² kernel (conv_igemm.hip) measured latency-bound at
// 15-23% MfmaUtil on shallow-Cred ResNet shapes (profiles/r50_pmc_final.md)
// — one K-tile in flight and 16 MFMAs per barrier give too little cover
// for ~900-cycle HBM latency.  This schedule quadruples both the MFMA
// work per phase and the staged bytes in flight.
//
// Quadrant order (0,0)->(0,1)->(1,1)->(1,0) re-uses the A fragments
// across each mh pair and the B fragments across the middle pair, so a
// tile reads 28 b128 fragments instead of 48.  Half-tile consumption
// order A0,B0 | B1 | A1 | (B0) matches the issue order A0,B0,B1,A1 of
// tile t+1 spread one half per phase; each half lands >= 3 phases
// before its first read, and vmcnt(4) (= 2 glds x 2 younger halves)
// before the q0/q1/q2 barriers is exactly the guarantee the next
// phase's reads need.
#include "conv_tile.h"
#include "api.h"

namespace {

enum Mode8 { FWD8 = 0, DGRAD8 = 1, DGRAD8_S2 = 3 };

struct DgradPhase8 {
  int ph2, pw2;
  int r0, s0;
  int Sp;
  int Hp, Wp;
  FastDiv fdWp, fdHpWp;
};

__device__ __align__(256) const unsigned char ddl8_zero_page[256] = {0};

typedef __attribute__((address_space(1))) const unsigned int as1_uint8t;
typedef __attribute__((address_space(3))) unsigned int as3_uint8t;

__device__ __forceinline__ void glds16_8(const void* gsrc, void* lds_base) {
  __builtin_amdgcn_global_load_lds((as1_uint8t*)gsrc, (as3_uint8t*)lds_base,
                                   16, 0, 0);
}

// Tile geometry
constexpr int BM8 = 256;
constexpr int BN8 = 256;
constexpr int HALF_BYTES = 128 * 128;  // 128 rows x 128 B = 16 KiB

template <typename T, int MODE>
__global__ __launch_bounds__(512, 2) void conv_igemm8_k(
    const T* __restrict__ Asrc, const T* __restrict__ Bsrc,
    const T* __restrict__ bias, T* __restrict__ out, ConvDims d, int64_t M,
    int Nc, int relu, DgradPhase8 gp, float* __restrict__ stats_ws,
    const T* __restrict__ accum) {
  static_assert(sizeof(T) == 2, "8-phase kernel is bf16/f16 only");
  // LDS: [ring 2][half 2] x 16 KiB per operand; B at +64 KiB.  ONE
  // __shared__ object (a second one forces vmcnt(0) before every ds_read
  // — guide §5 ".s-level trap (a)").
  __shared__ __align__(16) char smem[4 * HALF_BYTES * 2];
  auto lA = [&](int r, int h) { return smem + (r * 2 + h) * HALF_BYTES; };
  auto lB = [&](int r, int h) {
    return smem + 4 * HALF_BYTES + (r * 2 + h) * HALF_BYTES;
  };

  const int nt_n = (Nc + BN8 - 1) / BN8;
  const unsigned bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int64_t tile_m = (int64_t)(bid / nt_n) * BM8;
  const int tile_n = (int)(bid % nt_n) * BN8;

  const int lane = threadIdx.x & 63;
  const int wid = __builtin_amdgcn_readfirstlane(threadIdx.x >> 6);
  const int wm = (wid >> 2) * 64;   // wave row offset inside a quadrant
  const int wn = (wid & 3) * 32;    // wave col offset inside a quadrant
  const int trow = threadIdx.x >> 3;  // staging row base (0..63)
  const int tcol = threadIdx.x & 7;   // 16 B unit in the row

  f32x4_t acc[4][4][2] = {};  // [quadrant][mfrag][nfrag]

  // Per-thread A-row decode for global rows trow + {0,64,128,192}.
  int a_n[4], a_i1[4], a_i2[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int64_t m = tile_m + trow + 64 * i;
    if (m < M) {
      if (MODE == DGRAD8_S2) {
        const uint32_t mw = fd_div((uint32_t)m, gp.fdWp);
        a_i2[i] = (int)((uint32_t)m - mw * gp.Wp) * d.sw + gp.pw2;
        const uint32_t n = fd_div((uint32_t)m, gp.fdHpWp);
        a_i1[i] = (int)(mw - n * gp.Hp) * d.sh + gp.ph2;
        a_n[i] = (int)n;
      } else if (MODE == DGRAD8) {
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

  const bool isdg = (MODE != FWD8);
  const int Cred = isdg ? d.K : d.C;
  const int64_t a_ld = Cred;
  const int b_ld = Cred;
  const int Rp = (MODE == DGRAD8_S2) ? (d.R - gp.r0 + d.sh - 1) / d.sh : d.R;
  const int Sp = (MODE == DGRAD8_S2) ? gp.Sp : d.S;
  const int rs_total = Rp * Sp;
  const int nct = (Cred + 63) / 64;
  const int total_tiles = rs_total * nct;

  int cur_rs = -1;
  const T* a_ptr[4];
  const T* b_ptr[4];
  const T* zpad = reinterpret_cast<const T*>(ddl8_zero_page);

  auto update_rs_ptrs = [&](int rs) {
    int r, s;
    if (MODE == DGRAD8_S2) {
      r = gp.r0 + (rs / Sp) * d.sh;
      s = gp.s0 + (rs % Sp) * d.sw;
    } else {
      r = rs / d.S;
      s = rs % d.S;
    }
    cur_rs = rs;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      a_ptr[i] = nullptr;
      if (a_n[i] >= 0) {
        if (MODE == FWD8) {
          const int hi = a_i1[i] * d.sh + r - d.ph;
          const int wi = a_i2[i] * d.sw + s - d.pw;
          if (hi >= 0 && hi < d.H && wi >= 0 && wi < d.W)
            a_ptr[i] = Asrc + (((int64_t)a_n[i] * d.H + hi) * d.W + wi) * a_ld;
        } else if (MODE == DGRAD8_S2) {
          const int hn = a_i1[i] + d.ph - r;
          const int wn_ = a_i2[i] + d.pw - s;
          if (hn >= 0 && wn_ >= 0) {
            const int ho = hn / d.sh, wo = wn_ / d.sw;
            if (ho < d.P && wo < d.Q)
              a_ptr[i] = Asrc + (((int64_t)a_n[i] * d.P + ho) * d.Q + wo) * a_ld;
          }
        } else {
          const int hn = a_i1[i] + d.ph - r;
          const int wn_ = a_i2[i] + d.pw - s;
          if (hn >= 0 && wn_ >= 0 && hn % d.sh == 0 && wn_ % d.sw == 0) {
            const int ho = hn / d.sh, wo = wn_ / d.sw;
            if (ho < d.P && wo < d.Q)
              a_ptr[i] = Asrc + (((int64_t)a_n[i] * d.P + ho) * d.Q + wo) * a_ld;
          }
        }
      }
    }
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int row_n = tile_n + trow + 64 * i;
      b_ptr[i] = nullptr;
      if (row_n < Nc)
        b_ptr[i] = Bsrc + ((int64_t)row_n * d.R + r) * d.S * b_ld +
                   (int64_t)s * b_ld;
    }
  };

  // Stage ONE half-tile (2 glds per thread): rows (trow, trow+64) of the
  // half image.  Lane-linear LDS dest; bank swizzle on the SOURCE column
  // unit (rule 21): cu ^ (row&7); (trow+64)&7 == trow&7 so one scu works
  // for both sweeps.
  const int scu = tcol ^ (trow & 7);
  auto stage_half_A = [&](char* dst, int h, int c0, bool valid) {
    const int cel = c0 + scu * 8;  // 8 elements per 16 B unit (2-byte T)
    const bool cok = valid && (cel + 8 <= Cred);
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const T* p = a_ptr[h * 2 + j];
      const T* src = (cok && p != nullptr) ? p + cel : zpad;
      glds16_8(src, dst + j * 8192 + wid * 1024);
    }
  };
  auto stage_half_B = [&](char* dst, int h, int c0, bool valid) {
    const int cel = c0 + scu * 8;
    const bool cok = valid && (cel + 8 <= Cred);
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const T* p = b_ptr[h * 2 + j];
      const T* src = (cok && p != nullptr) ? p + cel : zpad;
      glds16_8(src, dst + j * 8192 + wid * 1024);
    }
  };

  const int frow = lane & 15;
  const int fcol_hi = (lane >> 4) * 16;  // byte offset of this lane's 16 B
  using Frag = typename MfmaFrag<T>::type;
  Frag af[4][2], bf[2][2];
  auto read_a = [&](const char* half) {
#pragma unroll
    for (int mr = 0; mr < 4; ++mr)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        uint4 v = *reinterpret_cast<const uint4*>(
            half + swz7_off(wm + mr * 16 + frow, kk * 64 + fcol_hi));
        af[mr][kk] = *reinterpret_cast<Frag*>(&v);
      }
  };
  auto read_b = [&](const char* half) {
#pragma unroll
    for (int nr = 0; nr < 2; ++nr)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        uint4 v = *reinterpret_cast<const uint4*>(
            half + swz7_off(wn + nr * 16 + frow, kk * 64 + fcol_hi));
        bf[nr][kk] = *reinterpret_cast<Frag*>(&v);
      }
  };
  auto mfma_quadrant = [&](int q) {
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mr = 0; mr < 4; ++mr)
#pragma unroll
      for (int nr = 0; nr < 2; ++nr)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          acc[q][mr][nr] = mfma_16x16x32<T>(af[mr][kk], bf[nr][kk],
                                            acc[q][mr][nr]);
    __builtin_amdgcn_s_setprio(0);
  };

  // ---- prologue: stage tile 0 (A0, B0, B1, A1), nothing waited yet -------
  if (total_tiles > 0) {
    update_rs_ptrs(0);
    stage_half_A(lA(0, 0), 0, 0, true);
    stage_half_B(lB(0, 0), 0, 0, true);
    stage_half_B(lB(0, 1), 1, 0, true);
    stage_half_A(lA(0, 1), 1, 0, true);
  }
  // guard tile 0's A0+B0 for q0's reads; B1[0], A1[0] stay in flight
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  __builtin_amdgcn_sched_barrier(0);

  int rsn = 0, c0n = 0;  // (rs, c0) of tile t+1
  for (int t = 0; t < total_tiles; ++t) {
    const int r = t & 1, rn = (t + 1) & 1;
    const bool more = (t + 1) < total_tiles;
    if (more) {
      c0n += 64;
      if (c0n >= Cred) { c0n = 0; ++rsn; }
      if (rsn != cur_rs) update_rs_ptrs(rsn);
    }
    // ---- q0: quadrant (0,0) — reads guarded by the previous barrier ----
    read_a(lA(r, 0));
    read_b(lB(r, 0));
    stage_half_A(lA(rn, 0), 0, c0n, more);
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");  // B1[t] landed
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    mfma_quadrant(0);
    // ---- q1: quadrant (0,1) — af reused ----
    read_b(lB(r, 1));
    stage_half_B(lB(rn, 0), 0, c0n, more);
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");  // A1[t] landed
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    mfma_quadrant(1);
    // ---- q2: quadrant (1,1) — bf reused ----
    read_a(lA(r, 1));
    stage_half_B(lB(rn, 1), 1, c0n, more);
    __builtin_amdgcn_s_barrier();  // no vmcnt: q3 reads B0[t], long landed
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    mfma_quadrant(2);
    // ---- q3: quadrant (1,0) ----
    read_b(lB(r, 0));
    stage_half_A(lA(rn, 1), 1, c0n, more);
    // guard tile t+1's A0+B0 for the next q0's reads
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    mfma_quadrant(3);
  }

  // ---- epilogue ----------------------------------------------------------
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // drain dummy stages
  __syncthreads();

  auto out_row_elem = [&](int64_t m) -> int64_t {
    if (MODE == DGRAD8_S2) {
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
  T* etile = reinterpret_cast<T*>(smem);  // [256][256] elements = 128 KiB
  const bool emit_stats = (MODE == FWD8) && stats_ws != nullptr;
  float cs[2][2] = {}, cq[2][2] = {};  // per (nh, nr) column partials
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    const int qmh = q >> 1;                       // quadrant order
    const int qnh = (q == 1 || q == 2) ? 1 : 0;   // (0,0),(0,1),(1,1),(1,0)
    const int rbase = qmh * 128 + wm;
    const int cbase = qnh * 128 + wn;
#pragma unroll
    for (int nr = 0; nr < 2; ++nr) {
      const int col = cbase + nr * 16 + en;
      const float bv = (!isdg && bias != nullptr && tile_n + col < Nc)
                           ? to_f32(bias[tile_n + col]) : 0.f;
#pragma unroll
      for (int mr = 0; mr < 4; ++mr) {
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          float f = acc[q][mr][nr][reg] + bv;
          if (relu) f = fmaxf(f, 0.f);
          if (emit_stats) {
            cs[qnh][nr] += f;
            cq[qnh][nr] += f * f;
          }
          etile[(rbase + mr * 16 + em + reg) * BN8 + col] = from_f32<T>(f);
        }
      }
    }
  }
  __syncthreads();
  const int nvalid = Nc - tile_n;
#pragma unroll
  for (int it = 0; it < 16; ++it) {
    const int unit = threadIdx.x + it * 512;
    const int row = unit >> 5;
    const int ub = (unit & 31) * 8;
    const int64_t m = tile_m + row;
    if (m < M && ub < nvalid) {
      uint4 u = *reinterpret_cast<const uint4*>(etile + row * BN8 + ub);
      const int64_t off = out_row_elem(m) + tile_n + ub;
      if (accum != nullptr) {
        uint4 a = *reinterpret_cast<const uint4*>(accum + off);
        T* ue = reinterpret_cast<T*>(&u);
        const T* ae = reinterpret_cast<const T*>(&a);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          ue[j] = from_f32<T>(to_f32(ue[j]) + to_f32(ae[j]));
      }
      *reinterpret_cast<uint4*>(out + off) = u;
    }
  }
  if (!emit_stats) return;
  // Column fold: lanes l, l+16, l+32, l+48 share a column
#pragma unroll
  for (int nh = 0; nh < 2; ++nh)
#pragma unroll
    for (int nr = 0; nr < 2; ++nr) {
#pragma unroll
      for (int off = 16; off <= 32; off <<= 1) {
        cs[nh][nr] += __shfl_xor(cs[nh][nr], off, 64);
        cq[nh][nr] += __shfl_xor(cq[nh][nr], off, 64);
      }
    }
  __syncthreads();
  // scratch [wid 8][nh 2][nr 2][en 16][2] floats = 8 KiB
  float* sred = reinterpret_cast<float*>(smem);
  if (lane < 16) {
#pragma unroll
    for (int nh = 0; nh < 2; ++nh)
#pragma unroll
      for (int nr = 0; nr < 2; ++nr) {
        const int idx = (((wid * 2 + nh) * 2 + nr) * 16 + lane) * 2;
        sred[idx] = cs[nh][nr];
        sred[idx + 1] = cq[nh][nr];
      }
  }
  __syncthreads();
  const int col = threadIdx.x & 255;
  if (threadIdx.x < 256 && tile_n + col < Nc) {
    const int nh = col >> 7, rem = col & 127;
    const int w3 = rem >> 5, nr = (rem >> 4) & 1, e = col & 15;
    float s = 0.f, q = 0.f;
#pragma unroll
    for (int wg = 0; wg < 2; ++wg) {  // the two m-row wave groups
      const int w = wg * 4 + w3;
      const int idx = (((w * 2 + nh) * 2 + nr) * 16 + e) * 2;
      s += sred[idx];
      q += sred[idx + 1];
    }
    float* dst = stats_ws + (int64_t)(bid / nt_n) * 2 * Nc + tile_n + col;
    dst[0] = s;
    dst[Nc] = q;
  }
}

}  // namespace

extern "C" {

// Applicability: 2-byte dtype, vectorizable C (handled by caller), a deep
// enough reduce pipeline and a grid that fills the chip at 1 WG/CU.
// Measured A/B (profiles/conv8_ab.md): at batch 256 the schedule wins
// from ~12 tiles of K-depth (l3.c2: +31% fwd / +35% dgrad); at batch
// 1024 the 128^2 kernel's larger grid already covers latency and only
// the deepest loops still win (l4.c2 72 tiles: +15% fwd / +7% dgrad).
// Route the depth the win is measured at on the headline batch; losses
// appear when the 256^2 grid underfills 256 CUs at 1 WG/CU or Nc < 192.
// DDL_CONV_8PH=0 disables, =2 forces (for sweeps).
int ddl_conv8_applicable(int dtype, int64_t M, int Nc, int rs_total,
                         int Cred) {
  if (dtype == 0) return 0;
  const char* env = getenv("DDL_CONV_8PH");
  if (env && env[0] == '0') return 0;
  if (env && env[0] == '2') return 1;
  const int64_t tiles = (int64_t)rs_total * cdiv(Cred, 64);
  const int64_t grid = cdiv(M, BM8) * cdiv(Nc, BN8);
  return tiles >= 48 && grid >= 160 && Nc >= 192;
}

void ddl_conv8_fwd(const void* x, const void* w, const void* bias, void* y,
                   float* stats_ws, int N, int H, int W, int C, int K, int R,
                   int S, int P, int Q, int sh, int sw, int ph, int pw,
                   int relu, int dtype, hipStream_t s) {
  ConvDims d{N, H, W, C, K, R, S, P, Q, sh, sw, ph, pw};
  const int64_t M = (int64_t)N * P * Q;
  const int64_t grid = cdiv(M, BM8) * cdiv(K, BN8);
  DgradPhase8 gp{};
  if (dtype == 1)
    conv_igemm8_k<bf16, FWD8><<<(unsigned)grid, 512, 0, s>>>(
        (const bf16*)x, (const bf16*)w, (const bf16*)bias, (bf16*)y, d, M, K,
        relu, gp, stats_ws, nullptr);
  else
    conv_igemm8_k<f16, FWD8><<<(unsigned)grid, 512, 0, s>>>(
        (const f16*)x, (const f16*)w, (const f16*)bias, (f16*)y, d, M, K,
        relu, gp, stats_ws, nullptr);
}

void ddl_conv8_dgrad(const void* dy, const void* w_t, void* dx,
                     const void* accum, int N, int H, int W, int C, int K,
                     int R, int S, int P, int Q, int sh, int sw, int ph,
                     int pw, int dtype, hipStream_t s) {
  ConvDims d{N, H, W, C, K, R, S, P, Q, sh, sw, ph, pw};
  if (sh == 1 && sw == 1) {
    const int64_t M = (int64_t)N * H * W;
    const int64_t grid = cdiv(M, BM8) * cdiv(C, BN8);
    DgradPhase8 gp{};
    if (dtype == 1)
      conv_igemm8_k<bf16, DGRAD8><<<(unsigned)grid, 512, 0, s>>>(
          (const bf16*)dy, (const bf16*)w_t, nullptr, (bf16*)dx, d, M, C, 0,
          gp, nullptr, (const bf16*)accum);
    else
      conv_igemm8_k<f16, DGRAD8><<<(unsigned)grid, 512, 0, s>>>(
          (const f16*)dy, (const f16*)w_t, nullptr, (f16*)dx, d, M, C, 0,
          gp, nullptr, (const f16*)accum);
    return;
  }
  for (int ph2 = 0; ph2 < sh; ++ph2) {
    for (int pw2 = 0; pw2 < sw; ++pw2) {
      DgradPhase8 gp;
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
      const int64_t grid = cdiv(M, BM8) * cdiv(C, BN8);
      if (dtype == 1)
        conv_igemm8_k<bf16, DGRAD8_S2><<<(unsigned)grid, 512, 0, s>>>(
            (const bf16*)dy, (const bf16*)w_t, nullptr, (bf16*)dx, d, M, C,
            0, gp, nullptr, (const bf16*)accum);
      else
        conv_igemm8_k<f16, DGRAD8_S2><<<(unsigned)grid, 512, 0, s>>>(
            (const f16*)dy, (const f16*)w_t, nullptr, (f16*)dx, d, M, C, 0,
            gp, nullptr, (const f16*)accum);
    }
  }
}

}  // extern "C"
