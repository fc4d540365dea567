// K3: wgrad implicit GEMM — dw[k,r,s,c] = sum_m dy[m,k] * im2col_x[m,(r,s,c)].
// Per (r,s) and m-chunk: GEMM tile dw[128k][128c], reduce over the chunk's
// rows 64 at a time.  Both operands need the reduce dim (m) innermost in the
// fragment, so dy and x tiles are TRANSPOSE-STAGED into LDS ([k][m] / [c][m])
// via an in-register 8x4 repack (4 coalesced 16B loads -> 8 b64 writes,
// bank-spread by the shared swizzle), then the MFMA loop reads b128 like the
// forward kernel.  Split-K over m writes per-chunk fp32 slabs (plain dense
// stores, no atomics) combined by a deterministic reduce+cast kernel.
#include "conv_tile.h"
#include "api.h"
#include <stdlib.h>

namespace {

enum WMode { TILED = 0, FLAT = 1 };

typedef __attribute__((ext_vector_type(4))) short s16x4_t;
typedef __attribute__((address_space(3))) s16x4_t* as3_s16x4p;

// SIMPLE = 1x1/stride-1/no-pad conv: the x operand is just row m of the
// input (no im2col decode; the per-tile q/p/n divides were a hot cost).
// TBM/TBN/NTHR: tile geometry.  (128,128,256) is the full 4-wave 2x2 tile;
// (64,64,64) is a single-wave tile for small K / small C*R*S convs (layer1
// of ResNet: a 128^2 tile on a 64^2 dw plane wastes 75% of the MFMA work —
// the small tile instead fills the chip with extra split-K chunks).
// The 64^2 TILED/SIMPLE variants sit at ~200 VGPR, just past the 192
// 2-waves/SIMD boundary; minWavesPerEU=2 squeezes them back under it
// (FLAT's im2col decode needs the registers more than the occupancy).
// TR16 (2-byte dtypes, SBUF 128^2 only, DDL_WGRAD_TR16=1): stage the
// tiles in their NATURAL [m][ch] layout — one plain ds_write_b128 per
// 16 B load instead of the 8x4 register transpose (8 b64 writes + ~50
// pack VALU) — and read MFMA fragments with gfx950's hardware transpose
// read `ds_read_b64_tr_b16`.  Image: [ch-block cb][m-half][32 m][16 ch]
// subtiles (32 B rows), slot base padded by cb*32 so the 8-lane b128
// write groups spread banks, and rows XOR'd (m ^ ((m>>3)&1)<<2) so the
// two 16-lane tr-read halves of a 32-lane service group (k-groups 8
// rows apart) land on different banks.
template <typename T, int MODE, bool SIMPLE, bool DBUF, int TBM = 128,
          int TBN = 128, int NTHR = 256, bool TR16 = false>
__global__ __launch_bounds__(NTHR, (NTHR == 64) ? 2 : 1)
void conv_wgrad_k(
    const T* __restrict__ dy, const T* __restrict__ x,
    float* __restrict__ dw_ws, ConvDims d, int64_t M, int64_t chunk_rows,
    int64_t slab_len, FastDiv fdq, FastDiv fdpq, int ntiles, int chunks,
    int xcd_map) {
  static_assert(TBM == TBN, "square tiles only");
  static_assert(!TR16 || (sizeof(T) == 2 && !DBUF),
                "TR16: 2-byte dtypes, single-buffer only");
  // TR16 operand tile: TBM/16 ch-blocks x 2 m-halves slots x 1 KiB
  // + cb padding
  constexpr int ATILE =
      TR16 ? (TBM / 16) * 2 * 1024 + (TBM / 16) * 32 : TBM * 128;
  // (tile, chunk) decode.  xcd_map=1: co-locate a chunk's tiles on one XCD
  // (block b runs on XCD b%8, MI355X_MICROARCH.md §Workgroup dispatch —
  // affinity used for speed only).  The tiles of one chunk sweep the same
  // 64-row m-window of dy/x in near-lockstep; co-located, that window is
  // read from HBM once per chunk instead of once per XCD that happens to
  // host a tile (per-XCD L2s are private) — wgrad re-reads dy ntc times
  // and x ntk times otherwise.  Used only when chunks >= 8 so all XCDs
  // stay loaded; xcd_map=0 is the plain (x=tile, z=chunk) grid.
  int tilei, chunk;
  if (xcd_map) {
    const int slot = (int)blockIdx.x >> 3;
    chunk = ((int)blockIdx.x & 7) + 8 * (slot / ntiles);
    tilei = slot % ntiles;
    if (chunk >= chunks) return;
  } else {
    tilei = blockIdx.x;
    chunk = blockIdx.z;
  }
  // each m-chunk owns a private fp32 slab: plain dense stores, no atomics
  // (a chunks-deep atomicAdd chain per dw element serializes at L2 like
  // the BN-stats case), and the final sum is deterministic.
  dw_ws += (int64_t)chunk * slab_len;
  __shared__ __align__(16) char smem[DBUF ? 4 * ATILE : 2 * ATILE];
  char* const lA[2] = {smem, DBUF ? smem + 2 * ATILE : smem};
  char* const lB[2] = {smem + ATILE,
                       DBUF ? smem + 3 * ATILE : smem + ATILE};

  const int kred_total = (MODE == FLAT) ? d.R * d.S * d.C : d.C;
  const int ntc = (kred_total + TBN - 1) / TBN;
  const int tile_k = (tilei / ntc) * TBM;
  const int tile_c = (tilei % ntc) * TBN;
  const int rs = (MODE == FLAT) ? 0 : blockIdx.y;
  const int r = rs / d.S, s = rs % d.S;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = (NTHR == 64) ? 0 : (wid >> 1) * 64;
  const int wc = (NTHR == 64) ? 0 : (wid & 1) * 64;
  // staging assignment (in-register transpose): each thread owns a
  // 16B-wide k/c group x 4 consecutive m, in PASSES m-column passes.
  // (128,256) 2-byte dtypes: 16 groups of 8 rows x 16 m-quads, 1 pass;
  // (64,64) 2-byte: 8 groups x 8 m-quads x 2 passes; fp32 analogous.
  constexpr int EPU = ConvElems<T>::PER_UNIT;   // k rows per 16 B load
  constexpr int NSU = TBM / EPU;                // staging k/c groups
  constexpr int BKT = ConvElems<T>::BKT;        // m columns per LDS tile
  constexpr int MPP = (NTHR / NSU) * 4;         // m columns per pass
  constexpr int PASSES = BKT / MPP;
  static_assert(PASSES >= 1 && BKT % MPP == 0, "staging geometry");
  const int su = threadIdx.x & (NSU - 1);       // k/c group
  const int sm4 = (threadIdx.x / NSU) * 4;      // first of 4 owned m columns

  f32x4_t acc[4][4] = {};

  const int64_t m_begin = (int64_t)chunk * chunk_rows;
  const int64_t m_end = min(M, m_begin + chunk_rows);
  const int64_t mtiles =
      (m_end > m_begin) ? (m_end - m_begin + BKT - 1) / BKT : 0;

  uint4 va[PASSES][4], vb[PASSES][4];
  auto load_tile = [&](int64_t ti) {
    const int64_t m0 = m_begin + ti * BKT;
    const int k0g = tile_k + su * EPU;
#pragma unroll
    for (int ps = 0; ps < PASSES; ++ps)
#pragma unroll
      for (int mm = 0; mm < 4; ++mm) {
        const int64_t m = m0 + ps * MPP + sm4 + mm;
        va[ps][mm] = zero16();
        if (m < m_end && k0g + EPU <= d.K)
          va[ps][mm] = *reinterpret_cast<const uint4*>(dy + m * d.K + k0g);
      }
#pragma unroll
    for (int ps = 0; ps < PASSES; ++ps)
#pragma unroll
      for (int mm = 0; mm < 4; ++mm) {
        const int64_t m = m0 + ps * MPP + sm4 + mm;
        vb[ps][mm] = zero16();
        T* e = reinterpret_cast<T*>(&vb[ps][mm]);
        if (SIMPLE) {
          const int c0 = tile_c + su * EPU;
          if (m < m_end && c0 + EPU <= d.C)
            vb[ps][mm] = *reinterpret_cast<const uint4*>(x + m * d.C + c0);
        } else if (m < m_end) {
          const uint32_t mq = fd_div((uint32_t)m, fdq);
          const int q = (int)((uint32_t)m - mq * d.Q);
          const uint32_t n = fd_div((uint32_t)m, fdpq);
          const int p = (int)(mq - n * d.P);
          if (MODE == TILED) {
            const int hi = p * d.sh + r - d.ph;
            const int wi = q * d.sw + s - d.pw;
            const int c0 = tile_c + su * EPU;
            if (hi >= 0 && hi < d.H && wi >= 0 && wi < d.W && c0 + EPU <= d.C)
              vb[ps][mm] = *reinterpret_cast<const uint4*>(
                  x + (((int64_t)n * d.H + hi) * d.W + wi) * d.C + c0);
          } else {
#pragma unroll
            for (int j = 0; j < EPU; ++j) {
              const int kred = tile_c + su * EPU + j;
              if (kred < kred_total) {
                const int rr = kred / (d.S * d.C);
                const int rem = kred % (d.S * d.C);
                const int ss = rem / d.C;
                const int cc = rem % d.C;
                const int hi = p * d.sh + rr - d.ph;
                const int wi = q * d.sw + ss - d.pw;
                if (hi >= 0 && hi < d.H && wi >= 0 && wi < d.W)
                  e[j] = x[(((int64_t)n * d.H + hi) * d.W + wi) * d.C + cc];
              }
            }
          }
        }
      }
  };

  // TR16 subtile addressing: slot (cb, mh) holds logical m-rows
  // mh*32..+31 of channels cb*16..+15; 32 B rows, row-XOR for tr reads.
  auto tr16_off = [&](int cb, int mh, int m31, int chbyte) {
    const int pm = m31 ^ (((m31 >> 3) & 1) << 2);
    return (cb * 2 + mh) * 1024 + cb * 32 + pm * 32 + chbyte;
  };
  auto transpose_write = [&](char* lds, const uint4 (&v)[4], int sm4p) {
    if constexpr (TR16) {
      const int cb = su >> 1, half = su & 1;
#pragma unroll
      for (int mm = 0; mm < 4; ++mm) {
        const int m = sm4p + mm;
        *reinterpret_cast<uint4*>(
            lds + tr16_off(cb, m >> 5, m & 31, half * 16)) = v[mm];
      }
    } else if constexpr (sizeof(T) == 2) {
      const uint16_t* e0 = reinterpret_cast<const uint16_t*>(&v[0]);
      const uint16_t* e1 = reinterpret_cast<const uint16_t*>(&v[1]);
      const uint16_t* e2 = reinterpret_cast<const uint16_t*>(&v[2]);
      const uint16_t* e3 = reinterpret_cast<const uint16_t*>(&v[3]);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        uint64_t pk = (uint64_t)e0[j] | ((uint64_t)e1[j] << 16) |
                      ((uint64_t)e2[j] << 32) | ((uint64_t)e3[j] << 48);
        *reinterpret_cast<uint64_t*>(lds + swz_off(su * 8 + j, sm4p * 2)) = pk;
      }
    } else {
      const float* e0 = reinterpret_cast<const float*>(&v[0]);
      const float* e1 = reinterpret_cast<const float*>(&v[1]);
      const float* e2 = reinterpret_cast<const float*>(&v[2]);
      const float* e3 = reinterpret_cast<const float*>(&v[3]);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float4 pk{e0[j], e1[j], e2[j], e3[j]};
        *reinterpret_cast<float4*>(lds + swz_off(su * 4 + j, sm4p * 4)) = pk;
      }
    }
  };
  auto write_tile = [&](char* la, char* lb) {
#pragma unroll
    for (int ps = 0; ps < PASSES; ++ps) {
      transpose_write(la, va[ps], sm4 + ps * MPP);
      transpose_write(lb, vb[ps], sm4 + ps * MPP);
    }
  };

  const int frow = lane & 15;
  auto mfma_tile = [&](const char* la, const char* lb) {
#pragma unroll
    for (int kk = 0; kk < ConvElems<T>::KSTEPS; ++kk) {
      if constexpr (TR16) {
        // hardware transpose read: each 16-lane group reads a
        // [4 m][16 ch] block; lane l gets ch (l&15), m = 8*(l>>4)+j.
        // Two reads (+4 m rows) build the 8-deep mfma fragment.
        const int lx = lane & 15;
        const int lg = lane >> 4;
        const int row0 = 8 * lg + (lx >> 2);
        const int pr0 = row0 ^ (((row0 >> 3) & 1) << 2);
        const int row1 = row0 + 4;
        const int pr1 = row1 ^ (((row1 >> 3) & 1) << 2);
        const int lane_off = (lx & 3) * 8;
        typename MfmaFrag<T>::type a[4], b[4];
#pragma unroll
        for (int mr = 0; mr < 4; ++mr) {
          const int cb = (wr + mr * 16) >> 4;
          const char* sb = la + (cb * 2 + kk) * 1024 + cb * 32 + lane_off;
          s16x4_t r0 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
              (as3_s16x4p)(sb + pr0 * 32));
          s16x4_t r1 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
              (as3_s16x4p)(sb + pr1 * 32));
          s16x4_t* h = reinterpret_cast<s16x4_t*>(&a[mr]);
          h[0] = r0;
          h[1] = r1;
        }
#pragma unroll
        for (int nr = 0; nr < 4; ++nr) {
          const int cb = (wc + nr * 16) >> 4;
          const char* sb = lb + (cb * 2 + kk) * 1024 + cb * 32 + lane_off;
          s16x4_t r0 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
              (as3_s16x4p)(sb + pr0 * 32));
          s16x4_t r1 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
              (as3_s16x4p)(sb + pr1 * 32));
          s16x4_t* h = reinterpret_cast<s16x4_t*>(&b[nr]);
          h[0] = r0;
          h[1] = r1;
        }
#pragma unroll
        for (int mr = 0; mr < 4; ++mr)
#pragma unroll
          for (int nr = 0; nr < 4; ++nr)
            acc[mr][nr] = mfma_16x16x32<T>(a[mr], b[nr], acc[mr][nr]);
      } else if constexpr (sizeof(T) == 2) {
        const int fcolb = kk * 64 + (lane >> 4) * 16;
        typename MfmaFrag<T>::type a[4], b[4];
#pragma unroll
        for (int mr = 0; mr < 4; ++mr) {
          uint4 v = lds_read16(la, wr + mr * 16 + frow, fcolb);
          a[mr] = *reinterpret_cast<typename MfmaFrag<T>::type*>(&v);
        }
#pragma unroll
        for (int nr = 0; nr < 4; ++nr) {
          uint4 v = lds_read16(lb, wc + nr * 16 + frow, fcolb);
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
              la + swz_off(wr + mr * 16 + frow, fcolb));
#pragma unroll
        for (int nr = 0; nr < 4; ++nr)
          b[nr] = *reinterpret_cast<const float*>(
              lb + swz_off(wc + nr * 16 + frow, fcolb));
#pragma unroll
        for (int mr = 0; mr < 4; ++mr)
#pragma unroll
          for (int nr = 0; nr < 4; ++nr)
            acc[mr][nr] = __builtin_amdgcn_mfma_f32_16x16x4f32(
                a[mr], b[nr], acc[mr][nr], 0, 0, 0);
      }
    }
  };

  if (mtiles > 0) {
    if (DBUF) {
      load_tile(0);
      write_tile(lA[0], lB[0]);
      __syncthreads();
      for (int64_t t = 0; t < mtiles; ++t) {
        const bool more = (t + 1) < mtiles;
        const int cur = (int)(t & 1);
        if (more) load_tile(t + 1);
        mfma_tile(cur ? lA[1] : lA[0], cur ? lB[1] : lB[0]);
        if (more) write_tile(cur ? lA[0] : lA[1], cur ? lB[0] : lB[1]);
        __syncthreads();
      }
    } else {
      load_tile(0);
      for (int64_t t = 0; t < mtiles; ++t) {
        write_tile(lA[0], lB[0]);
        __syncthreads();
        if (t + 1 < mtiles) load_tile(t + 1);
        mfma_tile(lA[0], lB[0]);
        __syncthreads();
      }
    }
  }

  // ---- epilogue: repack through LDS (fp32; SBUF smem fits half the tile
  // at a time), then coalesced 16B stores into this chunk's fp32 slab -----
  const int rsc = d.R * d.S * d.C;
  const int nvalid = kred_total - tile_c;
  float* etile = reinterpret_cast<float*>(smem);
  const int en = lane & 15;
  const int em = (lane >> 4) * 4;
  // SBUF 128^2: two passes of 64 k-rows (etile is half the tile); the
  // 64^2 tile's etile fits the staging smem whole.
  constexpr int halves = (DBUF || 2 * ATILE >= TBM * TBN * 4) ? 1 : 2;
  for (int h = 0; h < halves; ++h) {
    __syncthreads();
    const int rows_per = TBM / halves;
#pragma unroll
    for (int mr = 0; mr < 4; ++mr) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int krow = wr + mr * 16 + em + reg;
        if (krow < h * rows_per || krow >= (h + 1) * rows_per) continue;
#pragma unroll
        for (int nr = 0; nr < 4; ++nr) {
          etile[(krow - h * rows_per) * TBN + wc + nr * 16 + en] =
              acc[mr][nr][reg];
        }
      }
    }
    __syncthreads();
    const int nunits = rows_per * TBN / 4 / NTHR;  // 16B units per thread
    for (int it = 0; it < nunits; ++it) {
      const int unit = (int)threadIdx.x + it * NTHR;
      const int row = unit / (TBN / 4);
      const int ub = (unit % (TBN / 4)) * 4;
      const int k = tile_k + h * rows_per + row;
      if (k < d.K && ub < nvalid) {
        const int cidx = tile_c + ub;
        if (MODE == FLAT) {
          for (int j = 0; j < 4 && cidx + j < kred_total; ++j)
            dw_ws[(int64_t)k * rsc + cidx + j] = etile[row * TBN + ub + j];
        } else {
          const int64_t o = (int64_t)k * rsc + (int64_t)rs * d.C + cidx;
          *reinterpret_cast<float4*>(dw_ws + o) =
              *reinterpret_cast<const float4*>(etile + row * TBN + ub);
        }
      }
    }
  }
}

// dw[i] = cast(sum_ch ws[ch][i]) — the deterministic chunk combine.
// Block = 16 float4-element-groups x 16 chunk-slices, so the chunk loop is
// 16-way parallel: the 64^2-tile wgrad path runs ~2k chunks over a tiny dw
// plane (4k elements for a 64->64 1x1), where a 4-way split left the chip
// nearly idle.  The split order is fixed, so the sum stays deterministic.
template <typename T>
__global__ void wgrad_reduce_cast_k(const float* __restrict__ ws, T* dw,
                                    int64_t len, int chunks) {
  __shared__ float red[16][16][4];
  const int eg = threadIdx.x & 15;   // element group (4 floats)
  const int cq = threadIdx.x >> 4;   // chunk slice
  const int64_t e0 = ((int64_t)blockIdx.x * 16 + eg) * 4;
  float acc[4] = {0, 0, 0, 0};
  if (e0 < len) {
    for (int ch = cq; ch < chunks; ch += 16) {
      const float* p = ws + (int64_t)ch * len + e0;
      if (e0 + 4 <= len) {
        float4 v = *reinterpret_cast<const float4*>(p);
        acc[0] += v.x; acc[1] += v.y; acc[2] += v.z; acc[3] += v.w;
      } else {
        for (int j = 0; j < (int)(len - e0); ++j) acc[j] += p[j];
      }
    }
  }
#pragma unroll
  for (int j = 0; j < 4; ++j) red[cq][eg][j] = acc[j];
  __syncthreads();
  if (cq == 0 && e0 < len) {
#pragma unroll
    for (int sl = 1; sl < 16; ++sl)
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[j] += red[sl][eg][j];
    const int nt = (int)min((int64_t)4, len - e0);
    for (int j = 0; j < nt; ++j) dw[e0 + j] = from_f32<T>(acc[j]);
  }
}

struct WgradPlan {
  int64_t chunks, chunk_rows;
  int ntk, ntc, rs_total, kred_total, tile;
};

WgradPlan wgrad_plan(int N, int C, int K, int R, int S, int P, int Q) {
  WgradPlan pl;
  const int64_t M = (int64_t)N * P * Q;
  const bool flat = (C % 8) != 0;
  pl.kred_total = flat ? R * S * C : C;
  // 64^2 single-wave tile when BOTH dw-plane sides are <= 64 (ResNet
  // layer1 64->64 convs): the 128^2 tile would run 75% dead MFMA work.
  // One-sided-small shapes (64->256 etc.) measure FASTER on the 128^2
  // tile: halving the tile there doubles the re-read of the streamed
  // operand, and these kernels are traffic-bound, not MFMA-bound.  FLAT
  // also stays on 128^2 — its im2col decode spills at 64-tile pressure.
  pl.tile = (!flat && K <= 64 && pl.kred_total <= 64) ? 64 : 128;
  pl.ntk = (int)cdiv(K, pl.tile);
  pl.ntc = (int)cdiv(pl.kred_total, pl.tile);
  pl.rs_total = flat ? 1 : R * S;
  // block target: the 64^2 tile runs one wave per 64-thread block, so it
  // needs ~8x the blocks of the 4-wave 128^2 tile for the same occupancy.
  const char* bt = getenv(pl.tile == 64 ? "DDL_WGRAD_BLOCKS64"
                                        : "DDL_WGRAD_BLOCKS");
  const int64_t target =
      bt ? atoll(bt) : (pl.tile == 64 ? 2048 : 512);  // measured (swept)
  int64_t chunks = std::max<int64_t>(
      1, target / ((int64_t)pl.ntk * pl.ntc * pl.rs_total));
  chunks = std::min<int64_t>(chunks, cdiv(M, BK));
  chunks = std::min<int64_t>(chunks, pl.tile == 64 ? 2048 : 512);
  // cap slab memory at ~256 MB
  const int64_t slab = (int64_t)K * R * S * C * 4;
  if (slab > 0) chunks = std::min<int64_t>(chunks, std::max<int64_t>(1, (256LL << 20) / slab));
  // round DOWN to a multiple of 8 so the XCD-affinity block map (chunk%8
  // -> XCD) is load-balanced — but only when that keeps >=75% of the
  // block target (rounding 14->8 idled half the chip on l3.c2; rounding
  // UP past the resident-block capacity added a nearly-empty second
  // dispatch round that doubled l1.c2's wall time).  Unrounded counts
  // simply run the plain grid (launch checks chunks%8).
  if (chunks >= 8) {
    const int64_t dn = (chunks / 8) * 8;
    if (dn * 4 >= chunks * 3) chunks = dn;
  }
  pl.chunk_rows = cdiv(cdiv(M, chunks), BK) * BK;
  // keep the rounded count authoritative: trailing chunks may own zero
  // rows (their blocks write zero slabs, which the reduce sums away)
  pl.chunks = chunks;
  return pl;
}

}  // namespace

extern "C" {

int64_t ddl_conv_wgrad_ws_elems(int N, int C, int K, int R, int S, int P,
                                int Q) {
  WgradPlan pl = wgrad_plan(N, C, K, R, S, P, Q);
  return pl.chunks * (int64_t)K * R * S * C;
}

void ddl_conv_wgrad(const void* dy, const void* x, float* dw_ws, void* dw,
                    int N, int H, int W, int C, int K, int R, int S, int P,
                    int Q, int sh, int sw, int ph, int pw, int dtype,
                    hipStream_t s) {
  ConvDims d{N, H, W, C, K, R, S, P, Q, sh, sw, ph, pw};
  const int64_t M = (int64_t)N * P * Q;
  // must match wgrad_plan's rule (grid/mode are derived from the plan)
  const bool flat = (C % 8) != 0;
  WgradPlan pl = wgrad_plan(N, C, K, R, S, P, Q);
  const int64_t slab_len = (int64_t)K * R * S * C;
  const int ntiles = pl.ntk * pl.ntc;
  const char* xe = getenv("DDL_WGRAD_XCD");
  // balanced only when chunks is a multiple of 8 (plan rounds, but the
  // final cdiv can land one short) — otherwise use the plain grid
  const int xcd_map =
      (xe ? xe[0] == '1' : true) && pl.chunks >= 8 && pl.chunks % 8 == 0
          ? 1 : 0;
  dim3 grid;
  if (xcd_map) {
    grid = dim3((unsigned)(8 * cdiv(pl.chunks, 8) * ntiles),
                (unsigned)pl.rs_total, 1u);
  } else {
    grid = dim3((unsigned)ntiles, (unsigned)pl.rs_total,
                (unsigned)pl.chunks);
  }
  const bool simple = (R == 1 && S == 1 && sh == 1 && sw == 1 && ph == 0 &&
                       pw == 0 && !flat);
  const char* env = getenv("DDL_WGRAD_DBUF");
  bool dbuf = env ? env[0] == '1' : true;  // measured: dbuf wins here
  const char* te = getenv("DDL_WGRAD_TR16");
  // measured 2-2.5x over the register-transpose staging on every
  // 128^2-tile shape (l3.c1 218->649 TF); FLAT keeps the old path (its
  // im2col decode is register-bound)
  const bool tr16 = (te ? te[0] == '1' : true) && dtype != 0 && !flat;
  if (tr16) dbuf = false;
  FastDiv fdq = make_fastdiv((uint32_t)Q);
  FastDiv fdpq = make_fastdiv((uint32_t)P * Q);
#define RUN(T, MODE, SIMP, DB)                                                 \
  conv_wgrad_k<T, MODE, SIMP, DB><<<grid, 256, 0, s>>>(                        \
      (const T*)dy, (const T*)x, dw_ws, d, M, pl.chunk_rows, slab_len, fdq,    \
      fdpq, ntiles, (int)pl.chunks, xcd_map)
#define RUN64(T, MODE, SIMP)                                                   \
  conv_wgrad_k<T, MODE, SIMP, false, 64, 64, 64><<<grid, 64, 0, s>>>(          \
      (const T*)dy, (const T*)x, dw_ws, d, M, pl.chunk_rows, slab_len, fdq,    \
      fdpq, ntiles, (int)pl.chunks, xcd_map)
#define RUN64TR(T, MODE, SIMP)                                                 \
  conv_wgrad_k<T, MODE, SIMP, false, 64, 64, 64, true><<<grid, 64, 0, s>>>(    \
      (const T*)dy, (const T*)x, dw_ws, d, M, pl.chunk_rows, slab_len, fdq,    \
      fdpq, ntiles, (int)pl.chunks, xcd_map)
#define RUNTR(T, MODE, SIMP)                                                   \
  conv_wgrad_k<T, MODE, SIMP, false, 128, 128, 256, true>                      \
      <<<grid, 256, 0, s>>>((const T*)dy, (const T*)x, dw_ws, d, M,            \
                            pl.chunk_rows, slab_len, fdq, fdpq, ntiles,        \
                            (int)pl.chunks, xcd_map)
#define SELTR(T, MODE, SIMP)                                                   \
  if (tr16) RUNTR(T, MODE, SIMP);                                              \
  else if (dbuf) RUN(T, MODE, SIMP, true);                                     \
  else RUN(T, MODE, SIMP, false)
#define SEL(T, MODE, SIMP)                                                     \
  if (pl.tile == 64) { if (tr16) RUN64TR(T, MODE, SIMP);                       \
                       else RUN64(T, MODE, SIMP); }                            \
  else SELTR(T, MODE, SIMP)
#define SELF32(T, MODE, SIMP)                                                  \
  if (pl.tile == 64) RUN64(T, MODE, SIMP);                                     \
  else if (dbuf) RUN(T, MODE, SIMP, true);                                     \
  else RUN(T, MODE, SIMP, false)
  if (dtype == 0) {
    if (flat) { SELF32(float, FLAT, false); }
    else if (simple) { SELF32(float, TILED, true); }
    else { SELF32(float, TILED, false); }
  } else if (dtype == 1) {
    if (flat) { SEL(bf16, FLAT, false); }
    else if (simple) { SEL(bf16, TILED, true); }
    else { SEL(bf16, TILED, false); }
  } else if (dtype == 2) {
    if (flat) { SEL(f16, FLAT, false); }
    else if (simple) { SEL(f16, TILED, true); }
    else { SEL(f16, TILED, false); }
  }
#undef SELF32
#undef SEL
#undef SELTR
#undef RUNTR
#undef RUN64TR
#undef RUN64
#undef RUN
  int g = (int)cdiv(slab_len, 64);
#define RED(T)                                                                 \
  wgrad_reduce_cast_k<T><<<g, 256, 0, s>>>(dw_ws, (T*)dw, slab_len,            \
                                           (int)pl.chunks)
  if (dtype == 0) RED(float);
  else if (dtype == 1) RED(bf16);
  else if (dtype == 2) RED(f16);
#undef RED
}

}  // extern "C"
