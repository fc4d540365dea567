// Shared MFMA tile machinery for the implicit-GEMM conv kernels (K1-K3).
// gfx950: v_mfma_f32_16x16x32_{bf16,f16}, 128x128x64 block tile, 4 waves
// (2x2), 64x64 per wave as 4x4 fragments of 16x16.
//
// Fragment lane mapping (CDNA4, K=32 forms):
//   A[i][k]:  i = lane&15,          k = (lane>>4)*8 + j   (j = 0..7)
//   B[k][n]:  n = lane&15,          k = (lane>>4)*8 + j
//   D[m][n]:  m = (lane>>4)*4 + r,  n = lane&15           (r = reg 0..3)
//
// LDS tiles are [128 rows][64 cols] elements (128 B rows) holding the
// GEMM-transposed operand layouts (row = output row / output col, col =
// reduce dim), XOR-swizzled with ((row&15)<<4) so the 16-lane ds_read_b128
// groups (rows distinct mod 16) are bank-conflict-free (guide §6 G4 / T2).
#pragma once
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;
typedef __attribute__((ext_vector_type(8))) _Float16 f16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

template <typename T> struct MfmaFrag;
template <> struct MfmaFrag<bf16> { using type = bf16x8_t; };
template <> struct MfmaFrag<f16> { using type = f16x8_t; };
template <> struct MfmaFrag<float> { using type = float; };

// Elements per 16-byte staging unit / per 128-byte LDS row.  The byte
// geometry of every tile is dtype-invariant (rows are 128 B), so the
// swizzle, glds staging and LDS footprint are shared: bf16/f16 rows hold
// 64 elements (BK=64 reduce per tile, mfma K=32 x2), fp32 rows hold 32
// (BK=32, v_mfma_f32_16x16x4_f32 x8 — exact f32 at the vector rate;
// gfx950 has no TF32 shortcut, guide §3).
template <typename T> struct ConvElems {
  static constexpr int PER_UNIT = 16 / (int)sizeof(T);   // elements per 16 B
  static constexpr int BKT = 128 / (int)sizeof(T);       // reduce per tile
  static constexpr int KSTEPS = sizeof(T) == 4 ? 8 : 2;  // mfma issues/tile
};

// igemm-local swizzle (bits 4-6 only, so the glds source XOR stays inside
// a row's 128 B span — rule 21); the wgrad register-staged kernel uses the
// wider swz_off above.
__device__ __forceinline__ int swz7_off(int row, int colbyte) {
  return ((row << 7) + colbyte) ^ ((row & 7) << 4);
}

template <typename T>
__device__ __forceinline__ f32x4_t mfma_16x16x32(typename MfmaFrag<T>::type a,
                                                 typename MfmaFrag<T>::type b,
                                                 f32x4_t c);
template <>
__device__ __forceinline__ f32x4_t mfma_16x16x32<bf16>(bf16x8_t a, bf16x8_t b,
                                                       f32x4_t c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}
template <>
__device__ __forceinline__ f32x4_t mfma_16x16x32<f16>(f16x8_t a, f16x8_t b,
                                                      f32x4_t c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b, c, 0, 0, 0);
}

// Tile geometry
constexpr int BM = 128;   // output-row tile
constexpr int BN = 128;   // output-col tile
constexpr int BK = 64;    // reduce tile
constexpr int TILE_BYTES = BM * BK * 2;  // 16 KiB per operand tile

// XOR swizzle g(row) = ((row&15) ^ ((row>>3)&7)) << 4.  The (row&15) term
// makes the b128 fragment reads (16-lane groups, rows m0..m0+15, fixed col)
// conflict-free; folding in (row>>3)&7 additionally spreads the wgrad
// staging's b64 transposed writes (rows stride 8 across lanes) to <=2-way.
// Bits >=4 only, so 16B reads and 8B writes stay aligned; bijective.
__device__ __forceinline__ int swz_off(int row, int colbyte) {
  return ((row << 7) + colbyte) ^ ((((row & 15) ^ ((row >> 3) & 7))) << 4);
}

__device__ __forceinline__ void lds_write16(char* lds, int row, int colbyte,
                                            uint4 v) {
  *reinterpret_cast<uint4*>(lds + swz_off(row, colbyte)) = v;
}

__device__ __forceinline__ uint4 lds_read16(const char* lds, int row,
                                            int colbyte) {
  return *reinterpret_cast<const uint4*>(lds + swz_off(row, colbyte));
}

__device__ __forceinline__ uint4 zero16() { return uint4{0, 0, 0, 0}; }

// Decode helpers ------------------------------------------------------------
struct ConvDims {
  int N, H, W, C, K, R, S, P, Q, sh, sw, ph, pw;
};

// Granlund-Montgomery round-up magic division (valid for any uint32 x):
// m = ceil(2^(32+s)/d), q = (x*m) >> (32+s).  Integer divides in per-tile
// address decode cost ~12 v_rcp-based sequences per staged tile otherwise.
struct FastDiv {
  uint32_t mul;
  int shift;  // total shift = 32 + shift
};

inline FastDiv make_fastdiv(uint32_t d) {
  FastDiv f;
  if ((d & (d - 1)) == 0) {  // power of two: plain shift
    int s = 0;
    while ((1u << s) < d) ++s;
    f.mul = 0;  // sentinel
    f.shift = s;
    return f;
  }
  // s = floor(log2 d) keeps mul = ceil(2^(32+s)/d) < 2^32; round-up magic
  // is then exact for all x < 2^31 (Granlund-Montgomery).
  int s = 0;
  while ((1ull << (s + 1)) <= d) ++s;
  f.mul = (uint32_t)(((((unsigned __int128)1) << (32 + s)) + d - 1) / d);
  f.shift = s;
  return f;
}

__device__ __forceinline__ uint32_t fd_div(uint32_t x, FastDiv f) {
  if (f.mul == 0) return x >> f.shift;
  return (uint32_t)(((uint64_t)x * f.mul) >> 32 >> f.shift);
}
