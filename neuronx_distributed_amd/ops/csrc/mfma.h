// MFMA fragment helpers for v_mfma_f32_32x32x16_bf16 on gfx950.
//
// Layout contracts (verified by tests/test_ops_gpu.py numerics on MI355X):
//   A operand (32x32x16): lane l supplies A[m = l&31][k = 8*(l>>5) + j], j=0..7
//   B operand:            lane l supplies B[k = 8*(l>>5) + j][n = l&31]
//   C/D accumulator (16 f32/lane): element r holds
//       row m = (r&3) + 8*(r>>2) + 4*(l>>5),  col n = l&31
// (cdna_hip_programming.md §3 Fragment layout.)
#pragma once

#include "common.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef unsigned int uint4v __attribute__((ext_vector_type(4)));

__device__ __forceinline__ f32x16 mfma_bf16(bf16x8 a, bf16x8 b, f32x16 c) {
  return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
}

union frag_u {
  bf16x8 bf;
  uint4v u4;
  uint u[4];
};

// accumulator row index for C/D element r (lane-half hi = lane>>5)
__device__ __forceinline__ int acc_row(int r, int hi) {
  return (r & 3) + 8 * (r >> 2) + 4 * hi;
}

// pack two f32 into one u32 of 2 bf16: use the compiler's native casts so
// it can emit v_cvt_pk_bf16_f32 (a hand-rolled RNE costs ~5 VALU per pair,
// cdna_hip_programming.md T12)
__device__ __forceinline__ uint pack_bf16x2(float lo, float hi) {
  union { __bf16 b[2]; uint u; } r;
  r.b[0] = (__bf16)lo;
  r.b[1] = (__bf16)hi;
  return r.u;
}

// XOR swizzle: spread 16B slots of a row-major LDS tile over banks
// (cdna_hip_programming.md §6 G4: byte ^= (row&7)<<4)
__device__ __forceinline__ int swz(int row, int byte_in_row) {
  return byte_in_row ^ ((row & 7) << 4);
}

// full swizzle for 256-byte rows: conflict-FREE when a b128 lane group's
// rows are distinct mod 16 (they are: the 4x16 groups of ds_read_b128)
__device__ __forceinline__ int swz16(int row, int byte_in_row) {
  return byte_in_row ^ ((row & 15) << 4);
}
