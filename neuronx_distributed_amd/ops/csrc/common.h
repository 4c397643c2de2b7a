// Common helpers for the CDNA4 (gfx950) kernels.
// Wavefront = 64 on CDNA; block sizes are multiples of 64.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <cstdint>

#define WAVE 64

#define HIP_CHECK(cmd)                                                         \
  do {                                                                         \
    hipError_t e_ = (cmd);                                                     \
    if (e_ != hipSuccess) {                                                    \
      fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(e_),        \
              __FILE__, __LINE__);                                             \
      abort();                                                                 \
    }                                                                          \
  } while (0)

using bf16 = __hip_bfloat16;

// vectorized bf16 access (G13: hipcc does not auto-vectorize bf16 loads)
typedef short s8v __attribute__((ext_vector_type(8)));
typedef float f4v __attribute__((ext_vector_type(4)));

__device__ __forceinline__ float bf2f(bf16 v) { return __bfloat162float(v); }
__device__ __forceinline__ bf16 f2bf(float v) { return __float2bfloat16(v); }

__device__ __forceinline__ float bits2f(short s) {
  union { unsigned u; float f; } c;
  c.u = ((unsigned)(unsigned short)s) << 16;
  return c.f;
}
__device__ __forceinline__ short f2bits(float f) {
  union { float f; unsigned u; } c;
  c.f = f;
  unsigned r = c.u + 0x7FFF + ((c.u >> 16) & 1);  // round-to-nearest-even
  return (short)(r >> 16);
}

// wave-wide reduction over 64 lanes
template <typename Op>
__device__ __forceinline__ float wave_reduce(float v, Op op) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = op(v, __shfl_xor(v, off, 64));
  return v;
}

struct SumOp { __device__ float operator()(float a, float b) const { return a + b; } };
struct MaxOp { __device__ float operator()(float a, float b) const { return fmaxf(a, b); } };

// block-wide sum reduction (threads must be multiple of 64, <=1024)
__device__ __forceinline__ float block_reduce_sum(float v, float* lds_scratch) {
  int lane = threadIdx.x & 63;
  int wid = threadIdx.x >> 6;
  v = wave_reduce(v, SumOp());
  if (lane == 0) lds_scratch[wid] = v;
  __syncthreads();
  int nw = blockDim.x >> 6;
  v = (threadIdx.x < (unsigned)nw) ? lds_scratch[threadIdx.x] : 0.f;
  if (wid == 0) {
#pragma unroll
    for (int off = 8; off > 0; off >>= 1)
      v += __shfl_xor(v, off, 64);
    if (lane == 0) lds_scratch[0] = v;
  }
  __syncthreads();
  return lds_scratch[0];
}
