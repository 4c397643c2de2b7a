// Fused SwiGLU activation: out = silu(gate) * up, with [gate; up] packed in
// one tensor (the fused gate-up ColumnParallel GEMM output, reference
// modules/moe/experts.py:219-233 GLU path).  HBM-bound; vectorized.
//
// x (N, 2I) bf16 -> out (N, I) bf16;  bwd: dy (N, I) -> dx (N, 2I).

#include "common.h"

extern "C" __global__ void __launch_bounds__(256)
swiglu_fwd_kernel(const short* __restrict__ x, short* __restrict__ out,
                  long N, int I) {
  // ONE int64 div/mod at entry, then carry-advance (row, i) each
  // iteration; v_rcp_f32 for the sigmoid (1-ulp fp32 — far below bf16
  // output rounding).
  int nvec = I >> 3;
  const long stride = (long)gridDim.x * blockDim.x;
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long row = idx / nvec;
  int i = (int)(idx - row * nvec);
  const int di = (int)(stride % nvec);
  const long drow = stride / nvec;
  for (; idx < N * nvec; idx += stride) {
    const short* g = x + row * 2 * I + i * 8;
    const short* u = x + row * 2 * I + I + i * 8;
    s8v gv = *(const s8v*)g;
    s8v uv = *(const s8v*)u;
    s8v o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bits2f(gv[j]);
      float s = gf * __builtin_amdgcn_rcpf(
          1.f + __builtin_amdgcn_exp2f(-gf * 1.4426950408889634f));
      o[j] = f2bits(s * bits2f(uv[j]));
    }
    *(s8v*)(out + row * I + i * 8) = o;
    row += drow;
    i += di;
    if (i >= nvec) { i -= nvec; ++row; }
  }
}

extern "C" __global__ void __launch_bounds__(256)
swiglu_bwd_kernel(const short* __restrict__ x, const short* __restrict__ dy,
                  short* __restrict__ dx, long N, int I) {
  // same div-free iteration + fast-rcp structure as the forward
  int nvec = I >> 3;
  const long stride = (long)gridDim.x * blockDim.x;
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long row = idx / nvec;
  int i = (int)(idx - row * nvec);
  const int di = (int)(stride % nvec);
  const long drow = stride / nvec;
  for (; idx < N * nvec; idx += stride) {
    const short* g = x + row * 2 * I + i * 8;
    const short* u = x + row * 2 * I + I + i * 8;
    const short* d = dy + row * I + i * 8;
    s8v gv = *(const s8v*)g;
    s8v uv = *(const s8v*)u;
    s8v dv = *(const s8v*)d;
    s8v dg, du;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bits2f(gv[j]);
      float df = bits2f(dv[j]);
      float sig = __builtin_amdgcn_rcpf(
          1.f + __builtin_amdgcn_exp2f(-gf * 1.4426950408889634f));
      float s = gf * sig;
      dg[j] = f2bits(df * bits2f(uv[j]) * (sig + s * (1.f - sig)));
      du[j] = f2bits(df * s);
    }
    *(s8v*)(dx + row * 2 * I + i * 8) = dg;
    *(s8v*)(dx + row * 2 * I + I + i * 8) = du;
    row += drow;
    i += di;
    if (i >= nvec) { i -= nvec; ++row; }
  }
}

extern "C" void swiglu_fwd(const void* x, void* out, long N, int I,
                           hipStream_t stream) {
  long work = N * (I >> 3);
  int blocks = (int)((work + 255) / 256);
  if (blocks > 2048) blocks = 2048;
  swiglu_fwd_kernel<<<blocks, 256, 0, stream>>>((const short*)x, (short*)out,
                                                N, I);
}

extern "C" void swiglu_bwd(const void* x, const void* dy, void* dx, long N,
                           int I, hipStream_t stream) {
  long work = N * (I >> 3);
  int blocks = (int)((work + 255) / 256);
  if (blocks > 2048) blocks = 2048;
  swiglu_bwd_kernel<<<blocks, 256, 0, stream>>>((const short*)x,
                                                (const short*)dy, (short*)dx,
                                                N, I);
}
