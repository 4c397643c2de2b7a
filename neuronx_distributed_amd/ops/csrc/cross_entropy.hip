// Fused vocab-parallel cross-entropy pieces for CDNA4.
//
// Replaces the torch-op chain of the reference's _ParallelCrossEntropy
// (reference parallel_layers/loss_functions.py:10-129) that eager ROCm
// would run as ~8 elementwise/reduce passes over fp32-materialized logits.
// Three HBM-bound kernels over the bf16 (N, V/tp) logits shard:
//   ce_rowmax:  per-row local max -> f32 (N)          [TP all-reduce MAX]
//   ce_sumexp:  per-row sum(exp(l - m)) + owned-target logit fetch
//               -> f32 (N), f32 (N)                    [TP all-reduce SUM]
//   ce_bwd:     dlogits = (exp(l-m)/sumexp - onehot) * gout, bf16 out
// The softmax matrix is never materialized (recomputed in bwd).

#include "common.h"

extern "C" __global__ void __launch_bounds__(256)
ce_rowmax_kernel(const short* __restrict__ logits, float* __restrict__ out,
                 long N, int V) {
  __shared__ float scratch[16];
  for (long row = blockIdx.x; row < N; row += gridDim.x) {
    const short* r = logits + row * V;
    float m = -3.0e38f;
    int nvec = V >> 3;
    for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
      s8v v = *(const s8v*)(r + i * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) m = fmaxf(m, bits2f(v[j]));
    }
    for (int i = (nvec << 3) + threadIdx.x; i < V; i += blockDim.x)
      m = fmaxf(m, bits2f(r[i]));
    // block max reduce
    int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
    m = wave_reduce(m, MaxOp());
    if (lane == 0) scratch[wid] = m;
    __syncthreads();
    if (wid == 0) {
      float v = lane < (blockDim.x >> 6) ? scratch[lane] : -3.0e38f;
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        v = fmaxf(v, __shfl_xor(v, off, 64));
      if (lane == 0) out[row] = v;
    }
    __syncthreads();
  }
}

extern "C" __global__ void __launch_bounds__(256)
ce_sumexp_kernel(const short* __restrict__ logits,
                 const float* __restrict__ rowmax,
                 const long* __restrict__ targets,
                 float* __restrict__ sumexp, float* __restrict__ predicted,
                 long N, int V, long vocab_start) {
  __shared__ float scratch[16];
  for (long row = blockIdx.x; row < N; row += gridDim.x) {
    const short* r = logits + row * V;
    const float m = rowmax[row];
    float s = 0.f;
    int nvec = V >> 3;
    for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
      s8v v = *(const s8v*)(r + i * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) s += __builtin_amdgcn_exp2f((bits2f(v[j]) - m) * 1.4426950408889634f);
    }
    for (int i = (nvec << 3) + threadIdx.x; i < V; i += blockDim.x)
      s += __builtin_amdgcn_exp2f((bits2f(r[i]) - m) * 1.4426950408889634f);
    s = block_reduce_sum(s, scratch);
    if (threadIdx.x == 0) {
      sumexp[row] = s;
      long t = targets[row] - vocab_start;
      predicted[row] = (t >= 0 && t < V) ? bits2f(r[t]) - m : 0.f;
    }
    __syncthreads();
  }
}

extern "C" __global__ void __launch_bounds__(256)
ce_bwd_kernel(const short* __restrict__ logits,
              const float* __restrict__ rowmax,
              const float* __restrict__ sumexp,
              const long* __restrict__ targets,
              const float* __restrict__ gout, short* __restrict__ dlogits,
              long N, int V, long vocab_start) {
  for (long row = blockIdx.x; row < N; row += gridDim.x) {
    const short* r = logits + row * V;
    short* d = dlogits + row * V;
    const float m = rowmax[row];
    const float inv = 1.0f / sumexp[row];
    const float g = gout[row];
    const long t = targets[row] - vocab_start;
    int nvec = V >> 3;
    for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
      s8v v = *(const s8v*)(r + i * 8);
      s8v o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float p = __builtin_amdgcn_exp2f((bits2f(v[j]) - m) * 1.4426950408889634f) * inv;
        long col = i * 8 + j;
        if (col == t) p -= 1.0f;
        o[j] = f2bits(p * g);
      }
      *(s8v*)(d + i * 8) = o;
    }
    for (int i = (nvec << 3) + threadIdx.x; i < V; i += blockDim.x) {
      float p = __builtin_amdgcn_exp2f((bits2f(r[i]) - m) * 1.4426950408889634f) * inv;
      if (i == t) p -= 1.0f;
      d[i] = f2bits(p * g);
    }
  }
}

extern "C" void ce_rowmax(const void* logits, void* out, long N, int V,
                          hipStream_t s) {
  int blocks = N < 2048 ? (int)N : 2048;
  ce_rowmax_kernel<<<blocks, 256, 0, s>>>((const short*)logits, (float*)out,
                                          N, V);
}

extern "C" void ce_sumexp(const void* logits, const void* rowmax,
                          const void* targets, void* sumexp, void* predicted,
                          long N, int V, long vocab_start, hipStream_t s) {
  int blocks = N < 2048 ? (int)N : 2048;
  ce_sumexp_kernel<<<blocks, 256, 0, s>>>(
      (const short*)logits, (const float*)rowmax, (const long*)targets,
      (float*)sumexp, (float*)predicted, N, V, vocab_start);
}

extern "C" void ce_bwd(const void* logits, const void* rowmax,
                       const void* sumexp, const void* targets,
                       const void* gout, void* dlogits, long N, int V,
                       long vocab_start, hipStream_t s) {
  int blocks = N < 2048 ? (int)N : 2048;
  ce_bwd_kernel<<<blocks, 256, 0, s>>>(
      (const short*)logits, (const float*)rowmax, (const float*)sumexp,
      (const long*)targets, (const float*)gout, (short*)dlogits, N, V,
      vocab_start);
}
