// Fused rotary position embedding (neox rotate-half) for CDNA4.
//
// Replaces the reference's torch-level rotary that the Neuron compiler
// fuses (reference modules/attention/utils.py:42-77) — one HBM-bound pass
// applying RoPE to Q and K in a single launch.
//
// Layout: q (B, S, Hq, D) bf16, k (B, S, Hk, D) bf16, cos/sin (S, D/2) f32
// (caller slices cos/sin for position offsets / CP shards).
// rotate-half pairing: (d, d + D/2).
// backward = forward with sin negated (rotation transpose), same kernel.

#include "common.h"

extern "C" __global__ void __launch_bounds__(256)
rope_kernel(const short* __restrict__ q, const short* __restrict__ k,
            short* __restrict__ qo, short* __restrict__ ko,
            const float* __restrict__ cos_t, const float* __restrict__ sin_t,
            int B, int S, int Hq, int Hk, int D, float sign, int pos_offset) {
  // out-of-place (halves the traffic vs clone+inplace); one wave per
  // (b, s, h) row; 4 waves/block
  long total = (long)B * S * (Hq + Hk);
  int half = D >> 1;
  for (long row = blockIdx.x * 4 + (threadIdx.x >> 6); row < total;
       row += (long)gridDim.x * 4) {
    int lane = threadIdx.x & 63;
    long bs = row / (Hq + Hk);
    int h = (int)(row % (Hq + Hk));
    int s = (int)(bs % S);
    const short* src;
    short* dst;
    if (h < Hq) {
      src = q + ((bs * Hq + h) * (long)D);
      dst = qo + ((bs * Hq + h) * (long)D);
    } else {
      src = k + ((bs * Hk + (h - Hq)) * (long)D);
      dst = ko + ((bs * Hk + (h - Hq)) * (long)D);
    }
    const float* cr = cos_t + (long)(s + pos_offset) * half;
    const float* sr = sin_t + (long)(s + pos_offset) * half;
    for (int d = lane; d < half; d += 64) {
      float c = cr[d];
      float sn = sr[d] * sign;
      float x0 = bits2f(src[d]);
      float x1 = bits2f(src[d + half]);
      dst[d] = f2bits(x0 * c - x1 * sn);
      dst[d + half] = f2bits(x1 * c + x0 * sn);
    }
  }
}

extern "C" void rope_fwd(const void* q, const void* k, void* qo, void* ko,
                         const void* cos_t, const void* sin_t,
                         int B, int S, int Hq, int Hk, int D, int pos_offset,
                         int backward, hipStream_t stream) {
  long total = (long)B * S * (Hq + Hk);
  int blocks = (int)((total + 3) / 4);
  if (blocks > 4096) blocks = 4096;
  rope_kernel<<<blocks, 256, 0, stream>>>(
      (const short*)q, (const short*)k, (short*)qo, (short*)ko,
      (const float*)cos_t, (const float*)sin_t, B, S, Hq,
      Hk, D, backward ? -1.f : 1.f, pos_offset);
}
