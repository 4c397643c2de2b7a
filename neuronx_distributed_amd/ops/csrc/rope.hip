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
  // out-of-place (halves the traffic vs clone+inplace).
  long total = (long)B * S * (Hq + Hk);
  int half = D >> 1;
  if (half == 64) {
    // D=128 fast path: 16 lanes per (b,s,h) row, lane owns the d-quad
    // [4*sub, 4*sub+4) of BOTH halves via b64 loads — no cross-lane
    // exchange, 4-row-per-wave ILP, coalesced 128 B segments.  (The old
    // one-wave-per-row scalar-b16 form ran 3.2x off the traffic floor:
    // two elements per lane per row, every iteration a full vmcnt
    // stall.)
    const int sub = threadIdx.x & 15;
    const long row0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) >> 4;
    const long rstep = ((long)gridDim.x * blockDim.x) >> 4;
    typedef short s4 __attribute__((ext_vector_type(4)));
    for (long row = row0; row < total; row += rstep) {
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
      const long crow = (long)(s + pos_offset) * 64 + sub * 4;
      f4v c4 = *(const f4v*)(cos_t + crow);
      f4v n4 = *(const f4v*)(sin_t + crow);
      s4 x0 = *(const s4*)(src + sub * 4);
      s4 x1 = *(const s4*)(src + 64 + sub * 4);
      s4 y0, y1;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float a = bits2f(x0[j]), bb = bits2f(x1[j]);
        float sn = n4[j] * sign;
        y0[j] = f2bits(a * c4[j] - bb * sn);
        y1[j] = f2bits(bb * c4[j] + a * sn);
      }
      *(s4*)(dst + sub * 4) = y0;
      *(s4*)(dst + 64 + sub * 4) = y1;
    }
    return;
  }
  // generic D fallback: one wave per row
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
  // D=128 fast path packs 16 rows per 256-thread block; generic 4
  long per_blk = (D == 128) ? 16 : 4;
  long nb = (total + per_blk - 1) / per_blk;
  int blocks = nb > 8192 ? 8192 : (int)nb;
  rope_kernel<<<blocks, 256, 0, stream>>>(
      (const short*)q, (const short*)k, (short*)qo, (short*)ko,
      (const float*)cos_t, (const float*)sin_t, B, S, Hq,
      Hk, D, backward ? -1.f : 1.f, pos_offset);
}
