// Fused MoE decode (token-generation) kernels — K9 in the reference NKI
// inventory (reference kernels/fused_moe.py TKG path).
//
// Decode batches are small (T*top_k slots), so the expert MLPs are
// weight-bandwidth-bound GEMVs.  Slots are grouped per expert into blocks
// of <=16 (wrapper-prepared), each block streams the expert's weights
// EXACTLY ONCE:
//   moe_gateup_kernel: x gather + gate/up GEMV + SwiGLU   -> act (bf16)
//   moe_down_kernel:   act GEMV + affinity scale + fp32 atomic scatter
// No (T, E, *) materialization, no per-expert kernel launches, cold
// experts never touched.
//
// Layouts (bf16): x (T, H); gate_up w (E, H, 2I) local fused [gate|up];
// down w (E, I, H); act (NSLOT_pad, I); out (T, H) fp32 accumulator.
// Slot blocks: block b covers padded slots [16b, 16b+16); slot_token
// holds the TOKEN id per padded slot (padding duplicates a valid token,
// write-back is cut at block_len).

#include "common.h"

#define MOE_MB 16  // slots per block

extern "C" __global__ void __launch_bounds__(256)
moe_gateup_kernel(const short* __restrict__ x, const short* __restrict__ w,
                  const int* __restrict__ slot_token,
                  const int* __restrict__ block_expert,
                  const int* __restrict__ block_len,
                  short* __restrict__ act, int H, int I) {
  const int blk = blockIdx.x;
  const int i = blockIdx.y * 256 + threadIdx.x;  // intermediate column
  const int e = block_expert[blk];
  const int m = block_len[blk];
  const int s0 = blk * MOE_MB;
  const bool vi = i < I;

  __shared__ float xs[MOE_MB][64];

  float ag[MOE_MB], au[MOE_MB];
#pragma unroll
  for (int mm = 0; mm < MOE_MB; ++mm) { ag[mm] = 0.f; au[mm] = 0.f; }

  const long wbase = (long)e * H * (2 * I);
  for (int h0 = 0; h0 < H; h0 += 64) {
    __syncthreads();
    for (int p = threadIdx.x; p < MOE_MB * 64; p += 256) {
      int mm = p >> 6, hh = p & 63;
      xs[mm][hh] = bits2f(x[(long)slot_token[s0 + mm] * H + h0 + hh]);
    }
    __syncthreads();
    if (vi) {
      for (int hh = 0; hh < 64; ++hh) {
        const long row = wbase + (long)(h0 + hh) * (2 * I);
        float wg = bits2f(w[row + i]);
        float wu = bits2f(w[row + I + i]);
#pragma unroll
        for (int mm = 0; mm < MOE_MB; ++mm) {
          ag[mm] += xs[mm][hh] * wg;
          au[mm] += xs[mm][hh] * wu;
        }
      }
    }
  }
  if (vi) {
    for (int mm = 0; mm < m; ++mm) {
      float g = ag[mm];
      float s = g / (1.f + __builtin_amdgcn_exp2f(-g * 1.4426950408889634f));
      act[(long)(s0 + mm) * I + i] = f2bits(s * au[mm]);
    }
  }
}

extern "C" __global__ void __launch_bounds__(256)
moe_down_kernel(const short* __restrict__ act, const short* __restrict__ wd,
                const int* __restrict__ slot_token,
                const int* __restrict__ block_expert,
                const int* __restrict__ block_len,
                const float* __restrict__ aff, float* __restrict__ out,
                int I, int H) {
  const int blk = blockIdx.x;
  const int h = blockIdx.y * 256 + threadIdx.x;  // output column
  const int e = block_expert[blk];
  const int m = block_len[blk];
  const int s0 = blk * MOE_MB;
  const bool vh = h < H;

  __shared__ float as[MOE_MB][64];

  float acc[MOE_MB];
#pragma unroll
  for (int mm = 0; mm < MOE_MB; ++mm) acc[mm] = 0.f;

  const long wbase = (long)e * I * H;
  for (int i0 = 0; i0 < I; i0 += 64) {
    __syncthreads();
    for (int p = threadIdx.x; p < MOE_MB * 64; p += 256) {
      int mm = p >> 6, ii = p & 63;
      as[mm][ii] = bits2f(act[(long)(s0 + mm) * I + i0 + ii]);
    }
    __syncthreads();
    if (vh) {
      for (int ii = 0; ii < 64; ++ii) {
        float wv = bits2f(wd[wbase + (long)(i0 + ii) * H + h]);
#pragma unroll
        for (int mm = 0; mm < MOE_MB; ++mm) acc[mm] += as[mm][ii] * wv;
      }
    }
  }
  if (vh) {
    for (int mm = 0; mm < m; ++mm) {
      atomicAdd(&out[(long)slot_token[s0 + mm] * H + h],
                acc[mm] * aff[s0 + mm]);
    }
  }
}

extern "C" void moe_decode_glu(const void* x, const void* w_gu,
                               const void* w_down, const void* slot_token,
                               const void* block_expert,
                               const void* block_len, const void* aff,
                               void* act, void* out, int n_blocks, int H,
                               int I, hipStream_t stream) {
  dim3 g1(n_blocks, (I + 255) / 256);
  moe_gateup_kernel<<<g1, 256, 0, stream>>>(
      (const short*)x, (const short*)w_gu, (const int*)slot_token,
      (const int*)block_expert, (const int*)block_len, (short*)act, H, I);
  dim3 g2(n_blocks, (H + 255) / 256);
  moe_down_kernel<<<g2, 256, 0, stream>>>(
      (const short*)act, (const short*)w_down, (const int*)slot_token,
      (const int*)block_expert, (const int*)block_len, (const float*)aff,
      (float*)out, I, H);
}
