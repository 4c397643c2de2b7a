// Empirical MFMA layout probe for v_mfma_f32_32x32x16_bf16 (gfx950).
// Launch with ONE wave (64 threads).
//
// probe_c: A,B loaded via the ASSUMED maps (A: m=l&31,k=8hi+j; B: n=l&31,
//   k=8hi+j) from matrices chosen so D[m][n] = m*32+n for ANY consistent
//   k-permutation -> raw accumulator dump reveals the true C map.
// probe_k: B set register-direct to 2^j, A register-direct to delta(j==J0)
//   -> D[m][n] = sum over A's true k slots of B value at that k for col n,
//   revealing the relation between element index j and true k.

#include "common.h"
#include "mfma.h"

extern "C" __global__ void probe_c_kernel(const short* A, const short* B,
                                          float* out) {
  int l = threadIdx.x;
  int hi = l >> 5;
  frag_u a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a.bf[j] = *(const __bf16*)(A + (l & 31) * 16 + (8 * hi + j));
    b.bf[j] = *(const __bf16*)(B + (8 * hi + j) * 32 + (l & 31));
  }
  f32x16 acc = {};
  acc = mfma_bf16(a.bf, b.bf, acc);
#pragma unroll
  for (int r = 0; r < 16; ++r) out[l * 16 + r] = acc[r];
}

extern "C" __global__ void probe_k_kernel(float* out, int J0) {
  int l = threadIdx.x;
  frag_u a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a.bf[j] = (j == J0) ? (__bf16)1.0f : (__bf16)0.0f;
    b.bf[j] = (__bf16)(float)(1 << j);   // 2^j, exact in bf16
  }
  f32x16 acc = {};
  acc = mfma_bf16(a.bf, b.bf, acc);
#pragma unroll
  for (int r = 0; r < 16; ++r) out[l * 16 + r] = acc[r];
}

extern "C" void run_probe_c(const void* A, const void* B, void* out,
                            hipStream_t s) {
  probe_c_kernel<<<1, 64, 0, s>>>((const short*)A, (const short*)B,
                                  (float*)out);
}
extern "C" void run_probe_k(void* out, int J0, hipStream_t s) {
  probe_k_kernel<<<1, 64, 0, s>>>((float*)out, J0);
}
