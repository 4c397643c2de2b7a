// Fused decode attention for CDNA4: RoPE(q,k) + KV-cache append +
// flash-decode over the cache, GQA-grouped, split-KV two-phase.
//
// Performance history at the llama3-8b decode shape (B=32, Hkv=8,
// pos~1k; cache-read floor ~17 us):
//   v1  4 waves/WG, lane-per-2-dims, 6-step wave shuffle per row: 171 us
//   v2  16 waves + prefetch:                                       70 us
//   v3  + split-KV x4 two-phase:                                   80 us
//       (ablation: shuffle reduction ~32 us of it -> the ds-pipe
//        __shfl chain, not memory, was the wall)
//   v6  THIS: each 16-lane QUARTER-wave owns a row (8 dims/lane,
//       b128 loads), so the score reduction is 4 xor steps serving 4
//       rows per instruction, exp2 runs once per score, and a wave
//       retires 4 rows/iteration:                                  31 us
//
// Phase 1 (decode_attn_part): WG (b, kvh, split s) flash-decodes its
// pos/SPLIT row chunk, merges its quarter-wave partials in LDS, writes
// one fp32 partial (m, l, o[REP][128]) to workspace; split 0 also
// appends the new K/V row and adds its term.  Phase 2
// (decode_attn_merge) log-sum-exp-combines the splits -> bf16 out.
//
// Layouts (bf16 unless noted): q_lin (B, Hq*D) row-stride qstride,
// k_lin/v_lin (B, Hkv*D) row-stride kvstride (strided fused-QKV views
// feed directly); kcache/vcache (B, Hkv, Smax, D); cos/sin (Smax, D/2)
// fp32; pos_ptr device int64 (hipGraph-replayable); workspace part_o
// (B*Hkv*SPLIT, REP, D) fp32 + part_ml (..., 2) fp32.  D = 128.

#include "common.h"
#include "mfma.h"

#define DA_D 128
#define LOG2E 1.4426950408889634f

#define DA_WAVES 4  // 4 waves x 4 quarter-rows = 16 rows in flight per WG
#define DA_SPLIT 4  // KV-range splits per (batch, kv-head)
#define DA_NAME decode_attn

template <int REP>
__global__ void __launch_bounds__(DA_WAVES * 64)
decode_attn_part(const short* __restrict__ qlin,
                    const short* __restrict__ klin,
                    const short* __restrict__ vlin,
                    short* __restrict__ kcache, short* __restrict__ vcache,
                    const float* __restrict__ cosp,
                    const float* __restrict__ sinp,
                    const long* __restrict__ pos_ptr,
                    float* __restrict__ part_o, float* __restrict__ part_ml,
                    int B, int Hq, int Hkv, int Smax, float scale,
                    int qstride, int kvstride) {
  const int wg = blockIdx.x;
  const int s = wg % DA_SPLIT;
  const int pair = wg / DA_SPLIT;
  const int kvh = pair % Hkv;
  const int b = pair / Hkv;
  const int qh0 = kvh * REP;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int half = (lane >> 4);      // 0..3: which row of the quad
  const int hl = lane & 15;          // lane within quarter; 8 dims each
  const int wid = __builtin_amdgcn_readfirstlane(tid >> 6);
  const long pos = *pos_ptr;
  const long chunk = (pos + DA_SPLIT - 1) / DA_SPLIT;
  const long r0 = (long)s * chunk;
  const long r1 = min(pos, r0 + chunk);

  __shared__ float qr[REP][DA_D];
  __shared__ float knr[DA_D];
  __shared__ float vn[DA_D];
  __shared__ float merge_o[4 * DA_WAVES + 1][REP][DA_D];
  __shared__ float merge_ml[4 * DA_WAVES + 1][REP][2];

  if (tid < 64) {
    const float c = cosp[pos * (DA_D / 2) + tid];
    const float sn = sinp[pos * (DA_D / 2) + tid];
#pragma unroll
    for (int g = 0; g < REP; ++g) {
      const long base = (long)b * qstride + (qh0 + g) * DA_D;
      float x0 = bits2f(qlin[base + tid]);
      float x1 = bits2f(qlin[base + tid + 64]);
      qr[g][tid] = x0 * c - x1 * sn;
      qr[g][tid + 64] = x1 * c + x0 * sn;
    }
    {
      const long base = (long)b * kvstride + kvh * DA_D;
      float x0 = bits2f(klin[base + tid]);
      float x1 = bits2f(klin[base + tid + 64]);
      knr[tid] = x0 * c - x1 * sn;
      knr[tid + 64] = x1 * c + x0 * sn;
    }
  } else if (tid < 128) {
    int i = tid - 64;
    vn[i] = bits2f(vlin[(long)b * kvstride + kvh * DA_D + i]);
    vn[i + 64] = bits2f(vlin[(long)b * kvstride + kvh * DA_D + i + 64]);
  }
  __syncthreads();

  if (s == 0 && tid < DA_D) {
    const long cache_row = (((long)b * Hkv + kvh) * Smax + pos) * DA_D;
    kcache[cache_row + tid] = f2bits(knr[tid]);
    vcache[cache_row + tid] = f2bits(vn[tid]);
  }

  const short* kc = kcache + ((long)b * Hkv + kvh) * Smax * DA_D;
  const short* vc = vcache + ((long)b * Hkv + kvh) * Smax * DA_D;
  const float s2 = scale * LOG2E;
  // this lane's 8 q dims (per head): dims 8*hl .. 8*hl+7
  float qv[REP][8];
#pragma unroll
  for (int g = 0; g < REP; ++g)
#pragma unroll
    for (int j = 0; j < 8; ++j)
      qv[g][j] = qr[g][8 * hl + j];

  float m_run[REP], l_run[REP], ov[REP][8];
#pragma unroll
  for (int g = 0; g < REP; ++g) {
    m_run[g] = -1e30f;
    l_run[g] = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) ov[g][j] = 0.f;
  }

  // row quad per wave-iteration: row = base + wid*4 + quarter
  long r = r0 + 4 * wid + half;
  const long rstep = 4 * DA_WAVES;
  uint4v kp = {0, 0, 0, 0}, vp = {0, 0, 0, 0};
  if (r < r1) {
    kp = *(const uint4v*)(kc + r * DA_D + 8 * hl);
    vp = *(const uint4v*)(vc + r * DA_D + 8 * hl);
  }
  for (; r < r1;) {
    const long rn = r + rstep;
    uint4v kpn = {0, 0, 0, 0}, vpn = {0, 0, 0, 0};
    if (rn < r1) {
      kpn = *(const uint4v*)(kc + rn * DA_D + 8 * hl);
      vpn = *(const uint4v*)(vc + rn * DA_D + 8 * hl);
    }
    float kf[8], vf[8];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      kf[2 * j] = bits2f((short)(kp[j] & 0xffff));
      kf[2 * j + 1] = bits2f((short)(kp[j] >> 16));
      vf[2 * j] = bits2f((short)(vp[j] & 0xffff));
      vf[2 * j + 1] = bits2f((short)(vp[j] >> 16));
    }
    float part[REP];
#pragma unroll
    for (int g = 0; g < REP; ++g) {
      part[g] = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j) part[g] += qv[g][j] * kf[j];
    }
    // 4-step xor reduce WITHIN each 16-lane quarter (all 4 rows per op)
#pragma unroll
    for (int off = 8; off > 0; off >>= 1)
#pragma unroll
      for (int g = 0; g < REP; ++g)
        part[g] += __shfl_xor(part[g], off, 64);
#pragma unroll
    for (int g = 0; g < REP; ++g) {
      float sc = part[g] * s2;
      float m_new = fmaxf(m_run[g], sc);
      float alpha = __builtin_amdgcn_exp2f(m_run[g] - m_new);
      float p = __builtin_amdgcn_exp2f(sc - m_new);
      m_run[g] = m_new;
      l_run[g] = l_run[g] * alpha + p;
#pragma unroll
      for (int j = 0; j < 8; ++j) ov[g][j] = ov[g][j] * alpha + p * vf[j];
    }
    r = rn;
    kp = kpn;
    vp = vpn;
  }

  // merge: slot = 4*wid + quarter (each quarter-wave is an independent
  // partial covering dims 8*hl..8*hl+7)
  const int slot = 4 * wid + half;
#pragma unroll
  for (int g = 0; g < REP; ++g) {
#pragma unroll
    for (int j = 0; j < 8; ++j)
      merge_o[slot][g][8 * hl + j] = ov[g][j];
    if (hl == 0) {
      merge_ml[slot][g][0] = m_run[g];
      merge_ml[slot][g][1] = l_run[g];
    }
  }
  const int nslot = 4 * DA_WAVES + (s == 0 ? 1 : 0);
  if (s == 0 && wid == 0) {
    float part[REP];
#pragma unroll
    for (int g = 0; g < REP; ++g) {
      part[g] = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        part[g] += qv[g][j] * knr[8 * hl + j];
    }
#pragma unroll
    for (int off = 8; off > 0; off >>= 1)
#pragma unroll
      for (int g = 0; g < REP; ++g)
        part[g] += __shfl_xor(part[g], off, 64);
    // all quarters computed the same new-row score; each writes its dims
#pragma unroll
    for (int g = 0; g < REP; ++g) {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        merge_o[4 * DA_WAVES][g][8 * hl + j] = vn[8 * hl + j];
      if (lane == 0) {
        merge_ml[4 * DA_WAVES][g][0] = part[g] * s2;
        merge_ml[4 * DA_WAVES][g][1] = 1.f;
      }
    }
  }
  __syncthreads();

  for (int g = wid; g < REP; g += DA_WAVES) {
    // lane covers 2 dims like before (64 lanes x 2 = 128)
    float m_g = -1e30f;
    for (int w = 0; w < nslot; ++w)
      m_g = fmaxf(m_g, merge_ml[w][g][0]);
    float l_g = 0.f, a0 = 0.f, a1 = 0.f;
    for (int w = 0; w < nslot; ++w) {
      float sw = __builtin_amdgcn_exp2f(merge_ml[w][g][0] - m_g);
      l_g += merge_ml[w][g][1] * sw;
      a0 += merge_o[w][g][2 * lane] * sw;
      a1 += merge_o[w][g][2 * lane + 1] * sw;
    }
    float* po = part_o + ((long)wg * REP + g) * DA_D;
    po[2 * lane] = a0;
    po[2 * lane + 1] = a1;
    if (lane == 0) {
      part_ml[((long)wg * REP + g) * 2 + 0] = l_g > 0.f ? m_g : -1e30f;
      part_ml[((long)wg * REP + g) * 2 + 1] = l_g;
    }
  }
}

template <int REP>
__global__ void __launch_bounds__(256)
decode_attn_merge(const float* __restrict__ part_o,
                     const float* __restrict__ part_ml,
                     short* __restrict__ outp, int B, int Hq, int Hkv) {
  const int pair = blockIdx.x;
  const int kvh = pair % Hkv;
  const int b = pair / Hkv;
  const int qh0 = kvh * REP;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  for (int g = wid; g < REP; g += 4) {
    float m_g = -1e30f;
#pragma unroll
    for (int sp = 0; sp < DA_SPLIT; ++sp)
      m_g = fmaxf(m_g,
                  part_ml[(((long)pair * DA_SPLIT + sp) * REP + g) * 2]);
    float l_g = 0.f, a0 = 0.f, a1 = 0.f;
#pragma unroll
    for (int sp = 0; sp < DA_SPLIT; ++sp) {
      const long idx = ((long)pair * DA_SPLIT + sp) * REP + g;
      const float mw = part_ml[idx * 2];
      const float sw = __builtin_amdgcn_exp2f(mw - m_g);
      l_g += part_ml[idx * 2 + 1] * sw;
      const float* po = part_o + idx * DA_D;
      a0 += po[2 * lane] * sw;
      a1 += po[2 * lane + 1] * sw;
    }
    float inv = 1.f / l_g;
    uint outpair = pack_bf16x2(a0 * inv, a1 * inv);
    *(uint*)(outp + (long)b * Hq * DA_D + (qh0 + g) * DA_D + 2 * lane) =
        outpair;
  }
}

extern "C" void decode_attn(const void* qlin, const void* klin,
                        const void* vlin, void* kcache, void* vcache,
                        const void* cosp, const void* sinp,
                        const void* pos_ptr, void* part_o, void* part_ml,
                        void* outp, int B, int Hq, int Hkv, int Smax,
                        float scale, int qstride, int kvstride,
                        hipStream_t stream) {
  const int rep = Hq / Hkv;
  dim3 grid1(B * Hkv * DA_SPLIT);
  dim3 grid2(B * Hkv);
#define LAUNCH(R)                                                        \
  do {                                                                   \
    decode_attn_part<R><<<grid1, DA_WAVES * 64, 0, stream>>>(         \
        (const short*)qlin, (const short*)klin, (const short*)vlin,      \
        (short*)kcache, (short*)vcache, (const float*)cosp,              \
        (const float*)sinp, (const long*)pos_ptr, (float*)part_o,        \
        (float*)part_ml, B, Hq, Hkv, Smax, scale, qstride, kvstride);    \
    decode_attn_merge<R><<<grid2, 256, 0, stream>>>(                  \
        (const float*)part_o, (const float*)part_ml, (short*)outp, B,    \
        Hq, Hkv);                                                        \
  } while (0)
  switch (rep) {
    case 1: LAUNCH(1); break;
    case 2: LAUNCH(2); break;
    case 4: LAUNCH(4); break;
    case 8: LAUNCH(8); break;
    default: LAUNCH(1); break;
  }
#undef LAUNCH
}
