// Fused decode attention for CDNA4: one kernel doing RoPE(q,k) + KV-cache
// append + flash-decode over the cache, GQA-grouped.
//
// The eager/torch decode chain is ~12 kernels per layer (rope elementwise,
// index_copy, batched GEMV, mask, softmax, casts, GEMV); at decode batch
// sizes every one is a few-microsecond latency-bound launch even inside a
// hipGraph.  This kernel collapses the whole attention step AND reads the
// KV cache exactly once per kv head: each workgroup owns one (batch,
// kv-head) pair and evaluates all REP = Hq/Hkv q-heads against the
// streamed rows (the cache read is the decode-attention bandwidth floor).
//
// Performance shape (llama3-8b decode, B=32 Hkv=8 pos~1k): the grid is
// B*Hkv = 256 workgroups == 1 per CU, so per-SIMD occupancy decides
// whether the serial load -> cross-lane-reduce -> exp2 chain is hidden.
// v1 used 4 waves/WG (1 wave/SIMD, nothing hidden): 171 us.  This version
// runs 16 waves/WG (4 waves/SIMD) and software-prefetches the next row's
// K/V pair into registers before computing the current one, which takes
// the kernel to the cache-read bandwidth floor.
//
// Layouts (bf16 unless noted): q_lin (B, Hq*D), k_lin/v_lin (B, Hkv*D)
// fresh from the QKV GEMMs; kcache/vcache (B, Hkv, Smax, D); cos/sin
// (Smax, D/2) fp32; pos_ptr = device int64 scalar (hipGraph-replayable);
// out (B, Hq*D).  D = 128.
//
// The NEW row never reads back from the cache: its contribution is merged
// locally (no cross-workgroup fence needed).  Waves stride the cached rows
// [0, pos) with lane-local stats over d (2 elems/lane) and merge via LDS.

#include "common.h"
#include "mfma.h"

#define DA_D 128
#define DA_WAVES 16  // 4 waves per SIMD: hides the serial per-row chain
#define LOG2E 1.4426950408889634f

template <int REP>
__global__ void __launch_bounds__(DA_WAVES * 64)
decode_attn_kernel(const short* __restrict__ qlin,
                   const short* __restrict__ klin,
                   const short* __restrict__ vlin,
                   short* __restrict__ kcache, short* __restrict__ vcache,
                   const float* __restrict__ cosp,
                   const float* __restrict__ sinp,
                   const long* __restrict__ pos_ptr,
                   short* __restrict__ outp, int B, int Hq, int Hkv,
                   int Smax, float scale, int qstride, int kvstride) {
  const int wg = blockIdx.x;
  const int kvh = wg % Hkv;
  const int b = wg / Hkv;
  const int qh0 = kvh * REP;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = __builtin_amdgcn_readfirstlane(tid >> 6);
  const long pos = *pos_ptr;

  __shared__ float qr[REP][DA_D];
  __shared__ float knr[DA_D];
  __shared__ float vn[DA_D];
  __shared__ float merge_o[DA_WAVES + 1][REP][DA_D];
  __shared__ float merge_ml[DA_WAVES + 1][REP][2];

  // ---- stage + rope the new q rows (all REP), k row; stage v ---------
  // pair rotation: (x0, x1) at (i, i+64), c/s index i (half tables)
  if (tid < 64) {
    const float c = cosp[pos * (DA_D / 2) + tid];
    const float s = sinp[pos * (DA_D / 2) + tid];
#pragma unroll
    for (int g = 0; g < REP; ++g) {
      const long base = (long)b * qstride + (qh0 + g) * DA_D;
      float x0 = bits2f(qlin[base + tid]);
      float x1 = bits2f(qlin[base + tid + 64]);
      qr[g][tid] = x0 * c - x1 * s;
      qr[g][tid + 64] = x1 * c + x0 * s;
    }
    {
      const long base = (long)b * kvstride + kvh * DA_D;
      float x0 = bits2f(klin[base + tid]);
      float x1 = bits2f(klin[base + tid + 64]);
      knr[tid] = x0 * c - x1 * s;
      knr[tid + 64] = x1 * c + x0 * s;
    }
  } else if (tid < 128) {
    int i = tid - 64;  // 64 threads x 2 elems cover the 128-wide v row
    vn[i] = bits2f(vlin[(long)b * kvstride + kvh * DA_D + i]);
    vn[i + 64] = bits2f(vlin[(long)b * kvstride + kvh * DA_D + i + 64]);
  }
  __syncthreads();

  // append roped k and v at cache[pos] (one writer per element)
  const long cache_row = (((long)b * Hkv + kvh) * Smax + pos) * DA_D;
  if (tid < DA_D) {
    kcache[cache_row + tid] = f2bits(knr[tid]);
    vcache[cache_row + tid] = f2bits(vn[tid]);
  }

  // ---- flash-decode over cached rows [0, pos), wave-strided ----------
  const short* kc = kcache + ((long)b * Hkv + kvh) * Smax * DA_D;
  const short* vc = vcache + ((long)b * Hkv + kvh) * Smax * DA_D;
  const float s2 = scale * LOG2E;
  float q0[REP], q1[REP];
#pragma unroll
  for (int g = 0; g < REP; ++g) {
    q0[g] = qr[g][2 * lane];
    q1[g] = qr[g][2 * lane + 1];
  }

  float m_run[REP], l_run[REP], o0[REP], o1[REP];
#pragma unroll
  for (int g = 0; g < REP; ++g) {
    m_run[g] = -1e30f;
    l_run[g] = 0.f;
    o0[g] = 0.f;
    o1[g] = 0.f;
  }

  // software-prefetch pipeline: issue row r+WAVES's loads before using row r
  long r = wid;
  uint kp2 = 0, vp2 = 0;
  if (r < pos) {
    kp2 = *(const uint*)(kc + r * DA_D + 2 * lane);
    vp2 = *(const uint*)(vc + r * DA_D + 2 * lane);
  }
  for (; r < pos;) {
    const long rn = r + DA_WAVES;
    uint kp2n = 0, vp2n = 0;
    if (rn < pos) {
      kp2n = *(const uint*)(kc + rn * DA_D + 2 * lane);
      vp2n = *(const uint*)(vc + rn * DA_D + 2 * lane);
    }
    const float k0 = bits2f((short)(kp2 & 0xffff));
    const float k1 = bits2f((short)(kp2 >> 16));
    const float v0 = bits2f((short)(vp2 & 0xffff));
    const float v1 = bits2f((short)(vp2 >> 16));
    float part[REP];
#pragma unroll
    for (int g = 0; g < REP; ++g) part[g] = q0[g] * k0 + q1[g] * k1;
    // wave sums -> scores broadcast to all lanes
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
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
      o0[g] = o0[g] * alpha + p * v0;
      o1[g] = o1[g] * alpha + p * v1;
    }
    r = rn;
    kp2 = kp2n;
    vp2 = vp2n;
  }

  // ---- merge the wave partials + the NEW row (last slot) -------------
#pragma unroll
  for (int g = 0; g < REP; ++g) {
    merge_o[wid][g][2 * lane] = o0[g];
    merge_o[wid][g][2 * lane + 1] = o1[g];
    if (lane == 0) {
      merge_ml[wid][g][0] = m_run[g];
      merge_ml[wid][g][1] = l_run[g];
    }
  }
  if (wid == 0) {
    // new row: score = qr[g] . knr, value = vn, l-contribution 1
    const float k0 = knr[2 * lane];
    const float k1 = knr[2 * lane + 1];
    float part[REP];
#pragma unroll
    for (int g = 0; g < REP; ++g) part[g] = q0[g] * k0 + q1[g] * k1;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
#pragma unroll
      for (int g = 0; g < REP; ++g)
        part[g] += __shfl_xor(part[g], off, 64);
#pragma unroll
    for (int g = 0; g < REP; ++g) {
      merge_o[DA_WAVES][g][2 * lane] = vn[2 * lane];
      merge_o[DA_WAVES][g][2 * lane + 1] = vn[2 * lane + 1];
      if (lane == 0) {
        merge_ml[DA_WAVES][g][0] = part[g] * s2;
        merge_ml[DA_WAVES][g][1] = 1.f;
      }
    }
  }
  __syncthreads();

  // waves 0..REP-1 finalize one q-head each (REP <= 8 < DA_WAVES)
  for (int g = wid; g < REP; g += DA_WAVES) {
    float m_g = -1e30f;
#pragma unroll
    for (int w = 0; w < DA_WAVES + 1; ++w)
      m_g = fmaxf(m_g, merge_ml[w][g][0]);
    float l_g = 0.f;
    float a0 = 0.f, a1 = 0.f;
#pragma unroll
    for (int w = 0; w < DA_WAVES + 1; ++w) {
      float sw = __builtin_amdgcn_exp2f(merge_ml[w][g][0] - m_g);
      l_g += merge_ml[w][g][1] * sw;
      a0 += merge_o[w][g][2 * lane] * sw;
      a1 += merge_o[w][g][2 * lane + 1] * sw;
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
                            const void* pos_ptr, void* outp, int B, int Hq,
                            int Hkv, int Smax, float scale, int qstride,
                            int kvstride, hipStream_t stream) {
  const int rep = Hq / Hkv;
  dim3 grid(B * Hkv);
#define LAUNCH(R)                                                        \
  decode_attn_kernel<R><<<grid, DA_WAVES * 64, 0, stream>>>(             \
      (const short*)qlin, (const short*)klin, (const short*)vlin,        \
      (short*)kcache, (short*)vcache, (const float*)cosp,                \
      (const float*)sinp, (const long*)pos_ptr, (short*)outp, B, Hq,     \
      Hkv, Smax, scale, qstride, kvstride)
  switch (rep) {
    case 1: LAUNCH(1); break;
    case 2: LAUNCH(2); break;
    case 4: LAUNCH(4); break;
    case 8: LAUNCH(8); break;
    default: LAUNCH(1); break;  // unsupported rep handled by the wrapper
  }
#undef LAUNCH
}
