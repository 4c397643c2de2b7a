// FlashAttention forward for CDNA4 (gfx950) — hand-written MFMA kernel.
//
// Replaces the reference's NKI flash kernels K1 (SURVEY.md §2.3,
// reference kernels/flash_attn.py:18,51-63).  MI355X-first design per
// cdna_hip_programming.md §B:
//   * 8 waves/WG, each wave owns 32 q rows (WG = 256 q rows); KV tiles of
//     64 double-buffered in LDS with the (row&7)<<4 XOR swizzle (G4).
//   * swapped QK^T — mfma(A=K, B=Q^T) — so each lane's softmax stats are
//     for ONE q row (l&31): row max / rescale / denom are lane-local
//     (+ one shfl_xor with the partner half-wave).
//   * P is repacked to bf16 PV B-fragments with pack_bf16x2 +
//     v_permlane32_swap (T12/T21 primitive).
//   * O is accumulated TRANSPOSED (O^T[d][q], q = lane) so the online
//     rescale is a per-lane scalar multiply; V is staged transposed (Vt).
//   * GQA: kv head = q head / (Hq/Hkv); causal masking per element on
//     diagonal tiles, whole-tile skip below the diagonal.
//
// Layouts: q,k,v,out (B, H, S, D) bf16 contiguous, D = 128; lse (B,Hq,S)
// f32 (natural-log row logsumexp, saved for backward).

#include "common.h"
#include "mfma.h"

#define FA_D 128
#define FA_QW 32       // q rows per wave
#define FA_WAVES 8
#define FA_QBLK (FA_QW * FA_WAVES)  // 256 q rows per workgroup
#define FA_KV 64       // kv tile
#define LOG2E 1.4426950408889634f
#define NEG_INF (-1e30f)

// LDS: double-buffered K [64][128] bf16 (256B rows, swz16 = conflict-free
// b128 reads) and Vt [128][72] bf16 (144B padded rows: 16B-aligned, and
// 36*d = 4*(9d mod 16) banks are distinct within every b128 lane group
// since 9 is coprime to 16 -> conflict-free, no swizzle needed)
#define K_TILE_B (FA_KV * FA_D * 2)    // 16 KB
#define VT_PITCH 72
#define VT_TILE_B (FA_D * VT_PITCH * 2)  // 18 KB

extern "C" __global__ void __launch_bounds__(512, 2)
flash_fwd_kernel(const short* __restrict__ qp, const short* __restrict__ kp,
                 const short* __restrict__ vp, short* __restrict__ op,
                 float* __restrict__ lsep, int B, int Hq, int Hkv, int S,
                 float scale, int causal, int window,
                 long q_bs, long q_hs, long q_ss,
                 long k_bs, long k_hs, long k_ss,
                 long v_bs, long v_hs, long v_ss,
                 long o_bs, long o_hs, long o_ss) {
  // (*_bs, *_hs, *_ss) = element strides of (batch, head, seq); the last
  // dim (d) is always dense.  BHSD contiguous: (H*S*128, S*128, 128).
  // BSHD transpose views (no-copy model layout): (S*H*128, 128, H*128).
  // window > 0: Mistral-style sliding window — q row i attends kv rows
  // [i - window + 1, i] (causal implied); whole tiles outside the band
  // are skipped, boundary tiles masked per element.
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // THREE rotating K/Vt buffers: tile t is read from buf t%3 while t+1 is
  // written into (t+1)%3, whose previous readers (tile t-2) finished two
  // barriers ago -> ONE barrier per tile instead of two.
  auto kbuf = [&](int i) { return smem + (size_t)i * K_TILE_B; };
  auto vbuf = [&](int i) {
    return smem + 3 * K_TILE_B + (size_t)i * VT_TILE_B;
  };

  const int lane = threadIdx.x & 63;
  // readfirstlane: provably wave-uniform -> scalar branches for the
  // per-wave activity guards instead of exec-mask divergence (T20)
  const int wid = __builtin_amdgcn_readfirstlane(threadIdx.x >> 6);
  const int col = lane & 31;       // q column owned by this lane
  const int hi = lane >> 5;

  // Dispatch mapping (causal LPT): heads on x, q-blocks on y REVERSED so
  // the largest-causal-work blocks launch first.  With x = q-block the
  // hardware pairs CU c with the SAME q-block index every occupancy round
  // (512 WGs / 256 CUs at 1 WG/CU) -> worst CUs do ~1.9x the mean causal
  // work while others idle; biggest-first lets early finishers absorb the
  // small diagonal blocks (greedy LPT, near-ideal pairing).
  const int qblk = (int)(gridDim.y - 1 - blockIdx.y);
  const int h = blockIdx.x;
  const int b = blockIdx.z;
  const int hkv = h / (Hq / Hkv);

  const long q_base = (long)b * q_bs + (long)h * q_hs;
  const long kv_base_k = (long)b * k_bs + (long)hkv * k_hs;
  const long kv_base_v = (long)b * v_bs + (long)hkv * v_hs;

  const int q0 = qblk * FA_QBLK;
  const int qw0 = q0 + wid * FA_QW;      // this wave's first q row
  const int my_q = qw0 + col;            // this lane's q row
  const int q_row_ld = my_q < S ? my_q : S - 1;

  // ---- Q fragments: 8 chunks of (16 d), each lane 8 bf16 --------------
  frag_u qf[8];
#pragma unroll
  for (int c = 0; c < 8; ++c) {
    const short* src = qp + q_base + (long)q_row_ld * q_ss + c * 16 + hi * 8;
    qf[c].u4 = *(const uint4v*)src;
  }

  // ---- accumulators ----------------------------------------------------
  f32x16 ot[4] = {};          // O^T: d-block nb, rows d_local, col q
  float m_run = NEG_INF;      // running max (exp2 domain)
  float l_run = 0.f;

  const int kv_end = causal ? min(S, q0 + FA_QBLK) : S;
  const int ntiles = (kv_end + FA_KV - 1) / FA_KV;
  // this wave can skip tiles fully above its causal row range
  const int my_kv_end = causal ? min(S, qw0 + FA_QW) : S;
  // sliding window: tiles fully below ANY of this wave's rows' windows
  // are skipped; the whole WG starts at the block's earliest window tile
  const int my_kv_begin = window > 0 ? max(0, qw0 - window + 1) : 0;
  const int t_begin = window > 0 ? max(0, q0 - window + 1) / FA_KV : 0;

  const float s2 = scale * LOG2E;

  // ---- staging: thread t owns K/V rows {2rp, 2rp+1} at 16B slot c16 ----
  // (rp = t>>4 in 0..31, c16 = t&15).  K: 2 x ds_write_b128 swizzled;
  // V: the row PAIR transposes into 8 x b32 writes (two k-columns per
  // write) instead of 16 scalar b16 scatters.
  const int tid = threadIdx.x;
  const int st_rp = tid >> 4;
  const int st_c16 = tid & 15;
  const int st_r0 = 2 * st_rp, st_r1 = 2 * st_rp + 1;

  uint4v kreg[2], vreg[2];
  auto issue_loads = [&](int t) {
    int kv0 = t * FA_KV;
    int rr0 = kv0 + st_r0 < S ? kv0 + st_r0 : S - 1;
    int rr1 = kv0 + st_r1 < S ? kv0 + st_r1 : S - 1;
    kreg[0] = *(const uint4v*)(kp + kv_base_k + (long)rr0 * k_ss + st_c16 * 8);
    kreg[1] = *(const uint4v*)(kp + kv_base_k + (long)rr1 * k_ss + st_c16 * 8);
    vreg[0] = *(const uint4v*)(vp + kv_base_v + (long)rr0 * v_ss + st_c16 * 8);
    vreg[1] = *(const uint4v*)(vp + kv_base_v + (long)rr1 * v_ss + st_c16 * 8);
  };

  auto write_tile = [&](int buf) {
    *(uint4v*)(kbuf(buf) + swz16(st_r0, st_c16 * 16)
               + st_r0 * (FA_D * 2)) = kreg[0];
    *(uint4v*)(kbuf(buf) + swz16(st_r1, st_c16 * 16)
               + st_r1 * (FA_D * 2)) = kreg[1];
    union { uint4v u; short s[8]; } a, b;
    a.u = vreg[0];
    b.u = vreg[1];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int d = st_c16 * 8 + j;
      uint pair = ((uint)(unsigned short)a.s[j]) |
                  (((uint)(unsigned short)b.s[j]) << 16);
      *(uint*)(vbuf(buf) + d * (VT_PITCH * 2) + st_r0 * 2) = pair;
    }
  };

  issue_loads(t_begin);
  write_tile(t_begin % 3);
  __syncthreads();

  for (int t = t_begin; t < ntiles; ++t) {
    const int kv0 = t * FA_KV;
    const int cur = t % 3;
    if (t + 1 < ntiles) issue_loads(t + 1);

    const bool wave_active = kv0 < my_kv_end &&
        (window <= 0 || kv0 + FA_KV > my_kv_begin);
    if (wave_active) {
      // ---- QK^T: S^T[k][q] = sum_d K[k][d] Q^T[d][q] ------------------
      // All 16 K fragments are prefetched into registers BEFORE the mfma
      // burst (T3/T4): with ds_reads issued only 1-2 mfmas ahead the
      // compiler emits lgkmcnt(0) waits before every other mfma and the
      // ~64-cycle LDS latency stalls the matrix pipe; batching the reads
      // turns those into counted waits that are already satisfied.
      f32x16 acc[2] = {};
      frag_u kfr[2][8];
#pragma unroll
      for (int kb = 0; kb < 2; ++kb) {
        int row = col + 32 * kb;
#pragma unroll
        for (int c = 0; c < 8; ++c)
          kfr[kb][c].u4 = *(const uint4v*)(kbuf(cur) + row * (FA_D * 2)
                                           + swz16(row, (c * 16 + hi * 8) * 2));
      }
      __builtin_amdgcn_s_setprio(1);  // T5: keep the matrix pipe fed
#pragma unroll
      for (int kb = 0; kb < 2; ++kb)
#pragma unroll
        for (int c = 0; c < 8; ++c)
          acc[kb] = mfma_bf16(kfr[kb][c].bf, qf[c].bf, acc[kb]);
      __builtin_amdgcn_s_setprio(0);

      // ---- online softmax (exp2 domain), lane-local per q row ---------
      float sc[2][16];
      const bool need_mask =
          (causal && kv0 + FA_KV > qw0) || (kv0 + FA_KV > S) ||
          (window > 0 && kv0 < qw0 + FA_QW - 1 - window + 1 + FA_KV);
#pragma unroll
      for (int kb = 0; kb < 2; ++kb)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          float v = acc[kb][r] * s2;
          if (need_mask) {
            int kg = kv0 + 32 * kb + acc_row(r, hi);
            if ((causal && kg > my_q) || kg >= S ||
                (window > 0 && kg <= my_q - window)) v = NEG_INF;
          }
          sc[kb][r] = v;
        }

      float mt = NEG_INF;
#pragma unroll
      for (int kb = 0; kb < 2; ++kb)
#pragma unroll
        for (int r = 0; r < 16; ++r) mt = fmaxf(mt, sc[kb][r]);
      mt = fmaxf(mt, __shfl_xor(mt, 32, 64));  // combine partner half

      float m_new = fmaxf(m_run, mt);
      // m_eff floor: when a lane's rows are ALL masked so far, m_new is
      // -1e30 and exp2(sc - m_new) would be exp2(0)=1 for masked scores;
      // the floor keeps those at exp2(-9.9e29) = 0.
      float m_eff = fmaxf(m_new, -1e28f);
      float alpha = __builtin_amdgcn_exp2f(m_run - m_eff);  // 0 when m_run=-1e30
      m_run = m_new;

      float psum = 0.f;
#pragma unroll
      for (int kb = 0; kb < 2; ++kb)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          float p = __builtin_amdgcn_exp2f(sc[kb][r] - m_eff);
          sc[kb][r] = p;
          psum += p;
        }
      psum += __shfl_xor(psum, 32, 64);
      l_run = l_run * alpha + psum;

#pragma unroll
      for (int nb = 0; nb < 4; ++nb) ot[nb] *= alpha;

      // ---- pack P -> PV B-fragments (4 chunks of 16 k) ----------------
      frag_u pf[4];
#pragma unroll
      for (int kb = 0; kb < 2; ++kb)
#pragma unroll
        for (int cc = 0; cc < 2; ++cc) {
          uint b0 = pack_bf16x2(sc[kb][8 * cc + 0], sc[kb][8 * cc + 1]);
          uint b1 = pack_bf16x2(sc[kb][8 * cc + 2], sc[kb][8 * cc + 3]);
          uint b2 = pack_bf16x2(sc[kb][8 * cc + 4], sc[kb][8 * cc + 5]);
          uint b3 = pack_bf16x2(sc[kb][8 * cc + 6], sc[kb][8 * cc + 7]);
          {
            auto r01 = __builtin_amdgcn_permlane32_swap(b0, b2, false, false);
            b0 = r01[0]; b2 = r01[1];
          }
          {
            auto r23 = __builtin_amdgcn_permlane32_swap(b1, b3, false, false);
            b1 = r23[0]; b3 = r23[1];
          }
          frag_u& f = pf[2 * kb + cc];
          f.u[0] = b0; f.u[1] = b1; f.u[2] = b2; f.u[3] = b3;
        }

      // ---- PV: O^T[d][q] += V^T[d][k] P^T[k][q] -----------------------
      // Same batched-prefetch structure as QK, two d-blocks at a time
      // (8 fragments = 32 VGPRs in flight).
#pragma unroll
      for (int np = 0; np < 2; ++np) {
        frag_u vfr[2][4];
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          int d = 32 * (2 * np + ni) + col;
#pragma unroll
          for (int c16 = 0; c16 < 4; ++c16)
            vfr[ni][c16].u4 = *(const uint4v*)(vbuf(cur) + d * (VT_PITCH * 2)
                                               + (c16 * 16 + hi * 8) * 2);
        }
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
#pragma unroll
          for (int c16 = 0; c16 < 4; ++c16)
            ot[2 * np + ni] =
                mfma_bf16(vfr[ni][c16].bf, pf[c16].bf, ot[2 * np + ni]);
        __builtin_amdgcn_s_setprio(0);
      }
    }

    if (t + 1 < ntiles) write_tile((t + 1) % 3);
    __syncthreads();
  }

  // ---- epilogue: normalize, store O (transposed back) and LSE ---------
  if (my_q >= S) return;
  float inv_l = l_run > 0.f ? 1.f / l_run : 0.f;
  long o_row = (long)b * o_bs + (long)h * o_hs + (long)my_q * o_ss;
#pragma unroll
  for (int nb = 0; nb < 4; ++nb) {
#pragma unroll
    for (int rg = 0; rg < 4; ++rg) {
      int d = 32 * nb + 8 * rg + 4 * hi;
      uint w0 = pack_bf16x2(ot[nb][4 * rg + 0] * inv_l,
                            ot[nb][4 * rg + 1] * inv_l);
      uint w1 = pack_bf16x2(ot[nb][4 * rg + 2] * inv_l,
                            ot[nb][4 * rg + 3] * inv_l);
      uint2 st = {w0, w1};
      *(uint2*)(op + o_row + d) = st;
    }
  }
  if (hi == 0 && lsep) {
    // natural-log LSE: scores were in exp2 domain
    lsep[(long)(b * Hq + h) * S + my_q] =
        m_run * 0.6931471805599453f + __logf(l_run);
  }
}

extern "C" void flash_attn_fwd(const void* q, const void* k, const void* v,
                               void* out, void* lse, int B, int Hq, int Hkv,
                               int S, float scale, int causal,
                               hipStream_t stream) {
  dim3 grid(Hq, (S + FA_QBLK - 1) / FA_QBLK, B);
  size_t lds = 3 * (K_TILE_B + VT_TILE_B);
  flash_fwd_kernel<<<grid, 512, lds, stream>>>(
      (const short*)q, (const short*)k, (const short*)v, (short*)out,
      (float*)lse, B, Hq, Hkv, S, scale, causal, 0,
      (long)Hq * S * FA_D, (long)S * FA_D, FA_D,
      (long)Hkv * S * FA_D, (long)S * FA_D, FA_D,
      (long)Hkv * S * FA_D, (long)S * FA_D, FA_D,
      (long)Hq * S * FA_D, (long)S * FA_D, FA_D);
}

extern "C" void flash_attn_fwd_strided(
    const void* q, const void* k, const void* v, void* out, void* lse,
    int B, int Hq, int Hkv, int S, float scale, int causal, int window,
    const long* st, hipStream_t stream) {
  // st = 12 longs: (bs, hs, ss) x (q, k, v, o)
  dim3 grid(Hq, (S + FA_QBLK - 1) / FA_QBLK, B);
  size_t lds = 3 * (K_TILE_B + VT_TILE_B);
  flash_fwd_kernel<<<grid, 512, lds, stream>>>(
      (const short*)q, (const short*)k, (const short*)v, (short*)out,
      (float*)lse, B, Hq, Hkv, S, scale, causal, window,
      st[0], st[1], st[2], st[3], st[4], st[5], st[6], st[7], st[8],
      st[9], st[10], st[11]);
}

extern "C" void flash_attn_fwd_window(const void* q, const void* k,
                                      const void* v, void* out, void* lse,
                                      int B, int Hq, int Hkv, int S,
                                      float scale, int window,
                                      hipStream_t stream) {
  dim3 grid(Hq, (S + FA_QBLK - 1) / FA_QBLK, B);
  size_t lds = 3 * (K_TILE_B + VT_TILE_B);
  flash_fwd_kernel<<<grid, 512, lds, stream>>>(
      (const short*)q, (const short*)k, (const short*)v, (short*)out,
      (float*)lse, B, Hq, Hkv, S, scale, 1, window,
      (long)Hq * S * FA_D, (long)S * FA_D, FA_D,
      (long)Hkv * S * FA_D, (long)S * FA_D, FA_D,
      (long)Hkv * S * FA_D, (long)S * FA_D, FA_D,
      (long)Hq * S * FA_D, (long)S * FA_D, FA_D);
}
