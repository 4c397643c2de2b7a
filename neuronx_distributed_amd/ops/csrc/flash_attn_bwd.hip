// FlashAttention backward for CDNA4 (gfx950) — FA2-style recompute.
//
// Replaces the reference's NKI flash_attn_bwd (K2, SURVEY.md §2.3;
// reference kernels/flash_attn.py:18,76-88).  Three kernels:
//   * fa_bwd_delta: delta[q] = sum_d dO*O (rowwise fp32)
//   * fa_bwd_dkdv:  grid over 128-row KV blocks (4 waves x 32 kv rows);
//       sequential q-tile loop recomputes P^T from (Q,K,lse), forms
//       dS^T = P^T*(dP^T - delta), accumulates dK/dV in registers.
//       dK/dV written per Q-HEAD (B,Hq,S,D); the GQA reduction over the
//       Hq/Hkv replicas happens in the python wrapper.
//   * fa_bwd_dq:    grid over 128-row Q blocks (4 waves x 32 q rows);
//       sequential kv-tile loop, dQ accumulated in registers.
// No atomics anywhere: every output region is written by exactly one
// workgroup.
//
// Orientation trick shared with the forward kernel: all mfmas keep the
// softmax-normalized index (q) on the lane axis (col = l&31) so lse/delta
// are per-lane scalars; operands that need the other orientation go
// through small per-wave LDS tiles written from the accumulator layout.

#include "common.h"
#include "mfma.h"

// transposed tiles use a PADDED pitch instead of a swizzle: 80-byte rows
// (40 bf16) are 16B-aligned and put row d on bank 4*(5d mod 16) -> all
// distinct within a b128 lane group (5 coprime to 16) -> conflict-free.
#define TR_PITCH 40

#define FA_D 128
#define LOG2E 1.4426950408889634f

// ---------------------------------------------------------------------------
// delta = rowsum(dO * O)
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
fa_bwd_delta_kernel(const short* __restrict__ dout,
                    const short* __restrict__ out, float* __restrict__ delta,
                    long rows, int Hq, int S,
                    long do_bs, long do_hs, long do_ss,
                    long o_bs, long o_hs, long o_ss) {
  // row enumerates (b, h, s).  16 lanes own one 128-elem row (8 bf16 per
  // lane via b128 loads, coalesced 256 B per row segment); 4 shfl_xor
  // steps fold the 16 partials — no LDS, no block barrier.  This is a
  // pure streaming op (read 2 x B*Hq*S*128 bf16) and runs at HBM rate;
  // the previous one-block-per-row form was ~14x off the read floor.
  const int sub = threadIdx.x & 15;
  const long row0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) >> 4;
  const long row_step = ((long)gridDim.x * blockDim.x) >> 4;
  for (long row = row0; row < rows; row += row_step) {
    const long b = row / ((long)Hq * S);
    const long h = (row / S) % Hq;
    const long sq = row % S;
    s8v dv = *(const s8v*)(dout + b * do_bs + h * do_hs + sq * do_ss
                           + sub * 8);
    s8v ov = *(const s8v*)(out + b * o_bs + h * o_hs + sq * o_ss + sub * 8);
    float s = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) s += bits2f(dv[j]) * bits2f(ov[j]);
#pragma unroll
    for (int off = 8; off >= 1; off >>= 1) s += __shfl_xor(s, off, 64);
    if (sub == 0) delta[row] = s;
  }
}

// ---------------------------------------------------------------------------
// shared helpers for the two main kernels
// ---------------------------------------------------------------------------

// stage a 32x128 bf16 tile row-major (XOR-swizzled) + transposed [128][32]
// copy, using all 256 threads (2x16B pieces each for row-major; scatter b16
// for the transpose).  row stride row-major: 256 B; transposed: 64 B.

struct StageRegs { uint4v v0, v1; };

__device__ __forceinline__ StageRegs load_tile32(const short* __restrict__ src,
                                                 long src_row0,
                                                 long src_stride,
                                                 int rows_valid) {
  int rp = (threadIdx.x & 255) >> 4;
  int c16 = threadIdx.x & 15;
  int r0 = 2 * rp, r1 = 2 * rp + 1;
  int rr0 = r0 < rows_valid ? r0 : (rows_valid > 0 ? rows_valid - 1 : 0);
  int rr1 = r1 < rows_valid ? r1 : (rows_valid > 0 ? rows_valid - 1 : 0);
  StageRegs r;
  r.v0 = *(const uint4v*)(src + (src_row0 + rr0) * src_stride + c16 * 8);
  r.v1 = *(const uint4v*)(src + (src_row0 + rr1) * src_stride + c16 * 8);
  return r;
}

__device__ __forceinline__ void write_tile32(StageRegs r, char* lds_rm,
                                             char* lds_tr) {
  if (blockDim.x > 256 && threadIdx.x >= 256) return;
  int rp = threadIdx.x >> 4;
  int c16 = threadIdx.x & 15;
  int r0 = 2 * rp, r1 = 2 * rp + 1;
  *(uint4v*)(lds_rm + r0 * (FA_D * 2) + swz16(r0, c16 * 16)) = r.v0;
  *(uint4v*)(lds_rm + r1 * (FA_D * 2) + swz16(r1, c16 * 16)) = r.v1;
  union { uint4v u; short s[8]; } a, b;
  a.u = r.v0; b.u = r.v1;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int d = c16 * 8 + j;
    uint pair = ((uint)(unsigned short)a.s[j]) |
                (((uint)(unsigned short)b.s[j]) << 16);
    *(uint*)(lds_tr + d * (TR_PITCH * 2) + r0 * 2) = pair;
  }
}

__device__ __forceinline__ void write_tile32_rm(StageRegs r, char* lds_rm) {
  if (blockDim.x > 256 && threadIdx.x >= 256) return;
  int rp = threadIdx.x >> 4;
  int c16 = threadIdx.x & 15;
  int r0 = 2 * rp, r1 = 2 * rp + 1;
  *(uint4v*)(lds_rm + r0 * (FA_D * 2) + swz16(r0, c16 * 16)) = r.v0;
  *(uint4v*)(lds_rm + r1 * (FA_D * 2) + swz16(r1, c16 * 16)) = r.v1;
}

__device__ __forceinline__ void stage_tile32(const short* __restrict__ src,
                                             long src_row0, long src_stride,
                                             int rows_valid, char* lds_rm,
                                             char* lds_tr) {
  if (blockDim.x > 256 && threadIdx.x >= 256) return;
  // thread t owns rows {2rp, 2rp+1} at 16B slot c16 (rp = t>>4, c16 = t&15):
  // row-major: 2 x b128 writes; transposed: 8 x b32 (two k cols per write).
  // NOTE: swz32 flips byte bits 4-5; a b32 write at (k=2rp)*2 has the pair
  // within one 4-byte word only when the two k's share the swizzled word,
  // which holds because swz32 only permutes 16-byte groups.
  int rp = threadIdx.x >> 4;
  int c16 = threadIdx.x & 15;
  int r0 = 2 * rp, r1 = 2 * rp + 1;
  int rr0 = r0 < rows_valid ? r0 : (rows_valid > 0 ? rows_valid - 1 : 0);
  int rr1 = r1 < rows_valid ? r1 : (rows_valid > 0 ? rows_valid - 1 : 0);
  uint4v v0 = *(const uint4v*)(src + (src_row0 + rr0) * src_stride + c16 * 8);
  uint4v v1 = *(const uint4v*)(src + (src_row0 + rr1) * src_stride + c16 * 8);
  *(uint4v*)(lds_rm + r0 * (FA_D * 2) + swz16(r0, c16 * 16)) = v0;
  *(uint4v*)(lds_rm + r1 * (FA_D * 2) + swz16(r1, c16 * 16)) = v1;
  union { uint4v u; short s[8]; } a, b;
  a.u = v0; b.u = v1;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int d = c16 * 8 + j;
    uint pair = ((uint)(unsigned short)a.s[j]) |
                (((uint)(unsigned short)b.s[j]) << 16);
    *(uint*)(lds_tr + d * (TR_PITCH * 2) + r0 * 2) = pair;
  }
}

// ---------------------------------------------------------------------------
// dK/dV kernel (8 waves over a 128-row kv block; see kernel comment)
// LDS: Q rm 32KB + Qt/dOt transposed 36.9KB + lse/delta 512B — the P/dS
// tiles live entirely in registers (pack + permlane32_swap repack).
// ---------------------------------------------------------------------------
#define TR_TILE_B (FA_D * TR_PITCH * 2)  // 10240 B

// ---- 64-row tile staging (512 threads, 2-row pairs) ----------------------
#define TR64_PITCH 72   // 144-byte rows: 16B-aligned, banks 4*(9d mod 16)
#define TR64_TILE_B (FA_D * TR64_PITCH * 2)   // 18432 B
#define BW64_LDS_Q 0
#define BW64_LDS_DO (64 * FA_D * 2)
#define BW64_LDS_QT (2 * 64 * FA_D * 2)
#define BW64_LDS_DOT (2 * 64 * FA_D * 2 + TR64_TILE_B)
// 128 floats: lse*log2(e) for the 64 q rows at [0..64), delta at [64..128)
#define BW64_LDS_LD (2 * 64 * FA_D * 2 + 2 * TR64_TILE_B)
// cross-wave A-fragment exchange: 8 slots (kvg x q-half) x 4 frags x
// 64 lanes x 16 B = 32 KB.  The kernel is VGPR-bound at 1 WG/CU (254
// VGPRs -> 2 waves/SIMD), so this LDS is free occupancy-wise.
#define BW64_LDS_X (BW64_LDS_LD + 128 * 4)
#define BW64_X_SLOT(kvg, half, f) \
  ((((kvg) * 2 + (half)) * 4 + (f)) * 64 * 16)

struct Stage64Regs { uint4v v0, v1; };

__device__ __forceinline__ Stage64Regs load_tile64(
    const short* __restrict__ src, long src_row0, long src_stride,
    int rows_valid) {
  int rp = threadIdx.x >> 4;        // 0..31 -> rows 2rp, 2rp+1 (0..63)
  int c16 = threadIdx.x & 15;
  int r0 = 2 * rp, r1 = 2 * rp + 1;
  int rr0 = r0 < rows_valid ? r0 : (rows_valid > 0 ? rows_valid - 1 : 0);
  int rr1 = r1 < rows_valid ? r1 : (rows_valid > 0 ? rows_valid - 1 : 0);
  Stage64Regs r;
  r.v0 = *(const uint4v*)(src + (src_row0 + rr0) * src_stride + c16 * 8);
  r.v1 = *(const uint4v*)(src + (src_row0 + rr1) * src_stride + c16 * 8);
  return r;
}

__device__ __forceinline__ void write_tile64(Stage64Regs r, char* lds_rm,
                                             char* lds_tr) {
  int rp = threadIdx.x >> 4;
  int c16 = threadIdx.x & 15;
  int r0 = 2 * rp, r1 = 2 * rp + 1;
  *(uint4v*)(lds_rm + r0 * (FA_D * 2) + swz16(r0, c16 * 16)) = r.v0;
  *(uint4v*)(lds_rm + r1 * (FA_D * 2) + swz16(r1, c16 * 16)) = r.v1;
  union { uint4v u; short s[8]; } a, b;
  a.u = r.v0; b.u = r.v1;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int d = c16 * 8 + j;
    uint pair = ((uint)(unsigned short)a.s[j]) |
                (((uint)(unsigned short)b.s[j]) << 16);
    *(uint*)(lds_tr + d * (TR64_PITCH * 2) + r0 * 2) = pair;
  }
}

__device__ __forceinline__ void stage_tile64(const short* __restrict__ src,
                                             long src_row0, long src_stride,
                                             int rows_valid, char* lds_rm,
                                             char* lds_tr) {
  write_tile64(load_tile64(src, src_row0, src_stride, rows_valid), lds_rm,
               lds_tr);
}

extern "C" __global__ void __launch_bounds__(512, 2)
fa_bwd_dkdv_kernel(const short* __restrict__ qp, const short* __restrict__ kp,
                   const short* __restrict__ vp,
                   const short* __restrict__ dop,
                   const float* __restrict__ lsep,
                   const float* __restrict__ deltap,
                   short* __restrict__ dkp, short* __restrict__ dvp,
                   int B, int Hq, int Hkv, int S, float scale, int causal,
                   int window,
                   long q_bs, long q_hs, long q_ss,
                   long k_bs, long k_hs, long k_ss,
                   long v_bs, long v_hs, long v_ss,
                   long do_bs, long do_hs, long do_ss,
                   long dk_bs, long dk_hs, long dk_ss,
                   long dv_bs, long dv_hs, long dv_ss) {
  // 8 waves / 128-row kv block: wave w -> kv rows (w>>1)*32, d-half w&1.
  // Q-TILE = 64 rows per iteration, processed as two 32-q halves that
  // REUSE the st/dpt accumulators: one barrier pair and one staging round
  // per 64 q rows.
  //
  // The score/dS tiles never touch LDS: S^T/dP are computed with Q on the
  // A operand (q in accumulator REGISTER rows, kv in lanes), so the
  // pack_bf16x2 + permlane32_swap repack from the forward kernel turns
  // them directly into the A fragments of the dV/dK mfmas (k-dim = q,
  // lane = kv).  The softmax's per-q lse/delta become per-register-row
  // values read from a 128-float LDS tile staged once per q-tile.
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int lane = threadIdx.x & 63;
  // readfirstlane: provably wave-uniform -> scalar branches for the
  // per-wave activity guards instead of exec-mask divergence (T20)
  const int wid = __builtin_amdgcn_readfirstlane(threadIdx.x >> 6);
  const int col = lane & 31;
  const int hi = lane >> 5;
  const int kvg = wid >> 1;        // kv row group 0..3
  const int dhalf = wid & 1;       // d half 0..1

  // heads on x, kv-blocks on y: causal work DEcreases with kv-block index,
  // so ascending y already dispatches biggest-first (greedy LPT; see the
  // fwd kernel's dispatch note)
  const int kvblk = blockIdx.y;
  const int h = blockIdx.x;
  const int b = blockIdx.z;
  const int hkv = h / (Hq / Hkv);
  const long q_base = (long)b * q_bs + (long)h * q_hs;
  const long do_base = (long)b * do_bs + (long)h * do_hs;
  const long k_base = (long)b * k_bs + (long)hkv * k_hs;
  const long v_base = (long)b * v_bs + (long)hkv * v_hs;
  const long lse_base = (long)(b * Hq + h) * S;

  const int kv0 = kvblk * 128 + kvg * 32;   // this wave's kv rows
  const int my_k = kv0 + col;

  frag_u kf[8], vf[8];
  {
    int row = my_k < S ? my_k : S - 1;
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      kf[c].u4 = *(const uint4v*)(kp + k_base + (long)row * k_ss + c * 16 + hi * 8);
      vf[c].u4 = *(const uint4v*)(vp + v_base + (long)row * v_ss + c * 16 + hi * 8);
    }
  }

  f32x16 dv_acc[2] = {};
  f32x16 dk_acc[2] = {};

  float* ld_sm = (float*)(smem + BW64_LDS_LD);
  // threads 0..63 fetch lse (pre-scaled by log2 e), 64..127 delta
  auto load_ld = [&](int qt0) -> float {
    int qg = qt0 + (threadIdx.x & 63);
    int qc = qg < S ? qg : S - 1;
    if (threadIdx.x < 64) return lsep[lse_base + qc] * LOG2E;
    if (threadIdx.x < 128) return deltap[lse_base + qc];
    return 0.f;
  };
  auto write_ld = [&](float v) {
    if (threadIdx.x < 128) ld_sm[threadIdx.x] = v;
  };

  const float s2 = scale * LOG2E;
  int q_start = causal ? kvblk * 128 : 0;
  // sliding window: q attends k iff k <= q < k + window, so this kv
  // block's active q rows end at kv_max + window
  int q_end = S;
  if (window > 0) {
    int qe = kvblk * 128 + 127 + window;
    q_end = qe < S ? qe : S;
  }

  stage_tile64(qp + q_base, q_start, q_ss, S - q_start, smem + BW64_LDS_Q,
               smem + BW64_LDS_QT);
  stage_tile64(dop + do_base, q_start, do_ss, S - q_start, smem + BW64_LDS_DO,
               smem + BW64_LDS_DOT);
  write_ld(load_ld(q_start));

  Stage64Regs nq, ndo;
  float nld = 0.f;
  for (int q0 = q_start; q0 < q_end; q0 += 64) {
    __syncthreads();
    if (q0 + 64 < q_end) {
      nq = load_tile64(qp + q_base, q0 + 64, q_ss, S - q0 - 64);
      ndo = load_tile64(dop + do_base, q0 + 64, do_ss, S - q0 - 64);
      nld = load_ld(q0 + 64);
    }
    // ---- PRODUCE: this wave computes S^T/dP^T for q-half = dhalf ONLY.
    // Its dhalf-partner wave (same kvg) computes the other half; the
    // packed A fragments are exchanged through LDS, halving the score
    // mfma work per wave (the old form had both waves of a kvg pair
    // redo the identical full-d S^T/dP^T for both halves: 48 mfmas per
    // wave-iteration where 32 are algorithmically needed — measured as
    // dkdv running at 363 TF vs the forward's 533).
    {
      const int half = dhalf;
      const int qh0 = q0 + half * 32;
      const bool produce =
          (!causal || (qh0 + 31 >= kv0)) && qh0 < S &&
          (window <= 0 || qh0 < kv0 + 31 + window);
      if (produce) {
        const int rm_row = half * 32 + col;  // row inside the 64-row tile
        const int my_kv = kv0 + col;         // this lane's kv column

        // S^T2[q][kv]: A = Q rows (B-layout frag doubles as A: lane=q),
        // B = K rows (lane=kv) -> q in accumulator rows, kv in lanes
        f32x16 st = {};
        f32x16 dpt = {};
#pragma unroll
        for (int c = 0; c < 8; ++c) {
          frag_u qfr, dofr;
          qfr.u4 = *(const uint4v*)(smem + BW64_LDS_Q + rm_row * (FA_D * 2)
                                    + swz16(rm_row, (c * 16 + hi * 8) * 2));
          st = mfma_bf16(qfr.bf, kf[c].bf, st);
          dofr.u4 = *(const uint4v*)(smem + BW64_LDS_DO + rm_row * (FA_D * 2)
                                     + swz16(rm_row, (c * 16 + hi * 8) * 2));
          dpt = mfma_bf16(dofr.bf, vf[c].bf, dpt);
        }

        // p / dS per element; lse & delta indexed by the REGISTER q row.
        // acc rows r = 4g..4g+3 are CONSECUTIVE q rows 8g+4*hi.. so the
        // 32 per-row values batch into 8 b128 broadcast reads (one wait)
        // instead of 32 serialized b32 reads.
        float lse_v[16], dlt_v[16];
#pragma unroll
        for (int g = 0; g < 4; ++g) {
          *(float4*)(lse_v + 4 * g) =
              *(const float4*)(ld_sm + half * 32 + 8 * g + 4 * hi);
          *(float4*)(dlt_v + 4 * g) =
              *(const float4*)(ld_sm + 64 + half * 32 + 8 * g + 4 * hi);
        }
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int ql = half * 32 + acc_row(r, hi);
          int qg = q0 + ql;
          bool masked = (causal && my_kv > qg) || my_kv >= S || qg >= S ||
              (window > 0 && qg >= my_kv + window);
          float p =
              masked ? 0.f : __builtin_amdgcn_exp2f(st[r] * s2 - lse_v[r]);
          st[r] = p;
          dpt[r] = p * (dpt[r] - dlt_v[r]);
        }

        // repack accumulator rows (q) into A-fragment k-dim in REGISTERS
        // (same pack pairs + permlane32_swap as the forward kernel's P
        // repack; lane dim = kv is already in place)
        frag_u pA[2], dA[2];
#pragma unroll
        for (int cc = 0; cc < 2; ++cc) {
          uint b0 = pack_bf16x2(st[8 * cc + 0], st[8 * cc + 1]);
          uint b1 = pack_bf16x2(st[8 * cc + 2], st[8 * cc + 3]);
          uint b2 = pack_bf16x2(st[8 * cc + 4], st[8 * cc + 5]);
          uint b3 = pack_bf16x2(st[8 * cc + 6], st[8 * cc + 7]);
          {
            auto r01 = __builtin_amdgcn_permlane32_swap(b0, b2, false, false);
            b0 = r01[0]; b2 = r01[1];
          }
          {
            auto r23 = __builtin_amdgcn_permlane32_swap(b1, b3, false, false);
            b1 = r23[0]; b3 = r23[1];
          }
          pA[cc].u[0] = b0; pA[cc].u[1] = b1;
          pA[cc].u[2] = b2; pA[cc].u[3] = b3;
          uint c0 = pack_bf16x2(dpt[8 * cc + 0], dpt[8 * cc + 1]);
          uint c1 = pack_bf16x2(dpt[8 * cc + 2], dpt[8 * cc + 3]);
          uint c2 = pack_bf16x2(dpt[8 * cc + 4], dpt[8 * cc + 5]);
          uint c3 = pack_bf16x2(dpt[8 * cc + 6], dpt[8 * cc + 7]);
          {
            auto r01 = __builtin_amdgcn_permlane32_swap(c0, c2, false, false);
            c0 = r01[0]; c2 = r01[1];
          }
          {
            auto r23 = __builtin_amdgcn_permlane32_swap(c1, c3, false, false);
            c1 = r23[0]; c3 = r23[1];
          }
          dA[cc].u[0] = c0; dA[cc].u[1] = c1;
          dA[cc].u[2] = c2; dA[cc].u[3] = c3;
        }

        // publish this half's A fragments for both d-half waves of the
        // kvg pair (frag-major layout: consecutive lanes x 16 B ->
        // conflict-free ds_write_b128)
        char* xb = smem + BW64_LDS_X;
        *(uint4v*)(xb + BW64_X_SLOT(kvg, half, 0) + lane * 16) = pA[0].u4;
        *(uint4v*)(xb + BW64_X_SLOT(kvg, half, 1) + lane * 16) = pA[1].u4;
        *(uint4v*)(xb + BW64_X_SLOT(kvg, half, 2) + lane * 16) = dA[0].u4;
        *(uint4v*)(xb + BW64_X_SLOT(kvg, half, 3) + lane * 16) = dA[1].u4;
      }
    }
    __syncthreads();  // exchange visible

    // ---- CONSUME: both q halves on this wave's 64-column d-half -------
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      const int qh0 = q0 + half * 32;
      if ((causal && qh0 + 31 < kv0) || qh0 >= S ||
          (window > 0 && qh0 >= kv0 + 31 + window)) continue;

      // prefetch the dV/dK B fragments first — independent of the slot
      // reads, so all 12 b128 LDS loads pipeline into one counted wait
      frag_u dofr[2][2], qfr2[2][2];
#pragma unroll
      for (int cq = 0; cq < 2; ++cq)
#pragma unroll
        for (int nb = 0; nb < 2; ++nb) {
          int d = dhalf * 64 + nb * 32 + col;
          dofr[cq][nb].u4 = *(const uint4v*)(smem + BW64_LDS_DOT
                                             + d * (TR64_PITCH * 2)
                                             + (half * 32 + cq * 16 + hi * 8) * 2);
          qfr2[cq][nb].u4 = *(const uint4v*)(smem + BW64_LDS_QT
                                             + d * (TR64_PITCH * 2)
                                             + (half * 32 + cq * 16 + hi * 8) * 2);
        }
      frag_u pA[2], dA[2];
      const char* xb = smem + BW64_LDS_X;
      pA[0].u4 = *(const uint4v*)(xb + BW64_X_SLOT(kvg, half, 0) + lane * 16);
      pA[1].u4 = *(const uint4v*)(xb + BW64_X_SLOT(kvg, half, 1) + lane * 16);
      dA[0].u4 = *(const uint4v*)(xb + BW64_X_SLOT(kvg, half, 2) + lane * 16);
      dA[1].u4 = *(const uint4v*)(xb + BW64_X_SLOT(kvg, half, 3) + lane * 16);

      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int cq = 0; cq < 2; ++cq)
#pragma unroll
        for (int nb = 0; nb < 2; ++nb) {
          dv_acc[nb] = mfma_bf16(pA[cq].bf, dofr[cq][nb].bf, dv_acc[nb]);
          dk_acc[nb] = mfma_bf16(dA[cq].bf, qfr2[cq][nb].bf, dk_acc[nb]);
        }
      __builtin_amdgcn_s_setprio(0);
    }

    __syncthreads();
    if (q0 + 64 < q_end) {
      write_tile64(nq, smem + BW64_LDS_Q, smem + BW64_LDS_QT);
      write_tile64(ndo, smem + BW64_LDS_DO, smem + BW64_LDS_DOT);
      write_ld(nld);
    }
  }

  // ---- epilogue: per-wave 32x64 halves via LDS transpose ---------------
  __syncthreads();
  char* otile = smem + wid * (32 * 64 * 2);  // 4 KB per wave
#pragma unroll
  for (int nb = 0; nb < 2; ++nb)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int k = acc_row(r, hi);
      int dd = nb * 32 + col;
      *(short*)(otile + (k * 64 + dd) * 2) = f2bits(dv_acc[nb][r]);
    }
  __builtin_amdgcn_s_waitcnt(0);
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int p = lane + i * 64;
    int row = p >> 3, c16 = p & 7;
    int kg = kv0 + row;
    if (kg < S) {
      uint4v vv = *(const uint4v*)(otile + (row * 64 + c16 * 8) * 2);
      *(uint4v*)(dvp + (long)b * dv_bs + (long)h * dv_hs
                 + (long)kg * dv_ss + dhalf * 64
                 + c16 * 8) = vv;
    }
  }
  __builtin_amdgcn_s_waitcnt(0);
#pragma unroll
  for (int nb = 0; nb < 2; ++nb)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int k = acc_row(r, hi);
      int dd = nb * 32 + col;
      *(short*)(otile + (k * 64 + dd) * 2) = f2bits(dk_acc[nb][r] * scale);
    }
  __builtin_amdgcn_s_waitcnt(0);
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int p = lane + i * 64;
    int row = p >> 3, c16 = p & 7;
    int kg = kv0 + row;
    if (kg < S) {
      uint4v vv = *(const uint4v*)(otile + (row * 64 + c16 * 8) * 2);
      *(uint4v*)(dkp + (long)b * dk_bs + (long)h * dk_hs
                 + (long)kg * dk_ss + dhalf * 64
                 + c16 * 8) = vv;
    }
  }
}

// ---------------------------------------------------------------------------
// dQ kernel: 4 waves, wave owns q rows [q0 + 32*wid, +32)
// LDS: K rm 8KB + Kt 8KB + V rm 8KB + per-wave dS tiles 4*2.5KB = 34KB
// ---------------------------------------------------------------------------
#define DQ_LDS_K 0
#define DQ_LDS_KT (32 * FA_D * 2)
#define DQ_LDS_V (32 * FA_D * 2 + TR_TILE_B)

extern "C" __global__ void __launch_bounds__(256, 2)
fa_bwd_dq_kernel(const short* __restrict__ qp, const short* __restrict__ kp,
                 const short* __restrict__ vp, const short* __restrict__ dop,
                 const float* __restrict__ lsep,
                 const float* __restrict__ deltap, short* __restrict__ dqp,
                 int B, int Hq, int Hkv, int S, float scale, int causal,
                 int window,
                 long q_bs, long q_hs, long q_ss,
                 long k_bs, long k_hs, long k_ss,
                 long v_bs, long v_hs, long v_ss,
                 long do_bs, long do_hs, long do_ss,
                 long dq_bs, long dq_hs, long dq_ss) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int lane = threadIdx.x & 63;
  // readfirstlane: provably wave-uniform -> scalar branches for the
  // per-wave activity guards instead of exec-mask divergence (T20)
  const int wid = __builtin_amdgcn_readfirstlane(threadIdx.x >> 6);
  const int col = lane & 31;
  const int hi = lane >> 5;

  // heads on x, q-blocks on y reversed: biggest-causal-work first (LPT,
  // see the fwd kernel's dispatch note)
  const int qblk = (int)(gridDim.y - 1 - blockIdx.y);
  const int h = blockIdx.x;
  const int b = blockIdx.z;
  const int hkv = h / (Hq / Hkv);
  const long q_base = (long)b * q_bs + (long)h * q_hs;
  const long do_base = (long)b * do_bs + (long)h * do_hs;
  const long dq_base = (long)b * dq_bs + (long)h * dq_hs;
  const long k_base = (long)b * k_bs + (long)hkv * k_hs;
  const long v_base = (long)b * v_bs + (long)hkv * v_hs;
  const long lse_base = (long)(b * Hq + h) * S;

  const int q0 = qblk * 128 + wid * 32;
  const int my_q = q0 + col;
  const int q_ld = my_q < S ? my_q : S - 1;

  // Q, dO fragments in registers (lane: row q=col within wave tile)
  frag_u qf[8], dof[8];
#pragma unroll
  for (int c = 0; c < 8; ++c) {
    qf[c].u4 = *(const uint4v*)(qp + q_base + (long)q_ld * q_ss + c * 16 + hi * 8);
    dof[c].u4 = *(const uint4v*)(dop + do_base + (long)q_ld * do_ss + c * 16 + hi * 8);
  }
  const float lse2 = lsep[lse_base + q_ld] * LOG2E;
  const float dlt = deltap[lse_base + q_ld];

  f32x16 dq_acc[4] = {};

  // WG-uniform loop bound (all waves share barriers); per-wave causal
  // skipping happens via wave_active below
  const int kv_end = causal ? min(S, qblk * 128 + 128) : S;
  // sliding window: this q block's earliest active kv row (tile-aligned)
  const int kv_begin = window > 0
      ? ((qblk * 128 - window + 1 > 0 ? qblk * 128 - window + 1 : 0) & ~31)
      : 0;
  const float s2 = scale * LOG2E;

  stage_tile32(kp + k_base, kv_begin, k_ss, S - kv_begin, smem + DQ_LDS_K,
               smem + DQ_LDS_KT);
  write_tile32_rm(load_tile32(vp + v_base, kv_begin, v_ss, S - kv_begin),
                  smem + DQ_LDS_V);

  StageRegs nk, nv;
  for (int kv0 = kv_begin; kv0 < kv_end; kv0 += 32) {
    __syncthreads();
    if (kv0 + 32 < kv_end) {
      nk = load_tile32(kp + k_base, kv0 + 32, k_ss, S - kv0 - 32);
      nv = load_tile32(vp + v_base, kv0 + 32, v_ss, S - kv0 - 32);
    }
    const bool wave_active = (!causal || (kv0 <= q0 + 31)) &&
        (window <= 0 || kv0 + 31 + window > q0);

    if (wave_active) {
      // S^T[k][q]: A = K rows (LDS rm, row k=col), B = Q^T (regs)
      f32x16 st = {};
      f32x16 dpt = {};
#pragma unroll
      for (int c = 0; c < 8; ++c) {
        frag_u kfr;
        kfr.u4 = *(const uint4v*)(smem + DQ_LDS_K + col * (FA_D * 2)
                                  + swz16(col, (c * 16 + hi * 8) * 2));
        st = mfma_bf16(kfr.bf, qf[c].bf, st);
        frag_u vfr;
        vfr.u4 = *(const uint4v*)(smem + DQ_LDS_V + col * (FA_D * 2)
                                  + swz16(col, (c * 16 + hi * 8) * 2));
        dpt = mfma_bf16(vfr.bf, dof[c].bf, dpt);
      }

      f32x16 dst;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int kg = kv0 + acc_row(r, hi);
        bool masked = (causal && kg > my_q) || kg >= S || my_q >= S ||
            (window > 0 && kg <= my_q - window);
        float p = masked ? 0.f : __builtin_amdgcn_exp2f(st[r] * s2 - lse2);
        dst[r] = p * (dpt[r] - dlt);
      }

      // prefetch the 8 Kt B-fragments (independent of the pack below —
      // the b128 LDS reads land under the pack VALU)
      frag_u kfr2[2][4];
#pragma unroll
      for (int ck = 0; ck < 2; ++ck)
#pragma unroll
        for (int nb = 0; nb < 4; ++nb) {
          int d = nb * 32 + col;
          kfr2[ck][nb].u4 = *(const uint4v*)(smem + DQ_LDS_KT
                                             + d * (TR_PITCH * 2)
                                             + (ck * 16 + hi * 8) * 2);
        }

      // repack dS accumulator rows (kv) into A-fragment k-dim in
      // REGISTERS (lane dim q already in place) — same pack pairs +
      // permlane32_swap as the forward P repack; no pw LDS round-trip
      frag_u dA[2];
#pragma unroll
      for (int cc = 0; cc < 2; ++cc) {
        uint b0 = pack_bf16x2(dst[8 * cc + 0], dst[8 * cc + 1]);
        uint b1 = pack_bf16x2(dst[8 * cc + 2], dst[8 * cc + 3]);
        uint b2 = pack_bf16x2(dst[8 * cc + 4], dst[8 * cc + 5]);
        uint b3 = pack_bf16x2(dst[8 * cc + 6], dst[8 * cc + 7]);
        {
          auto r01 = __builtin_amdgcn_permlane32_swap(b0, b2, false, false);
          b0 = r01[0]; b2 = r01[1];
        }
        {
          auto r23 = __builtin_amdgcn_permlane32_swap(b1, b3, false, false);
          b1 = r23[0]; b3 = r23[1];
        }
        dA[cc].u[0] = b0; dA[cc].u[1] = b1;
        dA[cc].u[2] = b2; dA[cc].u[3] = b3;
      }

      // dQ[q][d] += sum_k dS[q][k] K[k][d]
      //   A = dS (registers, lane=q), B = K[k][d] via Kt rows d
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ck = 0; ck < 2; ++ck) {
#pragma unroll
        for (int nb = 0; nb < 4; ++nb) {
          dq_acc[nb] = mfma_bf16(dA[ck].bf, kfr2[ck][nb].bf, dq_acc[nb]);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }

    __syncthreads();
    if (kv0 + 32 < kv_end) {
      write_tile32(nk, smem + DQ_LDS_K, smem + DQ_LDS_KT);
      write_tile32_rm(nv, smem + DQ_LDS_V);
    }
  }

  // epilogue: dQ (col = d, rows q) -> LDS transpose -> coalesced store
  __syncthreads();
  char* otile = smem + wid * (32 * FA_D * 2);
#pragma unroll
  for (int nb = 0; nb < 4; ++nb)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int qr = acc_row(r, hi);
      int d = nb * 32 + col;
      *(short*)(otile + (qr * FA_D + d) * 2) = f2bits(dq_acc[nb][r] * scale);
    }
  __builtin_amdgcn_s_waitcnt(0);
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    int p = lane + i * 64;
    int row = p >> 4, c16 = p & 15;
    int qg = q0 + row;
    if (qg < S) {
      uint4v vv = *(const uint4v*)(otile + (row * FA_D + c16 * 8) * 2);
      *(uint4v*)(dqp + dq_base + (long)qg * dq_ss + c16 * 8) = vv;
    }
  }
}

// ---------------------------------------------------------------------------
extern "C" void flash_attn_bwd_strided(
    const void* q, const void* k, const void* v, const void* out,
    const void* dout, const void* lse, void* delta, void* dq, void* dk,
    void* dv, int B, int Hq, int Hkv, int S, float scale, int causal,
    int window, const long* st, hipStream_t stream) {
  // st = 24 longs: (bs, hs, ss) x (q, k, v, o, dout, dq, dk, dv)
  long rows = (long)B * Hq * S;
  // 16 lanes per row -> 16 rows per 256-thread block; cap well above the
  // 256-CU fill point and grid-stride the rest
  long blocks = (rows + 15) / 16;
  int nb = blocks < 8192 ? (int)blocks : 8192;
  fa_bwd_delta_kernel<<<nb, 256, 0, stream>>>(
      (const short*)dout, (const short*)out, (float*)delta, rows, Hq, S,
      st[12], st[13], st[14], st[9], st[10], st[11]);
  dim3 gkv(Hq, (S + 127) / 128, B);
  // staging (70,144 B) + A-fragment exchange (32,768 B) = 102,912 B; the
  // kernel is VGPR-bound at 1 WG/CU so the extra LDS costs no occupancy
  size_t lds1 = BW64_LDS_X + 8 * 4 * 64 * 16;
  fa_bwd_dkdv_kernel<<<gkv, 512, lds1, stream>>>(
      (const short*)q, (const short*)k, (const short*)v, (const short*)dout,
      (const float*)lse, (const float*)delta, (short*)dk, (short*)dv, B, Hq,
      Hkv, S, scale, causal, window, st[0], st[1], st[2], st[3], st[4],
      st[5], st[6], st[7], st[8], st[12], st[13], st[14], st[18], st[19],
      st[20], st[21], st[22], st[23]);
  dim3 gq(Hq, (S + 127) / 128, B);
  // staging layout needs 26.6 KB; the epilogue reuses LDS as 4 per-wave
  // 32x128 transpose tiles = 32 KB, which dominates
  size_t lds2 = 4 * 32 * FA_D * 2;
  fa_bwd_dq_kernel<<<gq, 256, lds2, stream>>>(
      (const short*)q, (const short*)k, (const short*)v, (const short*)dout,
      (const float*)lse, (const float*)delta, (short*)dq, B, Hq, Hkv, S,
      scale, causal, window, st[0], st[1], st[2], st[3], st[4], st[5],
      st[6], st[7], st[8], st[12], st[13], st[14], st[15], st[16],
      st[17]);
}

extern "C" void flash_attn_bwd(const void* q, const void* k, const void* v,
                               const void* out, const void* dout,
                               const void* lse, void* delta, void* dq,
                               void* dk, void* dv, int B, int Hq, int Hkv,
                               int S, float scale, int causal,
                               hipStream_t stream) {
  const long qd[3] = {(long)Hq * S * FA_D, (long)S * FA_D, FA_D};
  const long kd[3] = {(long)Hkv * S * FA_D, (long)S * FA_D, FA_D};
  long st[24];
  for (int i = 0; i < 3; ++i) {
    st[0 + i] = qd[i];            // q
    st[3 + i] = kd[i];            // k
    st[6 + i] = kd[i];            // v
    st[9 + i] = qd[i];            // out
    st[12 + i] = qd[i];           // dout
    st[15 + i] = qd[i];           // dq
    st[18 + i] = qd[i];           // dk (per-Q-head buffers)
    st[21 + i] = qd[i];           // dv
  }
  flash_attn_bwd_strided(q, k, v, out, dout, lse, delta, dq, dk, dv, B, Hq,
                         Hkv, S, scale, causal, 0, st, stream);
}
