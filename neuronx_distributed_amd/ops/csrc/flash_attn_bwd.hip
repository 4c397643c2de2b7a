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

// swizzle for 64-byte-row tiles (transposed [128][32] bf16): only bits 4-5
__device__ __forceinline__ int swz32(int row, int byte_in_row) {
  return byte_in_row ^ ((row & 3) << 4);
}

#define FA_D 128
#define LOG2E 1.4426950408889634f

// ---------------------------------------------------------------------------
// delta = rowsum(dO * O)
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
fa_bwd_delta_kernel(const short* __restrict__ dout,
                    const short* __restrict__ out, float* __restrict__ delta,
                    long rows) {
  __shared__ float scratch[16];
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const short* d = dout + row * FA_D;
    const short* o = out + row * FA_D;
    float s = 0.f;
    for (int i = threadIdx.x; i < FA_D / 8; i += blockDim.x) {
      s8v dv = *(const s8v*)(d + i * 8);
      s8v ov = *(const s8v*)(o + i * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) s += bits2f(dv[j]) * bits2f(ov[j]);
    }
    s = block_reduce_sum(s, scratch);
    if (threadIdx.x == 0) delta[row] = s;
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// shared helpers for the two main kernels
// ---------------------------------------------------------------------------

// stage a 32x128 bf16 tile row-major (XOR-swizzled) + transposed [128][32]
// copy, using all 256 threads (2x16B pieces each for row-major; scatter b16
// for the transpose).  row stride row-major: 256 B; transposed: 64 B.
__device__ __forceinline__ void stage_tile32(const short* __restrict__ src,
                                             long src_row0, long src_stride,
                                             int rows_valid, char* lds_rm,
                                             char* lds_tr) {
  int tid = threadIdx.x;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    int p = tid + i * 256;          // 512 pieces: row p>>4, slot p&15
    int row = p >> 4;
    int c16 = p & 15;
    int rr = row < rows_valid ? row : (rows_valid > 0 ? rows_valid - 1 : 0);
    uint4v vv = *(const uint4v*)(src + (src_row0 + rr) * src_stride + c16 * 8);
    *(uint4v*)(lds_rm + row * (FA_D * 2) + swz(row, c16 * 16)) = vv;
    union { uint4v u; short s[8]; } u;
    u.u = vv;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int d = c16 * 8 + j;
      *(short*)(lds_tr + d * 64 + swz32(d, row * 2)) = u.s[j];
    }
  }
}

// per-wave [32][40] bf16 tile (rows k or q, padded cols): write a 32x32
// accumulator-layout matrix (col = l&31, row = acc_row(r,hi)) as PAIRS
// (consecutive regs are consecutive rows) -> 2 cols packed... rows differ,
// so write scalar b16: 16 writes per lane.
#define PW_PITCH 40  // elements; 80 B rows -> bank stride 20 (conflict-lite)
__device__ __forceinline__ void write_acc_tile(char* tile, const f32x16& a,
                                               int lane) {
  int col = lane & 31, hi = lane >> 5;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int row = acc_row(r, hi);
    *(short*)(tile + (row * PW_PITCH + col) * 2) = f2bits(a[r]);
  }
}

// read an 8-element fragment (row fixed = l&31 style caller-supplied, cols
// contiguous) from a PW_PITCH tile
__device__ __forceinline__ bf16x8 read_pw_row(const char* tile, int row,
                                              int col0) {
  frag_u f;
#pragma unroll
  for (int j = 0; j < 8; ++j)
    f.bf[j] = *(const __bf16*)(tile + (row * PW_PITCH + col0 + j) * 2);
  return f.bf;
}

// ---------------------------------------------------------------------------
// dK/dV kernel: 4 waves, wave owns kv rows [kv0 + 32*wid, +32)
// ---------------------------------------------------------------------------
// LDS: Q rm 16KB?? 32x128x2 = 8KB; dO rm 8KB; Qt 8KB; dOt 8KB;
//      per-wave P^T + dS^T tiles 2*4*32*40*2 = 20KB  => ~52KB
#define BW_LDS_Q 0
#define BW_LDS_DO (32 * FA_D * 2)
#define BW_LDS_QT (2 * 32 * FA_D * 2)
#define BW_LDS_DOT (3 * 32 * FA_D * 2)
#define BW_LDS_PW (4 * 32 * FA_D * 2)
#define PW_BYTES (32 * PW_PITCH * 2)

extern "C" __global__ void __launch_bounds__(256, 1)
fa_bwd_dkdv_kernel(const short* __restrict__ qp, const short* __restrict__ kp,
                   const short* __restrict__ vp,
                   const short* __restrict__ dop,
                   const float* __restrict__ lsep,
                   const float* __restrict__ deltap,
                   short* __restrict__ dkp, short* __restrict__ dvp,
                   int B, int Hq, int Hkv, int S, float scale, int causal) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int col = lane & 31;
  const int hi = lane >> 5;

  const int kvblk = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = h / (Hq / Hkv);
  const long q_base = ((long)(b * Hq + h) * S) * FA_D;
  const long kv_base = ((long)(b * Hkv + hkv) * S) * FA_D;
  const long lse_base = (long)(b * Hq + h) * S;

  const int kv0 = kvblk * 128 + wid * 32;   // this wave's kv rows
  const int my_k = kv0 + col;               // lane's kv row (for masks only)

  // K,V fragments in registers: lane holds row (kv0 + col), d chunks
  frag_u kf[8], vf[8];
  {
    int row = my_k < S ? my_k : S - 1;
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      kf[c].u4 = *(const uint4v*)(kp + kv_base + (long)row * FA_D + c * 16 + hi * 8);
      vf[c].u4 = *(const uint4v*)(vp + kv_base + (long)row * FA_D + c * 16 + hi * 8);
    }
  }

  f32x16 dv_acc[4] = {};
  f32x16 dk_acc[4] = {};

  char* pw_p = smem + BW_LDS_PW + wid * 2 * PW_BYTES;        // P^T tile
  char* pw_ds = pw_p + PW_BYTES;                             // dS^T tile

  const float s2 = scale * LOG2E;
  int q_start = causal ? (kvblk * 128) / 32 * 32 : 0;
  // NOTE: q_start aligned to the WG's first kv row (not per-wave) so all
  // waves stay in the same barrier schedule.

  stage_tile32(qp + q_base, q_start, FA_D, S - q_start, smem + BW_LDS_Q,
               smem + BW_LDS_QT);
  stage_tile32(dop + q_base, q_start, FA_D, S - q_start, smem + BW_LDS_DO,
               smem + BW_LDS_DOT);

  for (int q0 = q_start; q0 < S; q0 += 32) {
    __syncthreads();
    const int my_q = q0 + col;             // lane's q (col axis)
    const bool wave_active = !causal || (q0 + 31 >= kv0);

    if (wave_active) {
      const float lse2 = lsep[lse_base + (my_q < S ? my_q : S - 1)] * LOG2E;
      const float dlt = deltap[lse_base + (my_q < S ? my_q : S - 1)];

      // S^T[k][q] = sum_d K[k][d] Q^T[d][q]
      f32x16 st = {};
#pragma unroll
      for (int c = 0; c < 8; ++c) {
        frag_u qfr;
        qfr.u4 = *(const uint4v*)(smem + BW_LDS_Q + col * (FA_D * 2)
                                  + swz(col, (c * 16 + hi * 8) * 2));
        st = mfma_bf16(kf[c].bf, qfr.bf, st);
      }
      // dP^T[k][q] = sum_d V[k][d] dO^T[d][q]
      f32x16 dpt = {};
#pragma unroll
      for (int c = 0; c < 8; ++c) {
        frag_u dofr;
        dofr.u4 = *(const uint4v*)(smem + BW_LDS_DO + col * (FA_D * 2)
                                   + swz(col, (c * 16 + hi * 8) * 2));
        dpt = mfma_bf16(vf[c].bf, dofr.bf, dpt);
      }

      // P^T = exp2(s*s2 - lse2); dS^T = P^T * (dP^T - delta)
      f32x16 pt, dst;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int kg = kv0 + acc_row(r, hi);
        bool masked = (causal && kg > my_q) || kg >= S || my_q >= S;
        float p = masked ? 0.f : __builtin_exp2f(st[r] * s2 - lse2);
        pt[r] = p;
        dst[r] = p * (dpt[r] - dlt);
      }
      write_acc_tile(pw_p, pt, lane);
      write_acc_tile(pw_ds, dst, lane);

      // dV[k][d] += sum_q P^T[k][q] dO[q][d]
      //   A = P^T rows k=col (b128 from pw_p), B = dO[q][d] via dOt rows d
#pragma unroll
      for (int cq = 0; cq < 2; ++cq) {
        frag_u pa;
        pa.bf = read_pw_row(pw_p, col, cq * 16 + hi * 8);
        frag_u da;
        da.bf = read_pw_row(pw_ds, col, cq * 16 + hi * 8);
#pragma unroll
        for (int nb = 0; nb < 4; ++nb) {
          int d = nb * 32 + col;
          frag_u dofr, qfr;
          dofr.u4 = *(const uint4v*)(smem + BW_LDS_DOT + d * 64
                                     + swz32(d, (cq * 16 + hi * 8) * 2));
          dv_acc[nb] = mfma_bf16(pa.bf, dofr.bf, dv_acc[nb]);
          qfr.u4 = *(const uint4v*)(smem + BW_LDS_QT + d * 64
                                    + swz32(d, (cq * 16 + hi * 8) * 2));
          dk_acc[nb] = mfma_bf16(da.bf, qfr.bf, dk_acc[nb]);
        }
      }
    }

    __syncthreads();
    if (q0 + 32 < S) {
      stage_tile32(qp + q_base, q0 + 32, FA_D, S - q0 - 32, smem + BW_LDS_Q,
                   smem + BW_LDS_QT);
      stage_tile32(dop + q_base, q0 + 32, FA_D, S - q0 - 32,
                   smem + BW_LDS_DO, smem + BW_LDS_DOT);
    }
  }

  // ---- epilogue: dK/dV out (B,Hq,S,D) bf16; transpose via LDS ----------
  // accumulators: col = d_local (n), rows k per reg.  Reuse smem tile:
  // write [k][d] rows then vector-store.
  __syncthreads();
  char* otile = smem + wid * (32 * FA_D * 2);  // per-wave 8KB scratch
#pragma unroll
  for (int nb = 0; nb < 4; ++nb)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int k = acc_row(r, hi);
      int d = nb * 32 + col;
      *(short*)(otile + (k * FA_D + d) * 2) = f2bits(dv_acc[nb][r] );
    }
  // wave-internal write->read; compiler orders via lgkmcnt on aliasing LDS
  __builtin_amdgcn_s_waitcnt(0);  // lgkmcnt(0) conservative
#pragma unroll
  for (int i = 0; i < 8; ++i) {   // 64 lanes * 8 = 512 pieces of 16B
    int p = lane + i * 64;
    int row = p >> 4, c16 = p & 15;
    int kg = kv0 + row;
    if (kg < S) {
      uint4v vv = *(const uint4v*)(otile + (row * FA_D + c16 * 8) * 2);
      *(uint4v*)(dvp + ((long)(b * Hq + h) * S + kg) * FA_D + c16 * 8) = vv;
    }
  }
  __builtin_amdgcn_s_waitcnt(0);
#pragma unroll
  for (int nb = 0; nb < 4; ++nb)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int k = acc_row(r, hi);
      int d = nb * 32 + col;
      *(short*)(otile + (k * FA_D + d) * 2) = f2bits(dk_acc[nb][r] * scale);
    }
  __builtin_amdgcn_s_waitcnt(0);
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    int p = lane + i * 64;
    int row = p >> 4, c16 = p & 15;
    int kg = kv0 + row;
    if (kg < S) {
      uint4v vv = *(const uint4v*)(otile + (row * FA_D + c16 * 8) * 2);
      *(uint4v*)(dkp + ((long)(b * Hq + h) * S + kg) * FA_D + c16 * 8) = vv;
    }
  }
}

// ---------------------------------------------------------------------------
// dQ kernel: 4 waves, wave owns q rows [q0 + 32*wid, +32)
// LDS: K rm 8KB + Kt 8KB + V rm 8KB + per-wave dS tiles 4*2.5KB = 34KB
// ---------------------------------------------------------------------------
#define DQ_LDS_K 0
#define DQ_LDS_KT (32 * FA_D * 2)
#define DQ_LDS_V (2 * 32 * FA_D * 2)
#define DQ_LDS_PW (3 * 32 * FA_D * 2)

extern "C" __global__ void __launch_bounds__(256, 2)
fa_bwd_dq_kernel(const short* __restrict__ qp, const short* __restrict__ kp,
                 const short* __restrict__ vp, const short* __restrict__ dop,
                 const float* __restrict__ lsep,
                 const float* __restrict__ deltap, short* __restrict__ dqp,
                 int B, int Hq, int Hkv, int S, float scale, int causal) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int col = lane & 31;
  const int hi = lane >> 5;

  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = h / (Hq / Hkv);
  const long q_base = ((long)(b * Hq + h) * S) * FA_D;
  const long kv_base = ((long)(b * Hkv + hkv) * S) * FA_D;
  const long lse_base = (long)(b * Hq + h) * S;

  const int q0 = blockIdx.x * 128 + wid * 32;
  const int my_q = q0 + col;
  const int q_ld = my_q < S ? my_q : S - 1;

  // Q, dO fragments in registers (lane: row q=col within wave tile)
  frag_u qf[8], dof[8];
#pragma unroll
  for (int c = 0; c < 8; ++c) {
    qf[c].u4 = *(const uint4v*)(qp + q_base + (long)q_ld * FA_D + c * 16 + hi * 8);
    dof[c].u4 = *(const uint4v*)(dop + q_base + (long)q_ld * FA_D + c * 16 + hi * 8);
  }
  const float lse2 = lsep[lse_base + q_ld] * LOG2E;
  const float dlt = deltap[lse_base + q_ld];

  f32x16 dq_acc[4] = {};
  char* pw_ds = smem + DQ_LDS_PW + wid * PW_BYTES;

  // WG-uniform loop bound (all waves share barriers); per-wave causal
  // skipping happens via wave_active below
  const int kv_end = causal ? min(S, (int)blockIdx.x * 128 + 128) : S;
  const float s2 = scale * LOG2E;

  stage_tile32(kp + kv_base, 0, FA_D, S, smem + DQ_LDS_K, smem + DQ_LDS_KT);
  {  // V row-major only: reuse stage but transposed target unused ->
     // cheap variant: stage V rm with same piece mapping
    int tid = threadIdx.x;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int p = tid + i * 256;
      int row = p >> 4, c16 = p & 15;
      int rr = row < S ? row : S - 1;
      *(uint4v*)(smem + DQ_LDS_V + row * (FA_D * 2) + swz(row, c16 * 16)) =
          *(const uint4v*)(vp + kv_base + (long)rr * FA_D + c16 * 8);
    }
  }

  for (int kv0 = 0; kv0 < kv_end; kv0 += 32) {
    __syncthreads();
    const bool wave_active = !causal || (kv0 <= q0 + 31);

    if (wave_active) {
      // S^T[k][q]: A = K rows (LDS rm, row k=col), B = Q^T (regs)
      f32x16 st = {};
      f32x16 dpt = {};
#pragma unroll
      for (int c = 0; c < 8; ++c) {
        frag_u kfr;
        kfr.u4 = *(const uint4v*)(smem + DQ_LDS_K + col * (FA_D * 2)
                                  + swz(col, (c * 16 + hi * 8) * 2));
        st = mfma_bf16(kfr.bf, qf[c].bf, st);
        frag_u vfr;
        vfr.u4 = *(const uint4v*)(smem + DQ_LDS_V + col * (FA_D * 2)
                                  + swz(col, (c * 16 + hi * 8) * 2));
        dpt = mfma_bf16(vfr.bf, dof[c].bf, dpt);
      }

      f32x16 dst;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int kg = kv0 + acc_row(r, hi);
        bool masked = (causal && kg > my_q) || kg >= S || my_q >= S;
        float p = masked ? 0.f : __builtin_exp2f(st[r] * s2 - lse2);
        dst[r] = p * (dpt[r] - dlt);
      }
      // write dS as [q][k] (row q = col): cols k = acc_row pairs packed b32
      {
#pragma unroll
        for (int r = 0; r < 16; r += 2) {
          int k0 = acc_row(r, hi);  // r even: k0, r+1 -> k0+1 (consecutive)
          uint pk = pack_bf16x2(dst[r], dst[r + 1]);
          *(uint*)(pw_ds + (col * PW_PITCH + k0) * 2) = pk;
        }
      }

      // dQ[q][d] += sum_k dS[q][k] K[k][d]
      //   A = dS rows q=col (pw tile), B = K[k][d] via Kt rows d
#pragma unroll
      for (int ck = 0; ck < 2; ++ck) {
        frag_u da;
        da.bf = read_pw_row(pw_ds, col, ck * 16 + hi * 8);
#pragma unroll
        for (int nb = 0; nb < 4; ++nb) {
          int d = nb * 32 + col;
          frag_u kfr;
          kfr.u4 = *(const uint4v*)(smem + DQ_LDS_KT + d * 64
                                    + swz32(d, (ck * 16 + hi * 8) * 2));
          dq_acc[nb] = mfma_bf16(da.bf, kfr.bf, dq_acc[nb]);
        }
      }
    }

    __syncthreads();
    if (kv0 + 32 < kv_end) {
      stage_tile32(kp + kv_base, kv0 + 32, FA_D, S - kv0 - 32,
                   smem + DQ_LDS_K, smem + DQ_LDS_KT);
      int tid = threadIdx.x;
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        int p = tid + i * 256;
        int row = p >> 4, c16 = p & 15;
        int rr = (kv0 + 32 + row) < S ? kv0 + 32 + row : S - 1;
        *(uint4v*)(smem + DQ_LDS_V + row * (FA_D * 2) + swz(row, c16 * 16)) =
            *(const uint4v*)(vp + kv_base + (long)rr * FA_D + c16 * 8);
      }
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
      *(uint4v*)(dqp + q_base + (long)qg * FA_D + c16 * 8) = vv;
    }
  }
}

// ---------------------------------------------------------------------------
extern "C" void flash_attn_bwd(const void* q, const void* k, const void* v,
                               const void* out, const void* dout,
                               const void* lse, void* delta, void* dq,
                               void* dk, void* dv, int B, int Hq, int Hkv,
                               int S, float scale, int causal,
                               hipStream_t stream) {
  long rows = (long)B * Hq * S;
  int nb = rows < 2048 ? (int)rows : 2048;
  fa_bwd_delta_kernel<<<nb, 256, 0, stream>>>((const short*)dout,
                                              (const short*)out,
                                              (float*)delta, rows);
  dim3 gkv((S + 127) / 128, Hq, B);
  size_t lds1 = 4 * 32 * FA_D * 2 + 4 * 2 * PW_BYTES;
  fa_bwd_dkdv_kernel<<<gkv, 256, lds1, stream>>>(
      (const short*)q, (const short*)k, (const short*)v, (const short*)dout,
      (const float*)lse, (const float*)delta, (short*)dk, (short*)dv, B, Hq,
      Hkv, S, scale, causal);
  dim3 gq((S + 127) / 128, Hq, B);
  size_t lds2 = 3 * 32 * FA_D * 2 + 4 * PW_BYTES;
  fa_bwd_dq_kernel<<<gq, 256, lds2, stream>>>(
      (const short*)q, (const short*)k, (const short*)v, (const short*)dout,
      (const float*)lse, (const float*)delta, (short*)dq, B, Hq, Hkv, S,
      scale, causal);
}
