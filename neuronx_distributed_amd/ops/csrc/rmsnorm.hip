// Fused RMSNorm forward/backward for CDNA4 (gfx950).
//
// Replaces the reference's torch-level RMSNorm that the Neuron compiler
// fuses (reference modules/rms_norm.py:10-36) — eager ROCm will not fuse,
// so this is a hand-written HBM-bound kernel: one pass, vectorized 16B
// bf16 loads (guide G13), fp32 accumulation, per-row rstd saved for bwd.
//
// Layout: x (N, H) bf16 rows; w (H) bf16; out (N, H) bf16; rstd (N) f32.

#include "common.h"

// ---------------------------------------------------------------------------
// forward: out = x * rsqrt(mean(x^2) + eps) * w
// one block per ROWS_PER_BLOCK rows; vector short8 loads
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
rmsnorm_fwd_kernel(const short* __restrict__ x, const short* __restrict__ w,
                   short* __restrict__ out, float* __restrict__ rstd_out,
                   int H, float eps, int rows) {
  __shared__ float scratch[16];
  int row = blockIdx.x;
  if (row >= rows) return;
  const short* xr = x + (long)row * H;
  short* outr = out + (long)row * H;

  int nvec = H >> 3;  // H % 8 == 0 required
  float ss = 0.f;
  // register-cache the row between the stats pass and the normalize
  // pass (4 s8v/thread covers H <= 8192): 2 HBM passes instead of 3
  s8v cache[4];
  const bool cached = nvec <= (int)blockDim.x * 4;
  int ci = 0;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x, ++ci) {
    s8v v = *(const s8v*)(xr + i * 8);
    if (cached && ci < 4) cache[ci] = v;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bits2f(v[j]);
      ss += f * f;
    }
  }
  ss = block_reduce_sum(ss, scratch);
  float rstd = rsqrtf(ss / (float)H + eps);
  if (threadIdx.x == 0 && rstd_out) rstd_out[row] = rstd;

  ci = 0;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x, ++ci) {
    s8v v = (cached && ci < 4) ? cache[ci] : *(const s8v*)(xr + i * 8);
    s8v wv = *(const s8v*)(w + i * 8);
    s8v o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = f2bits(bits2f(v[j]) * rstd * bits2f(wv[j]));
    *(s8v*)(outr + i * 8) = o;
  }
}

// ---------------------------------------------------------------------------
// backward:
//   dx = rstd * w * dy - rstd^3/H * x * sum(dy * w * x)
//   dw += sum_rows(dy * x * rstd)        (fp32 atomics into dw_f32)
// ---------------------------------------------------------------------------
// grid-stride over rows; per-block dw partial accumulated in LDS (H*4 B
// dynamic LDS, H<=32768), ONE atomicAdd per element per block at the end —
// avoids rows*H atomic contention.
extern "C" __global__ void __launch_bounds__(256)
rmsnorm_bwd_kernel(const short* __restrict__ x, const short* __restrict__ w,
                   const short* __restrict__ dy,
                   const float* __restrict__ rstd_in,
                   const short* __restrict__ dpass,
                   short* __restrict__ dx, float* __restrict__ dw_f32,
                   int H, int rows) {
  // dpass != nullptr: dx += dpass — the residual fork's pass-through
  // gradient is folded into this pass (training fused add+RMSNorm site),
  // replacing a separate 3-pass eager add kernel.
  extern __shared__ __attribute__((aligned(16))) float smem[];
  float* dw_part = smem;           // [H]
  float* scratch = smem + H;       // [16]

  for (int i = threadIdx.x; i < H; i += blockDim.x) dw_part[i] = 0.f;
  __syncthreads();

  int nvec = H >> 3;
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const short* xr = x + (long)row * H;
    const short* dyr = dy + (long)row * H;
    short* dxr = dx + (long)row * H;
    float rstd = rstd_in[row];

    float dot = 0.f;  // sum(dy * w * x)
    // register-cache x and dy between the dot pass and the dx pass
    // (2 x 4 s8v/thread covers H <= 8192): ~5 HBM passes instead of 7
    s8v xcache[4], dcache[4];
    const bool cached = nvec <= (int)blockDim.x * 4;
    int ci = 0;
    for (int i = threadIdx.x; i < nvec; i += blockDim.x, ++ci) {
      s8v xv = *(const s8v*)(xr + i * 8);
      s8v dv = *(const s8v*)(dyr + i * 8);
      s8v wv = *(const s8v*)(w + i * 8);
      if (cached && ci < 4) { xcache[ci] = xv; dcache[ci] = dv; }
#pragma unroll
      for (int j = 0; j < 8; ++j)
        dot += bits2f(dv[j]) * bits2f(wv[j]) * bits2f(xv[j]);
    }
    dot = block_reduce_sum(dot, scratch);
    float k = rstd * rstd * rstd * dot / (float)H;

    const short* pr = dpass ? dpass + (long)row * H : nullptr;
    ci = 0;
    for (int i = threadIdx.x; i < nvec; i += blockDim.x, ++ci) {
      s8v xv = (cached && ci < 4) ? xcache[ci] : *(const s8v*)(xr + i * 8);
      s8v dv = (cached && ci < 4) ? dcache[ci] : *(const s8v*)(dyr + i * 8);
      s8v wv = *(const s8v*)(w + i * 8);
      s8v pv;
      if (pr) pv = *(const s8v*)(pr + i * 8);
      s8v o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xf = bits2f(xv[j]);
        float df = bits2f(dv[j]);
        float g = rstd * bits2f(wv[j]) * df - k * xf;
        if (pr) g += bits2f(pv[j]);
        o[j] = f2bits(g);
        dw_part[i * 8 + j] += df * xf * rstd;  // thread-exclusive slot per i
      }
      *(s8v*)(dxr + i * 8) = o;
    }
    __syncthreads();  // dw_part reuse across rows is thread-local per index
  }
  __syncthreads();
  for (int i = threadIdx.x; i < H; i += blockDim.x)
    if (dw_part[i] != 0.f) atomicAdd(dw_f32 + i, dw_part[i]);
}

// ---------------------------------------------------------------------------
// fused residual-add + RMSNorm (inference):
//   res_out = res_in + delta;  normed = res_out * rsqrt(mean(res_out^2)+eps) * w
// Collapses the decode path's two elementwise passes per norm site into one
// kernel and reads res/delta exactly once (HBM-bound; decode steps are
// kernel-count-bound inside hipGraph replay).
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
add_rmsnorm_fwd_kernel(const short* __restrict__ res_in,
                       const short* __restrict__ delta,
                       const short* __restrict__ w,
                       short* __restrict__ res_out,
                       short* __restrict__ normed,
                       float* __restrict__ rstd_out, int H, float eps,
                       int rows) {
  __shared__ float scratch[16];
  int row = blockIdx.x;
  if (row >= rows) return;
  const short* rr = res_in + (long)row * H;
  const short* dr = delta + (long)row * H;
  short* ror = res_out + (long)row * H;
  short* nr = normed + (long)row * H;

  int nvec = H >> 3;
  float ss = 0.f;
  // register-cache the summed row (up to 4 s8v per thread = H <= 8192)
  // so the second pass never re-reads it — decode rows are tiny and the
  // kernel is latency-bound, not bandwidth-bound
  s8v cache[4];
  const bool cached = nvec <= (int)blockDim.x * 4;
  int ci = 0;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x, ++ci) {
    s8v rv = *(const s8v*)(rr + i * 8);
    s8v dv = *(const s8v*)(dr + i * 8);
    s8v o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bits2f(rv[j]) + bits2f(dv[j]);
      o[j] = f2bits(f);
      f = bits2f(o[j]);  // norm statistics on the ROUNDED bf16 sum so the
                         // result matches an unfused add -> rmsnorm chain
      ss += f * f;
    }
    if (cached && ci < 4) cache[ci] = o;
    *(s8v*)(ror + i * 8) = o;
  }
  ss = block_reduce_sum(ss, scratch);
  float rstd = rsqrtf(ss / (float)H + eps);
  if (rstd_out && threadIdx.x == 0) rstd_out[row] = rstd;

  ci = 0;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x, ++ci) {
    s8v v = (cached && ci < 4) ? cache[ci] : *(const s8v*)(ror + i * 8);
    s8v wv = *(const s8v*)(w + i * 8);
    s8v o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = f2bits(bits2f(v[j]) * rstd * bits2f(wv[j]));
    *(s8v*)(nr + i * 8) = o;
  }
}

extern "C" void add_rmsnorm_fwd(const void* res_in, const void* delta,
                                const void* w, void* res_out, void* normed,
                                int rows, int H, float eps,
                                hipStream_t stream) {
  add_rmsnorm_fwd_kernel<<<rows, 256, 0, stream>>>(
      (const short*)res_in, (const short*)delta, (const short*)w,
      (short*)res_out, (short*)normed, nullptr, H, eps, rows);
}

// training variant: also emits per-row rstd for the backward
extern "C" void add_rmsnorm_fwd_train(const void* res_in, const void* delta,
                                      const void* w, void* res_out,
                                      void* normed, void* rstd, int rows,
                                      int H, float eps, hipStream_t stream) {
  add_rmsnorm_fwd_kernel<<<rows, 256, 0, stream>>>(
      (const short*)res_in, (const short*)delta, (const short*)w,
      (short*)res_out, (short*)normed, (float*)rstd, H, eps, rows);
}

extern "C" void rmsnorm_fwd(const void* x, const void* w, void* out,
                            void* rstd, int rows, int H, float eps,
                            hipStream_t stream) {
  rmsnorm_fwd_kernel<<<rows, 256, 0, stream>>>(
      (const short*)x, (const short*)w, (short*)out, (float*)rstd, H, eps, rows);
}

extern "C" void rmsnorm_bwd(const void* x, const void* w, const void* dy,
                            const void* rstd, void* dx, void* dw_f32, int rows,
                            int H, hipStream_t stream) {
  int blocks = rows < 2048 ? rows : 2048;
  size_t lds = (size_t)(H + 16) * sizeof(float);
  rmsnorm_bwd_kernel<<<blocks, 256, lds, stream>>>(
      (const short*)x, (const short*)w, (const short*)dy, (const float*)rstd,
      nullptr, (short*)dx, (float*)dw_f32, H, rows);
}

// fused add+RMSNorm backward: dx = rmsnorm_bwd(dy) + dpass in one pass
extern "C" void rmsnorm_bwd_add(const void* x, const void* w, const void* dy,
                                const void* rstd, const void* dpass, void* dx,
                                void* dw_f32, int rows, int H,
                                hipStream_t stream) {
  int blocks = rows < 2048 ? rows : 2048;
  size_t lds = (size_t)(H + 16) * sizeof(float);
  rmsnorm_bwd_kernel<<<blocks, 256, lds, stream>>>(
      (const short*)x, (const short*)w, (const short*)dy, (const float*)rstd,
      (const short*)dpass, (short*)dx, (float*)dw_f32, H, rows);
}
