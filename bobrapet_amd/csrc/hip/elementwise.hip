// Fused elementwise / normalization kernels for MI355X (gfx950).
//
// All memory-bound: target is the HBM roofline (≈6.3 TB/s), so every
// kernel loads bf16 as ushort8 (16 B/lane) per guide G13 and fuses the
// adjacent elementwise work into one pass (residual-add into rmsnorm,
// silu into the gate*up product, rope pairs in one read).
#include "common.h"

// ---------------------------------------------------------------------------
// RMSNorm: out[r][c] = x[r][c] * rsqrt(mean(x[r]^2)+eps) * w[c]
// One 256-thread workgroup per row; row length H is a multiple of 8.
// fused variant: residual += x first (llama pre-norm block pattern), the
// updated residual is both written back and normalized.
// ---------------------------------------------------------------------------

template <bool FUSED_ADD>
__global__ __launch_bounds__(256) void rmsnorm_kernel(
    unsigned short* __restrict__ out,         // [R][H] bf16
    unsigned short* __restrict__ x,           // [R][H] bf16 (input)
    unsigned short* __restrict__ residual,    // [R][H] bf16 (in/out, FUSED_ADD)
    const unsigned short* __restrict__ w,     // [H]
    int H, float eps) {
  const int row = blockIdx.x;
  const long base = (long)row * H;
  const int tid = threadIdx.x;
  const int nthread = blockDim.x;

  float ssq = 0.f;
  // pass 1: (optional residual add) + sum of squares; values kept in regs
  // for H<=8192 with 256 threads → ≤4 ushort8 chunks per thread
  float vals[4][8];
  ushort8v raw[4];
  int nchunk = 0;
  for (int c = tid * 8; c < H; c += nthread * 8) {
    ushort8v v = *reinterpret_cast<const ushort8v*>(x + base + c);
    if (FUSED_ADD) {
      ushort8v r = *reinterpret_cast<const ushort8v*>(residual + base + c);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        // round the sum to bf16 FIRST: the normalized value must match the
        // bf16 residual that is written back (and the torch reference)
        v[j] = f2bf(bf2f(v[j]) + bf2f(r[j]));
        vals[nchunk][j] = bf2f(v[j]);
      }
      *reinterpret_cast<ushort8v*>(residual + base + c) = v;  // write back sum
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) vals[nchunk][j] = bf2f(v[j]);
    }
    raw[nchunk] = v;
#pragma unroll
    for (int j = 0; j < 8; ++j) ssq += vals[nchunk][j] * vals[nchunk][j];
    ++nchunk;
  }

  // wave reduce + cross-wave via LDS
  __shared__ float warp_ssq[8];
  ssq = wave_reduce_sum(ssq);
  const int wid = tid / WAVE;
  if ((tid & (WAVE - 1)) == 0) warp_ssq[wid] = ssq;
  __syncthreads();
  float total = 0.f;
#pragma unroll
  for (int i = 0; i < 8; ++i)
    if (i < (nthread + WAVE - 1) / WAVE) total += warp_ssq[i];
  const float scale = rsqrtf(total / (float)H + eps);

  // pass 2: scale by rsqrt * w (x values already in registers)
  int k = 0;
  for (int c = tid * 8; c < H; c += nthread * 8, ++k) {
    ushort8v wv = *reinterpret_cast<const ushort8v*>(w + c);
    ushort8v o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bf(vals[k][j] * scale * bf2f(wv[j]));
    *reinterpret_cast<ushort8v*>(out + base + c) = o;
  }
  (void)raw;
}

extern "C" void launch_rmsnorm(void* out, void* x, void* residual, const void* w,
                               int rows, int H, float eps, bool fused_add,
                               hipStream_t stream) {
  dim3 grid(rows), block(256);
  if (fused_add)
    hipLaunchKernelGGL((rmsnorm_kernel<true>), grid, block, 0, stream,
                       (unsigned short*)out, (unsigned short*)x,
                       (unsigned short*)residual, (const unsigned short*)w, H, eps);
  else
    hipLaunchKernelGGL((rmsnorm_kernel<false>), grid, block, 0, stream,
                       (unsigned short*)out, (unsigned short*)x, nullptr,
                       (const unsigned short*)w, H, eps);
}

// ---------------------------------------------------------------------------
// SiLU-mul (SwiGLU): out = silu(gate) * up, all [N] bf16, N % 8 == 0.
// Grid-stride, capped grid (guide G11).
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void silu_mul_kernel(
    unsigned short* __restrict__ out, const unsigned short* __restrict__ gate,
    const unsigned short* __restrict__ up, long n8) {
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    ushort8v g = reinterpret_cast<const ushort8v*>(gate)[i];
    ushort8v u = reinterpret_cast<const ushort8v*>(up)[i];
    ushort8v o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf2f(g[j]);
      float s = gf / (1.f + __expf(-gf));
      o[j] = f2bf(s * bf2f(u[j]));
    }
    reinterpret_cast<ushort8v*>(out)[i] = o;
  }
}

__global__ __launch_bounds__(256) void silu_mul_strided_kernel(
    unsigned short* __restrict__ out, const unsigned short* __restrict__ gate,
    const unsigned short* __restrict__ up, long rows, int inner,
    long row_stride) {
  const long n8 = rows * (inner / 8);
  const int i8 = inner / 8;
  for (long idx = blockIdx.x * blockDim.x + threadIdx.x; idx < n8;
       idx += (long)gridDim.x * blockDim.x) {
    long r = idx / i8;
    int c = (int)(idx % i8) * 8;
    ushort8v g = *reinterpret_cast<const ushort8v*>(gate + r * row_stride + c);
    ushort8v u = *reinterpret_cast<const ushort8v*>(up + r * row_stride + c);
    ushort8v o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf2f(g[j]);
      float sg = gf / (1.f + __expf(-gf));
      o[j] = f2bf(sg * bf2f(u[j]));
    }
    *reinterpret_cast<ushort8v*>(out + r * (long)inner + c) = o;
  }
}

extern "C" void launch_silu_mul_strided(void* out, const void* gate,
                                        const void* up, long rows, int inner,
                                        long row_stride, hipStream_t stream) {
  long n8 = rows * (inner / 8);
  int blocks = (int)((n8 + 255) / 256);
  if (blocks > 2048) blocks = 2048;
  hipLaunchKernelGGL(silu_mul_strided_kernel, dim3(blocks), dim3(256), 0,
                     stream, (unsigned short*)out, (const unsigned short*)gate,
                     (const unsigned short*)up, rows, inner, row_stride);
}

extern "C" void launch_silu_mul(void* out, const void* gate, const void* up,
                                long n, hipStream_t stream) {
  long n8 = n / 8;
  int blocks = (int)((n8 + 255) / 256);
  if (blocks > 2048) blocks = 2048;
  hipLaunchKernelGGL(silu_mul_kernel, dim3(blocks), dim3(256), 0, stream,
                     (unsigned short*)out, (const unsigned short*)gate,
                     (const unsigned short*)up, n8);
}

// ---------------------------------------------------------------------------
// RoPE (NeoX half-rotation, llama style), in-place on q and k.
//   q: [T, Hq, D] bf16, k: [T, Hk, D] bf16 (T = flattened batch*seq)
//   cos/sin: [T, D/2] f32 precomputed on host per position (guide: trig
//   tables on HOST; on-device sinf/cosf turns memory-bound into VALU-bound)
// Rotation: for d in [0, D/2): (a, b) = (x[d], x[d+D/2])
//   x[d] = a*cos - b*sin ; x[d+D/2] = b*cos + a*sin
// One wave handles one (token, head): D/2=64 pairs → 1 pair/lane.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void rope_kernel(
    unsigned short* __restrict__ q, unsigned short* __restrict__ k,
    const float* __restrict__ cos_t, const float* __restrict__ sin_t, int T,
    int Hq, int Hk, int D) {
  const int half = D / 2;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int total = T * (Hq + Hk);
  const int nwaves = (gridDim.x * blockDim.x) / WAVE;
  for (int idx = wave; idx < total; idx += nwaves) {
    const int t = idx / (Hq + Hk);
    const int h = idx % (Hq + Hk);
    unsigned short* base = (h < Hq) ? q + ((long)t * Hq + h) * D
                                    : k + ((long)t * Hk + (h - Hq)) * D;
    const float* crow = cos_t + (long)t * half;
    const float* srow = sin_t + (long)t * half;
    // each lane rotates pairs (lane, lane+64, ...) — D=128 → one pair/lane
    for (int d = lane; d < half; d += WAVE) {
      float a = bf2f(base[d]);
      float b = bf2f(base[d + half]);
      float c = crow[d], s = srow[d];
      base[d] = f2bf(a * c - b * s);
      base[d + half] = f2bf(b * c + a * s);
    }
  }
}

// Fused rope-from-qkv: reads q/k heads from the fused qkv projection
// output (row stride = qkv_out) and writes rotated CONTIGUOUS [T,H,D]
// tensors — removes the slice-contiguous copies on the model hot path.
__global__ __launch_bounds__(256) void rope_qkv_kernel(
    unsigned short* __restrict__ q_out,     // [T,Hq,D]
    unsigned short* __restrict__ k_out,     // [T,Hk,D]
    const unsigned short* __restrict__ qkv, // [T, row_stride elems]
    const float* __restrict__ cos_t, const float* __restrict__ sin_t, int T,
    int Hq, int Hk, int D, long row_stride) {
  const int half = D / 2;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int total = T * (Hq + Hk);
  const int nwaves = (gridDim.x * blockDim.x) / WAVE;
  for (int idx = wave; idx < total; idx += nwaves) {
    const int t = idx / (Hq + Hk);
    const int h = idx % (Hq + Hk);
    // q heads sit at [0, Hq*D); k heads follow at [Hq*D, (Hq+Hk)*D)
    const unsigned short* src = qkv + (long)t * row_stride + (long)h * D;
    unsigned short* dst = (h < Hq) ? q_out + ((long)t * Hq + h) * D
                                   : k_out + ((long)t * Hk + (h - Hq)) * D;
    const float* crow = cos_t + (long)t * half;
    const float* srow = sin_t + (long)t * half;
    for (int d = lane; d < half; d += WAVE) {
      float a = bf2f(src[d]);
      float b = bf2f(src[d + half]);
      float c = crow[d], sn = srow[d];
      dst[d] = f2bf(a * c - b * sn);
      dst[d + half] = f2bf(b * c + a * sn);
    }
  }
}

extern "C" void launch_rope_qkv(void* q_out, void* k_out, const void* qkv,
                                const void* cos_t, const void* sin_t, int T,
                                int Hq, int Hk, int D, long row_stride,
                                hipStream_t stream) {
  int total_waves = T * (Hq + Hk);
  int blocks = (total_waves * WAVE + 255) / 256;
  if (blocks > 4096) blocks = 4096;
  hipLaunchKernelGGL(rope_qkv_kernel, dim3(blocks), dim3(256), 0, stream,
                     (unsigned short*)q_out, (unsigned short*)k_out,
                     (const unsigned short*)qkv, (const float*)cos_t,
                     (const float*)sin_t, T, Hq, Hk, D, row_stride);
}

// Fused DECODE step head prep: one kernel replaces rope_inplace + two
// cache index_copys + the k/v contiguous copies on the b1 decode path
// (each was a ~5 us launch, x32 layers x2 copies per step).  Reads the
// fused qkv projection row, applies rope to q/k at position L (a DEVICE
// scalar — hipGraph-replayable), writes rotated q to a contiguous
// [B,Hq,D] buffer and rotated k + raw v DIRECTLY into the KV cache at
// sequence slot L.
__global__ __launch_bounds__(256) void rope_qkv_decode_kernel(
    unsigned short* __restrict__ q_out,      // [B,Hq,D]
    unsigned short* __restrict__ kc,         // [B,Hkv,Smax,D]
    unsigned short* __restrict__ vc,         // [B,Hkv,Smax,D]
    const unsigned short* __restrict__ qkv,  // [B, (Hq+2*Hkv)*D]
    const float* __restrict__ inv_freq,      // [D/2] rope inverse freqs
    const int* __restrict__ L_dev, int B, int Hq, int Hkv, int D, int Smax) {
  const int L = *L_dev;
  const int half = D / 2;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int nwaves = (gridDim.x * blockDim.x) / WAVE;
  const long row = (long)(Hq + 2 * Hkv) * D;
  const int total = B * (Hq + 2 * Hkv);
  for (int idx = wave; idx < total; idx += nwaves) {
    const int b = idx / (Hq + 2 * Hkv);
    const int h = idx % (Hq + 2 * Hkv);
    const unsigned short* src = qkv + b * row + (long)h * D;
    if (h < Hq + Hkv) {  // rope'd q or k head
      unsigned short* dst =
          (h < Hq) ? q_out + ((long)b * Hq + h) * D
                   : kc + (((long)b * Hkv + (h - Hq)) * Smax + L) * D;
      for (int d = lane; d < half; d += WAVE) {
        float a = bf2f(src[d]);
        float bb = bf2f(src[d + half]);
        // angles computed in-kernel from inv_freq (the
        // host-side cos/sin table build was ~6 at::native launches/step)
        const float ang = (float)L * inv_freq[d];
        float sn, c;
        __sincosf(ang, &sn, &c);
        dst[d] = f2bf(a * c - bb * sn);
        dst[d + half] = f2bf(bb * c + a * sn);
      }
    } else {  // v head: straight copy into the cache slot
      int hv = h - Hq - Hkv;
      unsigned short* dst = vc + (((long)b * Hkv + hv) * Smax + L) * D;
      for (int d = lane; d < D; d += WAVE) dst[d] = src[d];
    }
  }
}

extern "C" void launch_rope_qkv_decode(void* q_out, void* kc, void* vc,
                                       const void* qkv, const void* inv_freq,
                                       const void* L_dev, int B, int Hq,
                                       int Hkv, int D, int Smax,
                                       hipStream_t stream) {
  int waves_needed = B * (Hq + 2 * Hkv);
  int blocks = (waves_needed + 3) / 4;  // 4 waves per 256-thread block
  hipLaunchKernelGGL(rope_qkv_decode_kernel, dim3(blocks), dim3(256), 0,
                     stream, (unsigned short*)q_out, (unsigned short*)kc,
                     (unsigned short*)vc, (const unsigned short*)qkv,
                     (const float*)inv_freq, (const int*)L_dev, B, Hq, Hkv, D,
                     Smax);
}

// ---------------------------------------------------------------------------
// row argmax: ids[m] = argmax_n x[m][n]  (greedy sampling — replaces the
// at::native argmax on [B, V~128k] logits)
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void argmax_rows_kernel(
    long* __restrict__ ids, const unsigned short* __restrict__ x, int N) {
  const int m = blockIdx.x;
  const unsigned short* row = x + (long)m * N;
  const int tid = threadIdx.x;
  float best = -3.4e38f;
  int bi = 0;
  for (int n = tid * 8; n + 7 < N; n += 256 * 8) {
    ushort8v v = *reinterpret_cast<const ushort8v*>(row + n);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float f = bf2f(v[j]);
      if (f > best) {
        best = f;
        bi = n + j;
      }
    }
  }
  for (int n = (N / 8) * 8 + tid; n < N; n += 256) {
    const float f = bf2f(row[n]);
    if (f > best) {
      best = f;
      bi = n;
    }
  }
  // wave reduce (value, index) — ties resolve to the LOWEST index like
  // torch.argmax
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float ob = __shfl_xor(best, off, WAVE);
    const int oi = __shfl_xor(bi, off, WAVE);
    if (ob > best || (ob == best && oi < bi)) {
      best = ob;
      bi = oi;
    }
  }
  __shared__ float sb[4];
  __shared__ int si[4];
  const int w = tid >> 6;
  if ((tid & (WAVE - 1)) == 0) {
    sb[w] = best;
    si[w] = bi;
  }
  __syncthreads();
  if (tid == 0) {
    for (int i = 1; i < 4; ++i)
      if (sb[i] > best || (sb[i] == best && si[i] < bi)) {
        best = sb[i];
        bi = si[i];
      }
    ids[m] = (long)bi;
  }
}

extern "C" void launch_argmax_rows(void* ids, const void* x, int M, int N,
                                   hipStream_t stream) {
  hipLaunchKernelGGL(argmax_rows_kernel, dim3(M), dim3(256), 0, stream,
                     (long*)ids, (const unsigned short*)x, N);
}

extern "C" void launch_rope(void* q, void* k, const void* cos_t,
                            const void* sin_t, int T, int Hq, int Hk, int D,
                            hipStream_t stream) {
  int total_waves = T * (Hq + Hk);
  int blocks = (total_waves * WAVE + 255) / 256;
  if (blocks > 4096) blocks = 4096;
  hipLaunchKernelGGL(rope_kernel, dim3(blocks), dim3(256), 0, stream,
                     (unsigned short*)q, (unsigned short*)k,
                     (const float*)cos_t, (const float*)sin_t, T, Hq, Hk, D);
}

// ---------------------------------------------------------------------------
// Embed: fused gather + mean-pool + L2-normalize (the `embed` engram's hot
// op): out[b] = normalize(mean over s of table[ids[b][s]]).
//   table: [V, H] bf16; ids: [B, S] int32; out: [B, H] bf16
// Two phases for grid parallelism (one-WG-per-row was ~300 GB/s):
//   A: grid (B x S-chunks x H-slices) partial sums -> f32 atomics
//   B: grid (B) normalize: /S, L2, cast bf16
// ---------------------------------------------------------------------------

#define EMB_SCHUNK 8
#define EMB_HSLICE 2048  // elems per slice (256 thr x 8)

__global__ __launch_bounds__(256) void embed_pool_sum_kernel(
    float* __restrict__ pooled,               // [B][H] f32, pre-zeroed
    const unsigned short* __restrict__ table, // [V][H]
    const int* __restrict__ ids,              // [B][S]
    int B, int S, int H, int V) {
  const int nslice = (H + EMB_HSLICE - 1) / EMB_HSLICE;
  const int nchunk = (S + EMB_SCHUNK - 1) / EMB_SCHUNK;
  int wg = blockIdx.x;
  const int b = wg / (nchunk * nslice);
  const int chunk = (wg / nslice) % nchunk;
  const int slice = wg % nslice;
  const int tid = threadIdx.x;
  const int h0 = slice * EMB_HSLICE + tid * 8;
  if (h0 + 8 > H) return;

  float acc[8] = {};
  const int s0 = chunk * EMB_SCHUNK;
  const int s_end = min(s0 + EMB_SCHUNK, S);
  // preload the chunk's ids, then issue ALL row gathers before consuming:
  // the naive loop serializes id-load -> row-load dependent chains (8 HBM
  // round-trips per WG); splitting the phases keeps 8 gathers in flight
  int id_reg[EMB_SCHUNK];
#pragma unroll
  for (int i = 0; i < EMB_SCHUNK; ++i) {
    int sidx = s0 + i;
    id_reg[i] = (sidx < s_end) ? ids[(long)b * S + sidx] : -1;
  }
  ushort8v rows[EMB_SCHUNK];
#pragma unroll
  for (int i = 0; i < EMB_SCHUNK; ++i) {
    int id = id_reg[i];
    rows[i] = (id >= 0 && id < V)
                  ? *reinterpret_cast<const ushort8v*>(table + (long)id * H + h0)
                  : ushort8v{0, 0, 0, 0, 0, 0, 0, 0};
  }
#pragma unroll
  for (int i = 0; i < EMB_SCHUNK; ++i)
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] += bf2f(rows[i][j]);
  // per-chunk partial slot, plain coalesced stores — the atomicAdd version
  // of this kernel was atomic-throughput bound (16-way contention per
  // pooled element, ~0.57 TB/s effective); partials + merge-in-norm stream
  float* dst = pooled + (((long)b * nchunk + chunk) * H) + h0;
  float4v lo, hi;
#pragma unroll
  for (int j = 0; j < 4; ++j) { lo[j] = acc[j]; hi[j] = acc[j + 4]; }
  *reinterpret_cast<float4v*>(dst) = lo;
  *reinterpret_cast<float4v*>(dst + 4) = hi;
}

__global__ __launch_bounds__(256) void embed_pool_norm_kernel(
    unsigned short* __restrict__ out, const float* __restrict__ pooled, int S,
    int H, int nchunk) {
  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  const float inv_s = 1.f / (float)S;
  float vals[4][8];
  float ssq = 0.f;
  int k = 0;
  for (int c = tid * 8; c < H; c += 256 * 8, ++k) {
    float sum[8] = {};
    for (int ch = 0; ch < nchunk; ++ch) {
      const float* src = pooled + (((long)b * nchunk + ch) * H) + c;
      const float4v lo = *reinterpret_cast<const float4v*>(src);
      const float4v hic = *reinterpret_cast<const float4v*>(src + 4);
#pragma unroll
      for (int j = 0; j < 4; ++j) { sum[j] += lo[j]; sum[j + 4] += hic[j]; }
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) vals[k][j] = sum[j] * inv_s;
#pragma unroll
    for (int j = 0; j < 8; ++j) ssq += vals[k][j] * vals[k][j];
  }
  __shared__ float warp_ssq[8];
  ssq = wave_reduce_sum(ssq);
  if ((tid & (WAVE - 1)) == 0) warp_ssq[tid / WAVE] = ssq;
  __syncthreads();
  float total = 1e-12f;
#pragma unroll
  for (int i = 0; i < 4; ++i) total += warp_ssq[i];
  const float inv_norm = rsqrtf(total);
  k = 0;
  for (int c = tid * 8; c < H; c += 256 * 8, ++k) {
    ushort8v o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bf(vals[k][j] * inv_norm);
    *reinterpret_cast<ushort8v*>(out + (long)b * H + c) = o;
  }
}

extern "C" void launch_embed_pool(void* out, void* pooled_f32, const void* table,
                                  const void* ids, int B, int S, int H, int V,
                                  hipStream_t stream) {
  const int nslice = (H + EMB_HSLICE - 1) / EMB_HSLICE;
  const int nchunk = (S + EMB_SCHUNK - 1) / EMB_SCHUNK;
  hipLaunchKernelGGL(embed_pool_sum_kernel, dim3(B * nchunk * nslice), dim3(256),
                     0, stream, (float*)pooled_f32,
                     (const unsigned short*)table, (const int*)ids, B, S, H, V);
  hipLaunchKernelGGL(embed_pool_norm_kernel, dim3(B), dim3(256), 0, stream,
                     (unsigned short*)out, (const float*)pooled_f32, S, H,
                     nchunk);
}

// ---------------------------------------------------------------------------
// Residual add (plain): out = a + b, bf16, vectorized.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void add_kernel(
    unsigned short* __restrict__ out, const unsigned short* __restrict__ a,
    const unsigned short* __restrict__ b, long n8) {
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    ushort8v x = reinterpret_cast<const ushort8v*>(a)[i];
    ushort8v y = reinterpret_cast<const ushort8v*>(b)[i];
    ushort8v o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bf(bf2f(x[j]) + bf2f(y[j]));
    reinterpret_cast<ushort8v*>(out)[i] = o;
  }
}

extern "C" void launch_add_bf16(void* out, const void* a, const void* b, long n,
                                hipStream_t stream) {
  long n8 = n / 8;
  int blocks = (int)((n8 + 255) / 256);
  if (blocks > 2048) blocks = 2048;
  hipLaunchKernelGGL(add_kernel, dim3(blocks), dim3(256), 0, stream,
                     (unsigned short*)out, (const unsigned short*)a,
                     (const unsigned short*)b, n8);
}
