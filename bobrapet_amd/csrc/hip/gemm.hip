// bf16 MFMA GEMM for MI355X (gfx950): C[M,N] = A[M,K] @ B[N,K]^T.
//
// NT layout (B stored row-major as [N][K], the nn.Linear weight
// convention) so BOTH operands stage through LDS with the same
// lane-linear global_load_lds image — glds cannot transpose (guide §5
// "glds K-tile trap"), so the transposed-B read problem is solved in the
// weight layout instead.
//
// Structure (guide §5 ladder step 3 + T1/T2/T3 minimum 2-phase):
//  - 128x128 tile, 4 waves (2x2), per-wave 64x64 = 4x4 fragments of
//    v_mfma_f32_16x16x32_bf16; BK = 64 (two k-subtiles of 32)
//  - global->LDS staging via __builtin_amdgcn_global_load_lds width 16,
//    double-buffered; the LDS image is lane-linear so the bank-conflict
//    XOR swizzle lives on the per-lane SOURCE address and the matching
//    read offset (guide rule 21)
//  - XCD-aware bijective blockIdx swizzle (T1) for L2 locality
#include <cstdlib>

#include "common.h"

#define GM_BM 128
#define GM_BN 128
#define GM_BK 64

typedef float f32x4g __attribute__((ext_vector_type(4)));

// A/B LDS image: [128 rows][64 k] bf16 = 128-byte rows, 16 KB per tile.
// Swizzle: byte-in-row ^= ((row>>1)&7)<<4 — combined with the row-parity
// bank bit this makes the 16-lane ds_read_b128 fragment reads (rows
// distinct mod 16 within a group) conflict-free.
__device__ __forceinline__ int gm_swz(int row, int byte_in_row) {
  return row * 128 + (byte_in_row ^ (((row >> 1) & 7) << 4));
}

__global__ __launch_bounds__(256, 2) void gemm_bf16_nt_kernel(
    unsigned short* __restrict__ C,        // [M][N] bf16
    const unsigned short* __restrict__ A,  // [M][K] bf16
    const unsigned short* __restrict__ B,  // [N][K] bf16
    int M, int N, int K) {
  __shared__ __attribute__((aligned(16))) char smem[4 * GM_BM * GM_BK * 2];
  // buffers: [buf][A|B] each 16 KB
  auto tile_ptr = [&](int buf, int which) -> char* {
    return smem + (buf * 2 + which) * (GM_BM * GM_BK * 2);
  };

  // XCD-aware bijective remap (guide T1)
  const int nbm = M / GM_BM, nbn = N / GM_BN;
  unsigned int wgid = xcd_swizzle(blockIdx.x, nbm * nbn);
  const int bm = (int)(wgid / nbn) * GM_BM;
  const int bn = (int)(wgid % nbn) * GM_BN;

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;           // 4 waves as 2x2
  const int wr = (wid >> 1) * 64;     // wave row offset in the tile
  const int wc = (wid & 1) * 64;      // wave col offset
  const int l15 = lane & 15;
  const int lg = lane >> 4;           // 0..3 fragment k-group

  // ---- staging: each wave issues 4 glds per operand per tile; one glds
  // writes 1 KB of LDS (64 lanes x 16 B) lane-linearly at a wave-uniform
  // base.  chunk = (wid*4 + it)*64 + lane; row = chunk/8; the source
  // byte offset applies the inverse swizzle (rule 21).
  auto stage = [&](int buf, int kt) {
    const long k0 = (long)kt * GM_BK;
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      int piece = wid * 4 + it;           // 0..15, wave-uniform
      int chunk = piece * 64 + lane;      // 16-B chunk index
      int row = chunk >> 3;               // 8 chunks per 128-B row
      int bq = (chunk & 7) * 16;          // byte-in-row of the LDS image
      int src_byte = bq ^ (((row >> 1) & 7) << 4);
      {  // A tile: rows bm+row
        const unsigned short* src = A + (long)(bm + row) * K + k0 + src_byte / 2;
        char* dst = tile_ptr(buf, 0) + piece * 1024;
        __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) unsigned int*)src,
                                         (__attribute__((address_space(3))) unsigned int*)dst,
                                         16, 0, 0);
      }
      {  // B tile: rows bn+row
        const unsigned short* src = B + (long)(bn + row) * K + k0 + src_byte / 2;
        char* dst = tile_ptr(buf, 1) + piece * 1024;
        __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) unsigned int*)src,
                                         (__attribute__((address_space(3))) unsigned int*)dst,
                                         16, 0, 0);
      }
    }
  };

  f32x4g acc[4][4] = {};
  const int nkt = K / GM_BK;

  stage(0, 0);
  // drain the prologue stage and enter the loop (simple 2-phase: vmcnt(0)
  // + plain barrier per guide T3-minimum; the barrier itself re-waits)
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  for (int kt = 0; kt < nkt; ++kt) {
    const int cur = kt & 1, nxt = cur ^ 1;
    if (kt + 1 < nkt) stage(nxt, kt + 1);

    const char* a_lds = tile_ptr(cur, 0);
    const char* b_lds = tile_ptr(cur, 1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 af[4], bf[4];
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        int arow = wr + f * 16 + l15;
        int byte = (ks * 32 + lg * 8) * 2;
        af[f] = *reinterpret_cast<const bf16x8*>(a_lds + gm_swz(arow, byte));
        int brow = wc + f * 16 + l15;
        bf[f] = *reinterpret_cast<const bf16x8*>(b_lds + gm_swz(brow, byte));
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bf[j],
                                                              acc[i][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();
  }

  // ---- epilogue: C[i=row][j=col] 16x16 layout: col = l15, row = lg*4+r
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = bm + wr + i * 16 + lg * 4 + r;
        int col = bn + wc + j * 16 + l15;
        C[(long)row * N + col] = f2bf(acc[i][j][r]);
      }
    }
  }
}

extern "C" void launch_gemm_bf16_nt(void* C, const void* A, const void* B,
                                    int M, int N, int K, hipStream_t stream) {
  dim3 grid((M / GM_BM) * (N / GM_BN)), block(256);
  hipLaunchKernelGGL(gemm_bf16_nt_kernel, grid, block, 0, stream,
                     (unsigned short*)C, (const unsigned short*)A,
                     (const unsigned short*)B, M, N, K);
}


// ---------------------------------------------------------------------------
// Weight-streaming GEMV for decode (guide App. B "GEMV / M<=16 decode
// weights: operand streamed once, load straight to VGPRs, deep unroll"):
// C[M,N] = A[M,K] @ W[N,K]^T for small M.  One wave per output column:
// the wave streams W[n][:] coalesced (ushort8 per lane), A rows ride in
// L1/L2 (tiny), M dot products accumulate per lane, one wave reduction
// per (m, n).
// ---------------------------------------------------------------------------

#define GV_MMAX 8

template <int M>
__global__ __launch_bounds__(256) void gemv_bf16_nt_kernel(
    unsigned short* __restrict__ C,        // [M][N]
    const unsigned short* __restrict__ A,  // [M][K]
    const unsigned short* __restrict__ W,  // [N][K]
    int N, int K) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x >> 6;
  const int waves_total = (gridDim.x * blockDim.x) >> 6;
  for (int n = blockIdx.x * 4 + wid; n < N; n += waves_total) {
    const unsigned short* wrow = W + (long)n * K;
    float acc[M];
#pragma unroll
    for (int m = 0; m < M; ++m) acc[m] = 0.f;
    for (int k0 = lane * 8; k0 < K; k0 += WAVE * 8) {
      ushort8v wv = *reinterpret_cast<const ushort8v*>(wrow + k0);
#pragma unroll
      for (int m = 0; m < M; ++m) {
        ushort8v av = *reinterpret_cast<const ushort8v*>(A + (long)m * K + k0);
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[m] += bf2f(av[j]) * bf2f(wv[j]);
      }
    }
#pragma unroll
    for (int m = 0; m < M; ++m) {
      float r = wave_reduce_sum(acc[m]);
      if (lane == 0) C[(long)m * N + n] = f2bf(r);
    }
  }
}

// ---------------------------------------------------------------------------
// Decode GEMV epilogue variants (fewer, fused decode
// kernels).  Same weight-streaming structure as gemv_bf16_nt; the variants
// fold the surrounding elementwise work into the stream:
//   EPI 1 "norm entry":  out = s[m] * (x @ W^T) with s[m] computed from the
//         SAME streamed x rows (rsqrt(sumsq*mul + eps)) — rmsnorm folded
//         into the projection (per-channel gain folded into W on the host),
//         no stat buffer, no separate rmsnorm kernel.
//   EPI 2 "resid": out = x @ W^T + resid  (the residual-add fused away)
//   EPI 3 "swiglu norm": W rows interleaved (gate_i, up_i); each wave
//         streams BOTH rows of its output column: out = silu(s·g)·(s·u)
//         — gate_up intermediate and the silu_mul kernel both eliminated.
// ---------------------------------------------------------------------------

template <int M, int EPI>
__global__ __launch_bounds__(256) void gemv2_kernel(
    unsigned short* __restrict__ C,        // [M][N]
    const unsigned short* __restrict__ A,  // [M][K]
    const unsigned short* __restrict__ W,  // [N][K] (EPI 3: [2N][K])
    const unsigned short* __restrict__ resid,  // [M][N] (EPI 2)
    int N, int K, float stat_mul, float stat_eps) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x >> 6;
  const int waves_total = (gridDim.x * blockDim.x) >> 6;

  // EPI 1/3 row scales come from a sum-of-squares accumulated IN the dot
  // loop (x is streamed there anyway — a dedicated pre-pass measured
  // ~2.8 TB/s effective on the qkv projection); every wave computes the
  // same ss, so no cross-wave exchange is needed.
  float scale[M];
  bool have_scale = (EPI != 1 && EPI != 3);
#pragma unroll
  for (int m = 0; m < M; ++m) scale[m] = 1.0f;

  for (int n = blockIdx.x * 4 + wid; n < N; n += waves_total) {
    const unsigned short* wg = W + (long)(EPI == 3 ? 2 * n : n) * K;
    const unsigned short* wu = wg + K;  // EPI 3 only
    float accg[M], accu[M], ssq[M];
#pragma unroll
    for (int m = 0; m < M; ++m) accg[m] = accu[m] = ssq[m] = 0.f;
    const bool want_ss = !have_scale;
    for (int k0 = lane * 8; k0 < K; k0 += WAVE * 8) {
      ushort8v gv = *reinterpret_cast<const ushort8v*>(wg + k0);
      ushort8v uv;
      if (EPI == 3) uv = *reinterpret_cast<const ushort8v*>(wu + k0);
#pragma unroll
      for (int m = 0; m < M; ++m) {
        ushort8v av = *reinterpret_cast<const ushort8v*>(A + (long)m * K + k0);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float a = bf2f(av[j]);
          accg[m] += a * bf2f(gv[j]);
          if (EPI == 3) accu[m] += a * bf2f(uv[j]);
          if (want_ss) ssq[m] += a * a;
        }
      }
    }
    if (want_ss) {
#pragma unroll
      for (int m = 0; m < M; ++m)
        scale[m] = rsqrtf(wave_reduce_sum(ssq[m]) * stat_mul + stat_eps);
      have_scale = true;
    }
#pragma unroll
    for (int m = 0; m < M; ++m) {
      float g = wave_reduce_sum(accg[m]);
      float u = (EPI == 3) ? wave_reduce_sum(accu[m]) : 0.f;
      if (lane == 0) {
        float out;
        if (EPI == 3) {
          g *= scale[m];
          u *= scale[m];
          out = (g / (1.0f + __expf(-g))) * u;
        } else if (EPI == 2) {
          out = g + bf2f(resid[(long)m * N + n]);
        } else {
          out = g * scale[m];
        }
        C[(long)m * N + n] = f2bf(out);
      }
    }
  }
}

template <int EPI>
static void gemv2_dispatch(void* C, const void* A, const void* W,
                           const void* resid, int M, int N, int K,
                           float stat_mul, float stat_eps, hipStream_t stream) {
  int blocks = (N + 3) / 4;
  // Persistent-grid cap: dispatching thousands of 256-thread WGs costs
  // ~5-15 us of ramp on a kernel whose streaming work is itself ~10-40
  // us; fewer WGs walking more rows each amortize it (the n loop is
  // already persistent).  Overridable for probes via BOBRA_GEMV_CAP.
  int cap = 2048;
  if (const char* e = getenv("BOBRA_GEMV_CAP")) {
    int v = atoi(e);
    if (v > 0) cap = v;
  }
  if (blocks > cap) blocks = cap;
  dim3 grid(blocks), block(256);
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, block, 0, stream, (unsigned short*)C,
                       (const unsigned short*)A, (const unsigned short*)W,
                       (const unsigned short*)resid, N, K, stat_mul, stat_eps);
  };
  switch (M) {
    case 1: launch(gemv2_kernel<1, EPI>); break;
    case 2: launch(gemv2_kernel<2, EPI>); break;
    case 3: launch(gemv2_kernel<3, EPI>); break;
    case 4: launch(gemv2_kernel<4, EPI>); break;
    case 5: launch(gemv2_kernel<5, EPI>); break;
    case 6: launch(gemv2_kernel<6, EPI>); break;
    case 7: launch(gemv2_kernel<7, EPI>); break;
    default: launch(gemv2_kernel<8, EPI>); break;
  }
}

extern "C" void launch_gemv2(int epi, void* C, const void* A, const void* W,
                             const void* resid, int M, int N, int K,
                             float stat_mul, float stat_eps,
                             hipStream_t stream) {
  switch (epi) {
    case 1: gemv2_dispatch<1>(C, A, W, resid, M, N, K, stat_mul, stat_eps, stream); break;
    case 2: gemv2_dispatch<2>(C, A, W, resid, M, N, K, stat_mul, stat_eps, stream); break;
    case 3: gemv2_dispatch<3>(C, A, W, resid, M, N, K, stat_mul, stat_eps, stream); break;
    default: gemv2_dispatch<0>(C, A, W, resid, M, N, K, stat_mul, stat_eps, stream); break;
  }
}

extern "C" void launch_gemv_bf16_nt(void* C, const void* A, const void* W,
                                    int M, int N, int K, hipStream_t stream) {
  int blocks = (N + 3) / 4;
  if (blocks > 2048) blocks = 2048;
  dim3 grid(blocks), block(256);
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, block, 0, stream, (unsigned short*)C,
                       (const unsigned short*)A, (const unsigned short*)W, N, K);
  };
  switch (M) {
    case 1: launch(gemv_bf16_nt_kernel<1>); break;
    case 2: launch(gemv_bf16_nt_kernel<2>); break;
    case 3: launch(gemv_bf16_nt_kernel<3>); break;
    case 4: launch(gemv_bf16_nt_kernel<4>); break;
    case 5: launch(gemv_bf16_nt_kernel<5>); break;
    case 6: launch(gemv_bf16_nt_kernel<6>); break;
    case 7: launch(gemv_bf16_nt_kernel<7>); break;
    default: launch(gemv_bf16_nt_kernel<8>); break;
  }
}
