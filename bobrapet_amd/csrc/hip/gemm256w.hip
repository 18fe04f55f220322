// 256x256-tile bf16 GEMM on v_mfma_f32_32x32x16_bf16 (gfx950).
//
// Same tile geometry, staging ring, counted-vmcnt schedule and epilogues
// as gemm256b.hip, but the MFMA shape is 32x32x16 instead of 16x16x32:
//  - HALF the MFMA instruction count for the same FLOPs (32/K-tile/wave
//    vs 64) — attacks the measured WAIT_INST_ANY ~49% issue bound
//  - higher shape ceiling: 2382 TF bf16 vs 2075 for 16x16 (guide tables)
// Fragment maps (verified on hardware by attention.hip's numerics tests):
//   A[i][k]: i = lane&31, k = 8*(lane>>5) + j   (j = 0..7, one bf16x8)
//   B[k][j]: j = lane&31, k = 8*(lane>>5) + jj  (B image is [N][K] = NT)
//   C[i][j]: j = lane&31, i = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
// Wave tile 128x64 = 4 (M) x 2 (N) C-blocks of 32x32, acc[4][2] f32x16.
// Phase q computes M-block q (2 nj x 4 k16 = 8 MFMAs, two interleaved
// accumulator chains: dependent latency 64 cyc / issue 32 -> 2 chains
// saturate the pipe).
//
// LDS swizzle: the 32-row b128 read groups (row = lane&31) collide 2-4
// way under gemm256b's 16-row swizzle; XOR the 16-B chunk with
// ((row>>2)&3) instead — within each gfx950 mixed lane group the rows
// sharing row%4 are {r, r+12, r+20, r+24} (or {r+4,r+8,r+16,r+28}),
// whose (row>>2)&3 values are all distinct, so every 64-B bank quarter
// serves exactly one 16-B chunk per group.  Derivation mirrors
// gemm256.hip's; staging applies the same XOR on the glds SOURCE.
// Requires K % 64 == 0 (dispatcher falls back otherwise).
#include "common.h"

#define G4_NRING 10
#define G4_HALF 16384

typedef float f32x16g4 __attribute__((ext_vector_type(16)));

__device__ __forceinline__ int g4_swz(int row) { return ((row >> 2) & 3) * 16; }

__device__ __forceinline__ void g4_glds(const unsigned short* src, char* lds_dst) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)src,
      (__attribute__((address_space(3))) unsigned int*)lds_dst, 16, 0, 0);
}

__device__ __forceinline__ float g4_silu(float g) {
  return g / (1.0f + __expf(-g));
}

template <int EPI>
__global__ __launch_bounds__(512, 2) void gemm256w_kernel(
    unsigned short* __restrict__ C,
    const unsigned short* __restrict__ A,
    const unsigned short* __restrict__ B,
    const unsigned short* __restrict__ resid,  // EPI 2
    const float* __restrict__ stat_in,         // EPI 0/1 optional row stat
    float* __restrict__ stat_out,              // EPI 2 optional
    int M, int N, int K, float stat_mul, float stat_eps) {
  __shared__ __attribute__((aligned(16))) char smem[G4_NRING * G4_HALF];
  // ring bases: A0 slots 0..2, A1 3..5, B0 6..7, B1 8..9 (as gemm256b)

  const int nbn = N / 256;
  const int nbm = (M + 255) / 256;
  unsigned int wgid = xcd_swizzle(blockIdx.x, nbm * nbn);
  const int GN = 16;
  int bm_i, bn_i;
  {
    const int full = nbn / GN;
    const int per = nbm * GN;
    if ((int)wgid < full * per) {
      const int grp = (int)wgid / per, rem = (int)wgid % per;
      bm_i = rem / GN;
      bn_i = grp * GN + rem % GN;
    } else {
      const int tail_n = nbn - full * GN;
      const int rem = (int)wgid - full * per;
      bm_i = rem / tail_n;
      bn_i = full * GN + rem % tail_n;
    }
  }
  const int bm = bm_i * 256;
  const int bn = bn_i * 256;

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;  // 8 waves: 2 (M) x 4 (N)
  const int wr = wid >> 2;
  const int wc = wid & 3;
  const int l31 = lane & 31;
  const int hi = lane >> 5;

  // ---- staging (identical structure to gemm256b; swizzle is g4) ----
  const int p0 = wid * 2;
  const int st_ks = p0 >> 3;
  const int st_row = (p0 & 7) * 16 + (lane >> 2);
  const int st_kbx = ((lane & 3) * 16) ^ g4_swz(st_row);
  int ar0 = bm + st_row, ar1 = bm + 128 + st_row;
  if (ar0 >= M) ar0 = M - 1;
  if (ar1 >= M) ar1 = M - 1;
  int ar0b = bm + st_row + 16, ar1b = bm + 128 + st_row + 16;
  if (ar0b >= M) ar0b = M - 1;
  if (ar1b >= M) ar1b = M - 1;
  // the second piece sits 16 rows below; g4_swz(row+16) == g4_swz(row)
  const unsigned short* a0src = A + (long)ar0 * K + st_ks * 32 + st_kbx / 2;
  const unsigned short* a1src = A + (long)ar1 * K + st_ks * 32 + st_kbx / 2;
  const unsigned short* b0src =
      B + (long)(bn + st_row) * K + st_ks * 32 + st_kbx / 2;
  const int a0d = (ar0b - ar0) * K;
  const int a1d = (ar1b - ar1) * K;
  const int b0d = 16 * K;

  auto stage = [&](int half, int slot, int kt) {
    char* base = smem + slot * G4_HALF + p0 * 1024;
    const int k0 = kt * 64;
    const unsigned short* s0;
    int d;
    if (half == 0) {
      s0 = a0src + k0;
      d = a0d;
    } else if (half == 1) {
      s0 = a1src + k0;
      d = a1d;
    } else if (half == 2) {
      s0 = b0src + k0;
      d = b0d;
    } else {
      s0 = b0src + 128L * K + k0;
      d = b0d;
    }
    g4_glds(s0, base);
    g4_glds(s0 + d, base + 1024);
  };

  f32x16g4 acc[4][2] = {};
  const int nt = K / 64;

  // fragment reads: k16-step s -> ks-block s>>1, 32-B half s&1, 16-B
  // sub-chunk hi; row = block row + l31; chunk XOR g4_swz(row)
  const int a_row = l31;  // + mi*32 per phase
  auto rd_a = [&](const char* slotb, int mi, int s) {
    const int row = mi * 32 + a_row;
    return *reinterpret_cast<const bf16x8*>(
        slotb + (s >> 1) * 8192 + row * 64 +
        (((s & 1) * 32 + hi * 16) ^ g4_swz(row)));
  };
  const int b_row0 = (wc & 1) * 64 + l31;
  auto rd_b = [&](const char* slotb, int nj, int s) {
    const int row = b_row0 + nj * 32;
    return *reinterpret_cast<const bf16x8*>(
        slotb + (s >> 1) * 8192 + row * 64 +
        (((s & 1) * 32 + hi * 16) ^ g4_swz(row)));
  };
  char* aring = smem + (wr ? 3 : 0) * G4_HALF;
  char* bring = smem + ((wc >> 1) ? 8 : 6) * G4_HALF;

  bf16x8 bfr[2][4];  // B frags of the current tile (nj x s)
  bf16x8 aq[4];      // current M-block's A frags (s)

  // ---- prologue: 7 halves in flight (same order/waits as gemm256b)
  stage(0, 0, 0);
  stage(1, 3, 0);
  stage(2, 6, 0);
  stage(3, 8, 0);
  if (nt > 1) {
    stage(0, 1, 1);
    stage(1, 4, 1);
    stage(2, 7, 1);
  }
  if (nt > 1)
    asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
  else
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");

  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  // phase q computes M-block q: 8 MFMAs as two interleaved 4-deep
  // accumulator chains (nj inner)
#define G4_MFMA_BLOCK(mi)                                               \
    __builtin_amdgcn_s_setprio(1);                                      \
    _Pragma("unroll")                                                   \
    for (int s = 0; s < 4; ++s)                                         \
      _Pragma("unroll")                                                 \
      for (int nj = 0; nj < 2; ++nj)                                    \
        acc[mi][nj] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(          \
            aq[s], bfr[nj][s], acc[mi][nj], 0, 0, 0);                   \
    __builtin_amdgcn_s_setprio(0);

  for (int t = 0; t < nt; ++t) {
    const char* aslot = aring + (t % 3) * G4_HALF;
    const char* bslot = bring + (t & 1) * G4_HALF;
    const bool tail = t + 2 >= nt;

    // phase 0: B(t) + A block 0; stage B1(t+1); barrier propagates the
    // q3 vmcnt (tile t's halves landed for every wave)
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");
#pragma unroll
    for (int nj = 0; nj < 2; ++nj)
#pragma unroll
      for (int s = 0; s < 4; ++s) bfr[nj][s] = rd_b(bslot, nj, s);
#pragma unroll
    for (int s = 0; s < 4; ++s) aq[s] = rd_a(aslot, 0, s);
    if (t + 1 < nt) stage(3, 8 + ((t + 1) & 1), t + 1);
    G4_MFMA_BLOCK(0)

    // phase 1 (no barrier): A block 1; stage A0(t+2)
#pragma unroll
    for (int s = 0; s < 4; ++s) aq[s] = rd_a(aslot, 1, s);
    if (t + 2 < nt) stage(0, (t + 2) % 3, t + 2);
    G4_MFMA_BLOCK(1)

    // phase 2: A block 2; stage A1(t+2)
#pragma unroll
    for (int s = 0; s < 4; ++s) aq[s] = rd_a(aslot, 2, s);
    if (t + 2 < nt) stage(1, 3 + (t + 2) % 3, t + 2);
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");
    G4_MFMA_BLOCK(2)

    // phase 3 (no barrier): A block 3; stage B0(t+2); tile-boundary vmcnt
#pragma unroll
    for (int s = 0; s < 4; ++s) aq[s] = rd_a(aslot, 3, s);
    if (t + 2 < nt) stage(2, 6 + ((t + 2) & 1), t + 2);
    if (!tail)
      asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    G4_MFMA_BLOCK(3)
  }
#undef G4_MFMA_BLOCK

  // keep the epilogue's loads and address chains below the loop
  asm volatile("" ::: "memory");
  int row0 = bm + wr * 128 + 4 * hi;
  int col0 = bn + wc * 64 + l31;
  asm volatile("" : "+v"(row0), "+v"(col0));

  // ------------------------------------------------------------------
  // epilogue.  C map: acc[mi][nj][r] -> row = row0 + mi*32 + (r&3) +
  // 8*(r>>2), col = col0 + nj*32.
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
    float sc[16];
    if (EPI != 2 && stat_in != nullptr) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int row = row0 + mi * 32 + (r & 3) + 8 * (r >> 2);
        sc[r] = stat_in[row < M ? row : M - 1];
      }
#pragma unroll
      for (int r = 0; r < 16; ++r)
        sc[r] = rsqrtf(sc[r] * stat_mul + stat_eps);
    } else {
#pragma unroll
      for (int r = 0; r < 16; ++r) sc[r] = 1.0f;
    }
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = row0 + mi * 32 + (r & 3) + 8 * (r >> 2);
      const bool live = row < M;
      if (EPI == 0) {
#pragma unroll
        for (int nj = 0; nj < 2; ++nj) {
          float outv = acc[mi][nj][r] * sc[r];
          if (live) C[(long)row * N + col0 + nj * 32] = f2bf(outv);
        }
      } else if (EPI == 1) {
#pragma unroll
        for (int nj = 0; nj < 2; ++nj) {
          const float v = acc[mi][nj][r] * sc[r];
          const float partner = __shfl_xor(v, 1, WAVE);
          const float outv = g4_silu(v) * partner;
          if (live && (l31 & 1) == 0)
            C[(long)row * (N / 2) + ((col0 + nj * 32) >> 1)] = f2bf(outv);
        }
      } else {
        const long rb = (long)(live ? row : M - 1) * N;
        float ss = 0.0f;
#pragma unroll
        for (int nj = 0; nj < 2; ++nj) {
          float outv = acc[mi][nj][r] + bf2f(resid[rb + col0 + nj * 32]);
          ss += outv * outv;
          if (live) C[(long)row * N + col0 + nj * 32] = f2bf(outv);
        }
        if (stat_out != nullptr) {
          // reduce over the 32 lanes of this hi-half (same row)
#pragma unroll
          for (int off = 16; off > 0; off >>= 1)
            ss += __shfl_xor(ss, off, WAVE);
          if (l31 == 0 && live) atomicAdd(stat_out + row, ss);
        }
      }
    }
  }
}

extern "C" void launch_gemm256w(int epi, void* C, const void* A, const void* B,
                                const void* resid, const void* stat_in,
                                void* stat_out, int M, int N, int K,
                                float stat_mul, float stat_eps,
                                hipStream_t stream) {
  const int nbm = (M + 255) / 256;
  dim3 grid(nbm * (N / 256)), block(512);
  if (stat_out != nullptr)
    (void)hipMemsetAsync(stat_out, 0, (size_t)M * sizeof(float), stream);
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, block, 0, stream, (unsigned short*)C,
                       (const unsigned short*)A, (const unsigned short*)B,
                       (const unsigned short*)resid, (const float*)stat_in,
                       (float*)stat_out, M, N, K, stat_mul, stat_eps);
  };
  switch (epi) {
    case 0: launch(gemm256w_kernel<0>); break;
    case 1: launch(gemm256w_kernel<1>); break;
    default: launch(gemm256w_kernel<2>); break;
  }
}
