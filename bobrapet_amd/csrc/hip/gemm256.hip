// 256x256-tile bf16 MFMA GEMM for MI355X (gfx950) with fused epilogues.
//
// C[M,N] = A[M,K] @ B[N,K]^T   (NT: B row-major [N][K], nn.Linear layout)
//
// This is the llm-infer prefill hot-path GEMM: the
// structure follows the measured-fastest plain-HIP GEMM shape for CDNA4
// (guide §5 "256² 8-phase template" + "glds with >1 tile in flight"):
//   - 256x256 macro tile, 8 waves (2M x 4N), per-wave 128x64 output
//     (acc = 8x4 fragments of v_mfma_f32_16x16x32_bf16)
//   - K-slices of 32 staged global->LDS by __builtin_amdgcn_global_load_lds
//     (width 16) into a 4-slot ring (4 x 32 KB = 128 KB LDS), prefetched
//     3 slices ahead with COUNTED s_waitcnt vmcnt(8) — never drained to 0
//     in the main loop — and raw s_barrier (a __syncthreads would emit
//     vmcnt(0) while a glds is in flight and drain the pipeline)
//   - XOR swizzle on the per-lane glds SOURCE address + the matching XOR
//     on the ds_read_b128 side (glds writes lane-linear, so the swizzle
//     must live on the source: guide rule 21); the swizzle makes the
//     16-lane fragment-read groups bank-conflict-free
//   - s_setprio(1) around each MFMA cluster; XCD-aware bijective block
//     swizzle for L2 locality
//
// Fused epilogues (what hipBLASLt cannot do — the round-2 win condition):
//   EPI 0  plain store, optional row scale s[m] = rsqrt(stat_in[m]*mul+eps)
//          (the RMSNorm entry: the per-channel gain is folded into the
//          weights, the per-row scale commutes with the GEMM and is
//          applied here — no separate rmsnorm kernel, no normalized-x
//          round trip through HBM)
//   EPI 1  SwiGLU: B rows interleaved (gate_i, up_i) -> out[M][N/2] =
//          silu(g)*u, fused via one __shfl_xor(1); optional row scale
//   EPI 2  residual add: C = acc + resid (the new residual stream), plus
//          a per-row sum-of-squares atomically accumulated into stat_out
//          (feeds the NEXT projection's row scale == the next rmsnorm)
#include "common.h"

#define G2_BM 256
#define G2_BN 256
#define G2_BKS 32
#define G2_NSLOT 5                          // 5 x 32 KB = all 160 KB of LDS
#define G2_PART (G2_BM * G2_BKS * 2)        // one operand part: 16 KB
#define G2_SLOT (2 * G2_PART)               // A+B: 32 KB

typedef float f32x4g2 __attribute__((ext_vector_type(4)));

// Bank-spread XOR for the [row][64-byte k-row] LDS image.  A wave64
// ds_read_b128 is serviced in four 16-lane groups that MIX row (l15) and
// k-group (lg) lanes — e.g. {l15 0-3 & 12-15 at lg=0} ∪ {l15 4-11 at
// lg=1} (microarch §LDS).  With bank = (16·row + 4·(lg ^ f(row))) mod 64,
// the per-group distinctness condition reduces to
// {f(c), f(12+c), 1^f(4+c), 1^f(8+c)} all distinct for every c∈0..3,
// which f(row) = 3·((row>>3)&1) satisfies for all four groups — measured:
// the previous (row>>2)&3 variant left a 2-way conflict in every group
// (SQ_LDS_BANK_CONFLICT ≈ 10% of wave cycles).
__device__ __forceinline__ int g2_swz(int row) { return ((row >> 3) & 1) * 48; }

__device__ __forceinline__ void g2_glds(const unsigned short* src, char* lds_dst) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)src,
      (__attribute__((address_space(3))) unsigned int*)lds_dst, 16, 0, 0);
}

__device__ __forceinline__ float g2_silu(float g) {
  return g / (1.0f + __expf(-g));
}

template <int EPI>
__global__ __launch_bounds__(512, 2) void gemm256_kernel(
    unsigned short* __restrict__ C,
    const unsigned short* __restrict__ A,
    const unsigned short* __restrict__ B,
    const unsigned short* __restrict__ resid,  // EPI 2
    const float* __restrict__ stat_in,         // EPI 0/1 optional row stat
    float* __restrict__ stat_out,              // EPI 2 optional
    int M, int N, int K, float stat_mul, float stat_eps) {
  __shared__ __attribute__((aligned(16))) char smem[G2_NSLOT * G2_SLOT];

  const int nbn = N / G2_BN;
  const int nbm = (M + G2_BM - 1) / G2_BM;
  unsigned int wgid = xcd_swizzle(blockIdx.x, nbm * nbn);
  // grouped supertile order: bn varies fastest within GN-wide column
  // groups, bm within a group next.  An XCD's contiguous chunk then
  // covers a compact (many-bm × GN-bn) window: the A panel (2 MB) stays
  // L2-resident across each 16-tile bn run and the GN B panels stay
  // LLC-resident across bm — measured FETCH dropped ~4x vs the flat
  // bn-fastest order on the gate/up shape.
  const int GN = 16;
  int bm_i, bn_i;
  {
    const int full = nbn / GN;             // full-width column groups
    const int per = nbm * GN;
    if ((int)wgid < full * per) {
      const int grp = (int)wgid / per, rem = (int)wgid % per;
      bm_i = rem / GN;
      bn_i = grp * GN + rem % GN;
    } else {
      const int tail_n = nbn - full * GN;  // last, narrower group
      const int rem = (int)wgid - full * per;
      bm_i = rem / tail_n;
      bn_i = full * GN + rem % tail_n;
    }
  }
  const int bm = bm_i * G2_BM;
  const int bn = bn_i * G2_BN;

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;           // 8 waves: 2 (M) x 4 (N)
  const int wr = wid >> 2;            // 0..1  -> rows wr*128
  const int wc = wid & 3;             // 0..3  -> cols wc*64
  const int l15 = lane & 15;
  const int lg = lane >> 4;           // k-group 0..3 (8 bf16 each)

  // per-wave glds piece assignment: 16 x 1 KB pieces per operand part,
  // wave w stages pieces {2w, 2w+1}; chunk = piece*64 + lane (16 B each);
  // LDS image row = chunk/4 (64-B k-rows), source k-byte inverse-swizzled.
  // Source pointers are computed ONCE (incl. the M-tail row clamp) and
  // only advance by kt*BKS columns per slice — keeps the hot loop's
  // register pressure down to one scalar offset per stage.
  // piece it=1 is piece it=0 shifted by 16 rows; the swizzle term is
  // invariant under row+16 ((row>>3)&1 unchanged), so its source is the
  // it=0 pointer + a UNIFORM 16*K elements — one pointer pair to carry.
  const unsigned short* asrc0;
  const unsigned short* bsrc0;
  long arow1_fix = 16L * K;  // it=1 offset; adjusted below if clamped
  {
    const int chunk = (wid * 2) * 64 + lane;
    const int row = chunk >> 2;
    const int kbx = ((chunk & 3) * 16) ^ g2_swz(row);
    int arow = bm + row;
    if (arow >= M) arow = M - 1;  // M-tail: clamp (stores are predicated)
    int arow1 = bm + row + 16;
    if (arow1 >= M) arow1 = M - 1;
    arow1_fix = (long)(arow1 - arow) * K;
    asrc0 = A + (long)arow * K + kbx / 2;
    bsrc0 = B + (long)(bn + row) * K + kbx / 2;
  }
  auto stage = [&](int slot, int kt) {
    const int k0 = kt * G2_BKS;
    char* abase = smem + slot * G2_SLOT + (wid * 2) * 1024;
    char* bbase = abase + G2_PART;
    g2_glds(asrc0 + k0, abase);
    g2_glds(bsrc0 + k0, bbase);
    g2_glds(asrc0 + arow1_fix + k0, abase + 1024);
    g2_glds(bsrc0 + 16L * K + k0, bbase + 1024);
  };

  f32x4g2 acc[8][4] = {};
  const int nk = K / G2_BKS;

  // fragment-read addresses are loop-invariant except the slot base
  const int a_off = (wr * 128 + l15) * 64 + ((lg * 16) ^ g2_swz(wr * 128 + l15));
  const int b_off = (wc * 64 + l15) * 64 + ((lg * 16) ^ g2_swz(wc * 64 + l15));
  // i*16 rows stride: 16*64 bytes; the swizzle depends on row>>2 which
  // changes every 4 rows — but row = base + i*16 keeps (row>>2)&3
  // invariant, so the XOR term is the same for every fragment index.
  auto rd_a = [&](const char* ab, int i) {
    return *reinterpret_cast<const bf16x8*>(ab + a_off + i * 16 * 64);
  };
  auto rd_b = [&](const char* bb, int j) {
    return *reinterpret_cast<const bf16x8*>(bb + b_off + j * 16 * 64);
  };

  // software-pipelined schedule (template-style read/barrier pairing):
  // each phase issues the CURRENT slice's remaining fragment reads (B +
  // A-tail, slot s) and the glds prefetch BEFORE the barrier, so their
  // latency overlaps the barrier wait; after the barrier the MFMA
  // cluster starts on fragments that are already (or nearly) resident,
  // with only the next slice's 4 leading A reads interleaved into the
  // cluster.  glds runs 4 slices ahead through the 5-slot ring with
  // counted vmcnt — 2 slices always in flight across the barrier.
  bf16x8 afp1[2], afp2[2];  // leading A frags of s (alternating sets)
  bf16x8 aft[6];            // trailing A frags of s (read + used in-phase)
  bf16x8 bf[4];

  stage(0, 0);
  if (nk > 1) stage(1, 1);
  if (nk > 2) stage(2, 2);
  if (nk > 3) stage(3, 3);
  // slices 0 (all reads at phase-0) and 1 (interleaved A prefetch) must
  // land before phase 0's cluster; slices 2,3 stay in flight
  if (nk >= 4)
    asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
  else if (nk == 3)
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  else
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
  if constexpr (EPI != 0) {
#pragma unroll
    for (int i = 0; i < 2; ++i) afp1[i] = rd_a(smem, i);
  }

  int slot_cur = 0;            // slot of slice s
  int slot_pf = 4 % G2_NSLOT;  // slot for slice s+4

  // A-set selection must be STATIC (a runtime-selected pointer to a
  // register array spills to scratch — guide rule 20): 2x hand-unroll.
  auto phase = [&](int s, bf16x8(&afp_cur)[2], bf16x8(&afp_next)[2]) {
    const char* ab = smem + slot_cur * G2_SLOT;
    const char* bb = ab + G2_PART;
    const char* abn = smem + (slot_cur + 1 == G2_NSLOT ? 0 : slot_cur + 1) * G2_SLOT;
    if (++slot_cur == G2_NSLOT) slot_cur = 0;
    // pre-barrier: this slice's B + trailing-A reads and the prefetch
    // glds — their latency hides under the barrier wait.  (EPI 0's
    // allocator mishandles the cross-phase afp live ranges — fragments
    // round-trip through scratch — so it reads ALL its A fragments at
    // the head instead of software-pipelining the leading pair.)
#pragma unroll
    for (int j = 0; j < 4; ++j) bf[j] = rd_b(bb, j);
#pragma unroll
    for (int i = 0; i < 6; ++i) aft[i] = rd_a(ab, 2 + i);
    if constexpr (EPI == 0) {
#pragma unroll
      for (int i = 0; i < 2; ++i) afp_cur[i] = rd_a(ab, i);
    }
    if (s + 4 < nk) {
      stage(slot_pf, s + 4);
      if (++slot_pf == G2_NSLOT) slot_pf = 0;
    }
    // slice s+1 must be landed for the interleaved A reads below
    if (s + 3 < nk)
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    else if (s + 2 < nk)
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");
    const bool pf = s + 1 < nk;
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 2; ++i) {
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afp_cur[i], bf[j], acc[i][j], 0, 0, 0);
      if (EPI != 0 && pf) afp_next[i] = rd_a(abn, i);
    }
#pragma unroll
    for (int i = 0; i < 6; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[2 + i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            aft[i], bf[j], acc[2 + i][j], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
  };

  if constexpr (EPI == 0) {
    // no cross-phase register state (afp read at the head): plain loop
    for (int s = 0; s < nk; ++s) phase(s, afp1, afp2);
  } else {
    for (int s = 0; s < nk;) {
      phase(s, afp1, afp2);
      if (++s >= nk) break;
      phase(s, afp2, afp1);
      ++s;
    }
  }

  // keep the epilogue's global loads AND address chains BELOW the main
  // loop: hoisted stat loads / 32 precomputed 64-bit C addresses extend
  // live ranges across the whole K loop and spill the accumulators (an
  // asm memory clobber pins the loads; the "+v" below pins the
  // register-only address arithmetic, which a clobber cannot order)
  asm volatile("" ::: "memory");
  int row0 = bm + wr * 128 + lg * 4;
  int col0 = bn + wc * 64 + l15;
  asm volatile("" : "+v"(row0), "+v"(col0));

  // ------------------------------------------------------------------
  // epilogue.  acc[i][j][r] -> row = bm + wr*128 + i*16 + lg*4 + r,
  //                            col = bn + wc*64 + j*16 + l15
  // Loads (row stats / residual) are issued BATCHED ahead of their uses
  // with clamped addresses — a load inside a predicated per-element
  // branch would serialize 32 dependent global round trips (guide §5
  // ".s-level traps" (c)).

  float sc[8][4];
  if (EPI != 2 && stat_in != nullptr) {
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = row0 + i * 16 + r;
        sc[i][r] = stat_in[row < M ? row : M - 1];
      }
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        sc[i][r] = rsqrtf(sc[i][r] * stat_mul + stat_eps);
  } else {
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int r = 0; r < 4; ++r) sc[i][r] = 1.0f;
  }

#pragma unroll
  for (int i = 0; i < 8; ++i) {
    float rsv[4][4];
    if (EPI == 2) {
      // batch the 16 residual loads of this i-block, clamped rows
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = row0 + i * 16 + r;
        const long rb = (long)(row < M ? row : M - 1) * N;
#pragma unroll
        for (int j = 0; j < 4; ++j) rsv[r][j] = bf2f(resid[rb + col0 + j * 16]);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = row0 + i * 16 + r;
      const bool live = row < M;
      if (EPI == 0) {
        float outs[4];
#pragma unroll
        for (int j = 0; j < 4; ++j) outs[j] = acc[i][j][r] * sc[i][r];
        if (live) {
#pragma unroll
          for (int j = 0; j < 4; ++j)
            C[(long)row * N + col0 + j * 16] = f2bf(outs[j]);
        }
      } else if (EPI == 1) {
        // interleaved (gate, up) columns: even fused col = gate_i,
        // odd = up_i; one half-swap via shfl_xor(1) pairs them.
        float outs[4];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const float v = acc[i][j][r] * sc[i][r];
          const float partner = __shfl_xor(v, 1, WAVE);
          outs[j] = g2_silu(v) * partner;
        }
        if (live && (l15 & 1) == 0) {
#pragma unroll
          for (int j = 0; j < 4; ++j)
            C[(long)row * (N / 2) + ((col0 + j * 16) >> 1)] = f2bf(outs[j]);
        }
      } else {  // EPI == 2: residual add + row sumsq
        float ss = 0.0f;
        float outs[4];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          outs[j] = acc[i][j][r] + rsv[r][j];
          ss += outs[j] * outs[j];
        }
        if (live) {
#pragma unroll
          for (int j = 0; j < 4; ++j)
            C[(long)row * N + col0 + j * 16] = f2bf(outs[j]);
        }
        if (stat_out != nullptr) {
          // reduce the 16 lanes of this row's lane group
#pragma unroll
          for (int off = 8; off > 0; off >>= 1) ss += __shfl_xor(ss, off, WAVE);
          if (l15 == 0 && live) atomicAdd(stat_out + row, ss);
        }
      }
    }
  }
}

extern "C" void launch_gemm256(int epi, void* C, const void* A, const void* B,
                               const void* resid, const void* stat_in,
                               void* stat_out, int M, int N, int K,
                               float stat_mul, float stat_eps,
                               hipStream_t stream) {
  const int nbm = (M + G2_BM - 1) / G2_BM;
  dim3 grid(nbm * (N / G2_BN)), block(512);
  if (stat_out != nullptr)
    hipMemsetAsync(stat_out, 0, (size_t)M * sizeof(float), stream);
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, block, 0, stream, (unsigned short*)C,
                       (const unsigned short*)A, (const unsigned short*)B,
                       (const unsigned short*)resid, (const float*)stat_in,
                       (float*)stat_out, M, N, K, stat_mul, stat_eps);
  };
#ifdef G2_SOLO_EPI0
  launch(gemm256_kernel<0>);
  (void)epi;
#else
  switch (epi) {
    case 0: launch(gemm256_kernel<0>); break;
    case 1: launch(gemm256_kernel<1>); break;
    default: launch(gemm256_kernel<2>); break;
  }
#endif
}

// ---------------------------------------------------------------------------
// row sum-of-squares: stat[m] = sum_k x[m,k]^2  (f32) — the norm entry for
// rows produced OUTSIDE a fused-epilogue GEMM (embedding output).  One wave
// per row, vectorized ushort8 loads.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void rowsumsq_kernel(
    float* __restrict__ stat, const unsigned short* __restrict__ x, int M,
    int K) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x >> 6;
  const int row = blockIdx.x * 4 + wid;
  if (row >= M) return;
  const unsigned short* xr = x + (long)row * K;
  float ss = 0.0f;
  for (int k0 = lane * 8; k0 < K; k0 += WAVE * 8) {
    ushort8v v = *reinterpret_cast<const ushort8v*>(xr + k0);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float f = bf2f(v[j]);
      ss += f * f;
    }
  }
  ss = wave_reduce_sum(ss);
  if (lane == 0) stat[row] = ss;
}

extern "C" void launch_rowsumsq(void* stat, const void* x, int M, int K,
                                hipStream_t stream) {
  dim3 grid((M + 3) / 4), block(256);
  hipLaunchKernelGGL(rowsumsq_kernel, grid, block, 0, stream, (float*)stat,
                     (const unsigned short*)x, M, K);
}
