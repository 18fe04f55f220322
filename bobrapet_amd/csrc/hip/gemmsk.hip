// Skinny-M bf16 MFMA GEMM (gfx950) — the decode-batch 9..32 projection
// path (docs/NEXT.md #10).  C[M<=32][N] = A[M][K] @ B[N][K]^T.
//
// Why a separate kernel: at M<=32 the 256-row gemm256 tile wastes 8-32x
// compute AND starves the grid (qkv N=6144 -> 24 WGs), while the
// weight-streaming GEMV's per-row uniform pointers spill SGPRs beyond
// M=8.  Here the parallelism comes from (N/64 tiles) x (K split):
//   - each WG owns all M rows, a 64-col N-tile and a K-chunk (<=1024)
//   - A (32 x Kc bf16, <=64 KB) is staged once into LDS, double-buffered
//     in 512-wide k-blocks; B streams straight from HBM (each weight row
//     is read exactly once across the grid)
//   - 4 waves x (2 M-frags x 1 N-frag) of v_mfma_f32_16x16x32_bf16;
//     acc is 2 f32x4 per lane — no register pressure at any M
//   - KS == 1 (grid already fills the chip, e.g. gate/up N=28672):
//     the WG walks the WHOLE K and writes bf16 directly with the fused
//     epilogue (rowscale / SwiGLU pairing via shfl_xor(1))
//   - KS > 1: WGs write f32 partials to ws[KS][M][Ntile]; a small
//     reduce kernel sums the K-splits and applies the epilogue
//     (rowscale / residual-add + row-sumsq for the stat chain) —
//     deterministic (no atomics on the accumulation path)
//
// LDS A image: ks-blocks of [32 rows][32 k] (64-B rows), byte chunk
// XOR'd with T(row>>2), T = {0,2,3,1} (nibble table 0x1320).  Derivation:
// an A-frag ds_read_b128 group mixes rows r in 0..15 at chunk lg and
// rows at chunk lg^1; bank slot = (row&3, chunk^T(row>>2)); the table
// makes every slot distinct within each gfx950 mixed lane group (the
// plain ((row>>2)&3) XOR collides rows 0-3@lg0 with rows 4-7@lg1).
#include "common.h"

#define SK_M 32      // compile-time row bucket (runtime Mr <= 32)
#define SK_NT 64     // N-tile per WG
#define SK_KB 512    // k-block staged per double-buffer half
#define SK_KC_MAX 1024

typedef float f32x4sk __attribute__((ext_vector_type(4)));

__device__ __forceinline__ int sk_swz(int row) {
  return ((0x1320 >> (((row) >> 2) & 3) * 4) & 3) * 16;
}

__device__ __forceinline__ float sk_silu(float g) {
  return g / (1.0f + __expf(-g));
}

// EPI (direct KS==1 path): 0 = rowscale (stat_in), 1 = SwiGLU+rowscale
template <int EPI>
__global__ __launch_bounds__(256, 2) void gemmsk_kernel(
    unsigned short* __restrict__ C,   // bf16 out (KS==1) — [Mr][N] or [Mr][N/2]
    float* __restrict__ ws,           // f32 partials (KS>1): [KS][SK_M][N]
    const unsigned short* __restrict__ A,  // [Mr][K]
    const unsigned short* __restrict__ B,  // [N][K]
    const float* __restrict__ stat_in,     // row sumsq (optional)
    int Mr, int N, int K, int KS, float stat_mul, float stat_eps) {
  // grid.x = N/SK_NT, grid.y = KS
  __shared__ __attribute__((aligned(16))) unsigned short a_lds[2][SK_M * SK_KB];

  const int nt = blockIdx.x;
  const int ks = blockIdx.y;
  const int n0 = nt * SK_NT;
  const int kc = K / KS;           // K-chunk of this WG (K % (KS*32) == 0)
  const int k0 = ks * kc;

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;        // 4 waves: wave w owns N cols n0+w*16..+15
  const int l15 = lane & 15;
  const int lg = lane >> 4;

  // cooperative A stage: a k-block image is 32 rows x 512 k x 2 B =
  // 32 KB = 2048 16-B chunks; 256 threads x 8 iterations.  Chunk ci maps
  // to (ks-block b = ci>>7, row r, 16-B slot c) with the swizzle on the
  // DEST slot; rows >= Mr duplicate row Mr-1 (never read back as output).
  auto stage_full = [&](int buf, int kb) {
#pragma unroll
    for (int it = 0; it < 8; ++it) {
      const int ci = it * 256 + tid;
      const int b = ci >> 7;
      const int r = (ci >> 2) & 31;
      const int c = ci & 3;
      const int src_row = r < Mr ? r : Mr - 1;
      const int kk = k0 + kb * SK_KB + b * 32 + c * 8;
      const bf16x8 v =
          *reinterpret_cast<const bf16x8*>(A + (long)src_row * K + kk);
      unsigned short* dst =
          a_lds[buf] + b * (32 * 32) + r * 32 + ((c * 16) ^ sk_swz(r)) / 2;
      *reinterpret_cast<bf16x8*>(dst) = v;
    }
  };

  f32x4sk acc[2] = {};  // M-frags 0 (rows 0-15) and 1 (rows 16-31)

  const int nkb = kc / SK_KB;          // full 512-k blocks
  const int krem = kc - nkb * SK_KB;   // remainder (multiple of 32)

  // B row base for this wave's N-frag: rows n0 + wid*16 + l15, starting
  // at this WG's K-chunk
  const unsigned short* brow = B + (long)(n0 + wid * 16 + l15) * K + k0;

  int bofs = 0;  // k progress of the B stream (elements)
  auto mfma_block = [&](int buf, int nks) {
    // nks ks-blocks of 32 k each; unroll so 4 B loads stay in flight —
    // indexed addressing (brow + b*32), NOT a serial pointer bump, so
    // the unrolled iterations' load addresses are independent
#pragma unroll 4
    for (int b = 0; b < nks; ++b) {
      const unsigned short* ab = a_lds[buf] + b * (32 * 32);
      // A frags: rows l15 (+16), k = lg*8..+7 within the ks-block
      const bf16x8 a0 = *reinterpret_cast<const bf16x8*>(
          ab + l15 * 32 + (((lg * 16) ^ sk_swz(l15)) >> 1));
      const bf16x8 a1 = *reinterpret_cast<const bf16x8*>(
          ab + (16 + l15) * 32 + (((lg * 16) ^ sk_swz(16 + l15)) >> 1));
      // B frag straight from HBM (each weight row read once)
      const bf16x8 bf =
          *reinterpret_cast<const bf16x8*>(brow + bofs + b * 32 + lg * 8);
      acc[0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, bf, acc[0], 0, 0, 0);
      acc[1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, bf, acc[1], 0, 0, 0);
    }
    bofs += nks * 32;
  };

  if (nkb > 0) stage_full(0, 0);
  __syncthreads();
  for (int kb = 0; kb < nkb; ++kb) {
    const int buf = kb & 1;
    if (kb + 1 < nkb) {
      stage_full(buf ^ 1, kb + 1);
    } else if (krem > 0) {
      // stage the remainder into the other buffer (partial image)
      const int base_kk = k0 + nkb * SK_KB;
      for (int ci = tid; ci < (krem / 32) * 128; ci += 256) {
        const int b = ci >> 7;
        const int r = (ci >> 2) & 31;
        const int c = ci & 3;
        const int src_row = r < Mr ? r : Mr - 1;
        const bf16x8 v = *reinterpret_cast<const bf16x8*>(
            A + (long)src_row * K + base_kk + b * 32 + c * 8);
        unsigned short* dst = a_lds[buf ^ 1] + b * (32 * 32) + r * 32 +
                              ((c * 16) ^ sk_swz(r)) / 2;
        *reinterpret_cast<bf16x8*>(dst) = v;
      }
    }
    mfma_block(buf, SK_KB / 32);
    __syncthreads();
  }
  if (krem > 0) {
    if (nkb == 0) {
      for (int ci = tid; ci < (krem / 32) * 128; ci += 256) {
        const int b = ci >> 7;
        const int r = (ci >> 2) & 31;
        const int c = ci & 3;
        const int src_row = r < Mr ? r : Mr - 1;
        const bf16x8 v = *reinterpret_cast<const bf16x8*>(
            A + (long)src_row * K + k0 + b * 32 + c * 8);
        unsigned short* dst =
            a_lds[0] + b * (32 * 32) + r * 32 + ((c * 16) ^ sk_swz(r)) / 2;
        *reinterpret_cast<bf16x8*>(dst) = v;
      }
      __syncthreads();
    }
    mfma_block(nkb == 0 ? 0 : (nkb & 1), krem / 32);
  }

  // C[i][j]: i (k-dim of frag) = row, j = col; 16x16x32 C map:
  // col = l15, row = lg*4 + reg  -> out row = lg*4+reg (+16 for acc[1]),
  // out col = n0 + wid*16 + l15
  const int col = n0 + wid * 16 + l15;
  if (KS > 1) {
    float* slot = ws + ((long)blockIdx.y * SK_M) * N;
#pragma unroll
    for (int f = 0; f < 2; ++f)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = f * 16 + lg * 4 + r;
        slot[(long)row * N + col] = acc[f][r];
      }
    return;
  }
  // KS == 1: fused epilogue, bf16 out
  float sc[2][4];
#pragma unroll
  for (int f = 0; f < 2; ++f)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = f * 16 + lg * 4 + r;
      const int rm = row < Mr ? row : Mr - 1;
      sc[f][r] = stat_in != nullptr
                     ? rsqrtf(stat_in[rm] * stat_mul + stat_eps)
                     : 1.0f;
    }
#pragma unroll
  for (int f = 0; f < 2; ++f)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = f * 16 + lg * 4 + r;
      if (row >= Mr) continue;
      float v = acc[f][r] * sc[f][r];
      if (EPI == 1) {
        const float partner = __shfl_xor(v, 1, WAVE);
        const float out = sk_silu(v) * partner;
        if ((l15 & 1) == 0) C[(long)row * (N / 2) + (col >> 1)] = f2bf(out);
      } else {
        C[(long)row * N + col] = f2bf(v);
      }
    }
}

// reduce ws[KS][M][N] -> bf16 with epilogue.
// EPI 0: rowscale (stat_in); EPI 2: + resid, row-sumsq -> stat_out.
// Grid (Mr rows x N-chunks of 2048) so the reduce uses the whole chip;
// the EPI2 row statistic accumulates via one atomicAdd per (row, chunk)
// block — the same discipline as gemm256's resid epilogue (stat_out is
// zeroed by the launcher).
#define SK_RC 2048
template <int EPI>
__global__ __launch_bounds__(256) void gemmsk_reduce_kernel(
    unsigned short* __restrict__ C, const float* __restrict__ ws,
    const unsigned short* __restrict__ resid,
    const float* __restrict__ stat_in, float* __restrict__ stat_out,
    int Mr, int N, int KS, float stat_mul, float stat_eps) {
  const int row = blockIdx.x;
  const int c0 = blockIdx.y * SK_RC;
  const int cend = min(c0 + SK_RC, N);
  if (row >= Mr) return;
  __shared__ float ssq_lds[256];
  const float sc = (EPI == 0 && stat_in != nullptr)
                       ? rsqrtf(stat_in[row] * stat_mul + stat_eps)
                       : 1.0f;
  float ss = 0.0f;
  for (int n = c0 + threadIdx.x; n < cend; n += 256) {
    float v = 0.0f;
    for (int k = 0; k < KS; ++k) v += ws[((long)k * SK_M + row) * N + n];
    if (EPI == 2) {
      v += bf2f(resid[(long)row * N + n]);
      ss += v * v;
      C[(long)row * N + n] = f2bf(v);
    } else {
      C[(long)row * N + n] = f2bf(v * sc);
    }
  }
  if (EPI == 2 && stat_out != nullptr) {
    ssq_lds[threadIdx.x] = ss;
    __syncthreads();
    for (int off = 128; off > 0; off >>= 1) {
      if (threadIdx.x < off) ssq_lds[threadIdx.x] += ssq_lds[threadIdx.x + off];
      __syncthreads();
    }
    if (threadIdx.x == 0) atomicAdd(stat_out + row, ssq_lds[0]);
  }
}

// epi: 0 rowscale, 1 swiglu(+rowscale), 2 resid(+stat_out)
extern "C" void launch_gemmsk(int epi, void* C, void* ws, const void* A,
                              const void* B, const void* resid,
                              const void* stat_in, void* stat_out, int M,
                              int N, int K, float stat_mul, float stat_eps,
                              hipStream_t stream) {
  // pick KS: fill the chip (>=256 WGs) with Kc <= SK_KC_MAX, Kc % 32 == 0
  const int ntiles = N / SK_NT;
  int KS = 1;
  if (epi != 1) {  // swiglu callers (gate/up) have huge N; keep KS=1 there
    while (KS * ntiles < 256 && (K / (KS * 2)) % 32 == 0 && K / (KS * 2) >= 64)
      KS *= 2;
    // bound per-WG serial K only while the grid still has headroom
    // (e.g. lm_head N=128256 already fills the chip at KS=1)
    while (KS * ntiles < 1024 && K / KS > SK_KC_MAX &&
           (K / (KS * 2)) % 32 == 0)
      KS *= 2;
  }
  if (epi == 2 && KS == 1) KS = 2;  // resid epilogue lives in the reduce
  dim3 grid(ntiles, KS), block(256);
  if (KS == 1) {
    if (epi == 1)
      hipLaunchKernelGGL((gemmsk_kernel<1>), grid, block, 0, stream,
                         (unsigned short*)C, (float*)nullptr,
                         (const unsigned short*)A, (const unsigned short*)B,
                         (const float*)stat_in, M, N, K, 1, stat_mul, stat_eps);
    else
      hipLaunchKernelGGL((gemmsk_kernel<0>), grid, block, 0, stream,
                         (unsigned short*)C, (float*)nullptr,
                         (const unsigned short*)A, (const unsigned short*)B,
                         (const float*)(epi == 0 ? stat_in : nullptr), M, N, K,
                         1, stat_mul, stat_eps);
    return;
  }
  hipLaunchKernelGGL((gemmsk_kernel<0>), grid, block, 0, stream,
                     (unsigned short*)nullptr, (float*)ws,
                     (const unsigned short*)A, (const unsigned short*)B,
                     (const float*)nullptr, M, N, K, KS, stat_mul, stat_eps);
  dim3 g2(SK_M, (N + SK_RC - 1) / SK_RC), b2(256);
  if (epi == 2 && stat_out != nullptr)
    (void)hipMemsetAsync(stat_out, 0, (size_t)M * sizeof(float), stream);
  if (epi == 2)
    hipLaunchKernelGGL((gemmsk_reduce_kernel<2>), g2, b2, 0, stream,
                       (unsigned short*)C, (const float*)ws,
                       (const unsigned short*)resid, (const float*)nullptr,
                       (float*)stat_out, M, N, KS, stat_mul, stat_eps);
  else
    hipLaunchKernelGGL((gemmsk_reduce_kernel<0>), g2, b2, 0, stream,
                       (unsigned short*)C, (const float*)ws,
                       (const unsigned short*)nullptr, (const float*)stat_in,
                       (float*)nullptr, M, N, KS, stat_mul, stat_eps);
}

// how many f32 elements of workspace launch_gemmsk may touch
extern "C" long gemmsk_ws_elems(int N, int K) {
  int ntiles = N / SK_NT;
  int KS = 1;
  while (KS * ntiles < 256 && (K / (KS * 2)) % 32 == 0 && K / (KS * 2) >= 64)
    KS *= 2;
  while (KS * ntiles < 1024 && K / KS > SK_KC_MAX && (K / (KS * 2)) % 32 == 0)
    KS *= 2;
  if (KS == 1) KS = 2;  // epi==2 may force a split; allocate for it
  return (long)KS * SK_M * N;
}
