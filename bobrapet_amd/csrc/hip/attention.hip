// Flash attention (prefill + decode) for MI355X (gfx950), bf16, D=128.
//
// Prefill structure (guide §B "fused attention prefill"):
//  - workgroup = 4 waves (256 thr), each wave owns QBLK=32 q rows
//    → 128 q rows per workgroup; grid = B * Hq * ceil(S/128)
//  - KV tiles of 64 staged in LDS: K as a XOR-swizzled [64][128] image
//    (bank-conflict fix, guide G4/T2), V transposed to [128][64] so the
//    PV B-operand reads are contiguous 16-byte ds_read_b128
//  - swapped QK^T: mfma(A=K, B=Q) gives S^T with the q index in the lane
//    (col = lane&31), so the online-softmax row reductions are per-lane
//    over registers + one shfl_xor(32) (guide T12 structure)
//  - P → bf16 A-fragments via pack + permlane32_swap (T12/T21 primitive)
//  - online softmax with running (m, l); O accumulated in f32 MFMA regs
//
// MFMA v_mfma_f32_32x32x16_bf16 fragment maps (guide §3; verified on
// hardware by the numerics tests against a torch fp32 reference):
//   A[i][k]: i = lane&31, k = 8*(lane>>5) + j   (j in 0..7)
//   B[k][j]: j = lane&31, k = 8*(lane>>5) + jj
//   C[i][j]: j = lane&31, i = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
#include <cstdlib>

#include "common.h"

#define D_HEAD 128
#define QBLK 32
#define KVBLK 64
#define NWAVES 8        // 8 waves (512 thr) => 2 waves/SIMD co-resident:
                        // one wave's softmax VALU hides under its partner's
                        // MFMA segment (guide §Two waves per SIMD)
#define WG_QROWS 256    // NWAVES * QBLK

typedef float f32x16 __attribute__((ext_vector_type(16)));

__device__ __forceinline__ unsigned int pack_bf16(float lo, float hi) {
  return (unsigned int)f2bf(lo) | ((unsigned int)f2bf(hi) << 16);
}

// K image swizzle: row stride 256 B, XOR byte bits 4..7 with kv&15 so the
// 16-lane ds_read_b128 groups land on distinct 16-B slots (guide T2).
__device__ __forceinline__ int k_lds_off(int kv, int byte_in_row) {
  return kv * 256 + (byte_in_row ^ ((kv & 15) << 4));
}

// V^T image: [d][kv] rows of 128 B.  XOR byte bits 4..6 with (d>>1)&7:
// combined with the row-parity bank bit (32*d mod 64) the b128 read bank is
// injective in d mod 16, which is exactly the spread of a 16-lane
// ds_read_b128 group — conflict-free PV reads (guide T2 derivation).
__device__ __forceinline__ int vt_lds_off(int d, int byte_in_row) {
  return d * 128 + (byte_in_row ^ (((d >> 1) & 7) << 4));
}

template <int VARIANT>  // ablation bitmask: 1=stage 2=qk+softmax 4=pv (7=full)
__global__ __launch_bounds__(512, 2) void attn_prefill_kernel(
    unsigned short* __restrict__ out,      // [B,S,Hq,D] (contiguous)
    const unsigned short* __restrict__ q,  // [B,S,Hq,D] (seq stride q_sstride_in)
    const unsigned short* __restrict__ k,  // [B,S,Hkv,D]
    const unsigned short* __restrict__ v,  // [B,S,Hkv,D] (row stride may differ)
    int B, int Hq, int Hkv, int S, long v_sstride, long q_sstride_in,
    const float* __restrict__ inv_freq,  // optional [64]: rope Q on load
    int pos0,                            // rope position offset (seq shards)
    float scale, int causal,
    float* __restrict__ stats) {  // optional [B,Hq,S,2] (m, l) exp2-domain
  // two K+V^T buffer pairs; pointers computed per use (an addrspace(3)
  // pointer array fails to compile as a static initializer)
  __shared__ __attribute__((aligned(16))) char smem[2 * (KVBLK * 256 + D_HEAD * 128)];
  const int BUFSTRIDE = KVBLK * 256 + D_HEAD * 128;
  auto k_buf = [&](int i) -> char* { return smem + i * BUFSTRIDE; };
  auto v_buf = [&](int i) -> char* { return smem + i * BUFSTRIDE + KVBLK * 256; };

  const int wg = blockIdx.x;
  const int nqblk = (S + WG_QROWS - 1) / WG_QROWS;
  // global depth-descending order (LPT): ALL deepest q-blocks dispatch
  // first, so under causal masking the stragglers are the short ones
  const int nbh = gridDim.x / nqblk;
  const int bh = wg % nbh;
  const int qblk = nqblk - 1 - (wg / nbh);
  const int b = bh / Hq;
  const int hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;
  const int hi = lane >> 5;    // half-wave
  const int l31 = lane & 31;

  // BSHD strides: consecutive sequence positions are H*D elements apart.
  // q may be a strided view straight into the fused qkv projection
  // (q_sstride_in), in which case inv_freq != nullptr applies rope to the
  // Q rows ON LOAD (pairs (d, d+64) sit in the same lane: qf[s]/qf[s+4])
  // — the rope kernel then only processes K and Q never round-trips HBM.
  const long o_sstride = (long)Hq * D_HEAD;
  const long kv_sstride = (long)Hkv * D_HEAD;
  const long q_base = (long)b * S * q_sstride_in + (long)hq * D_HEAD;
  const long o_base = (long)b * S * o_sstride + (long)hq * D_HEAD;
  const long kv_base = (long)b * S * kv_sstride + (long)hkv * D_HEAD;
  // v may be a strided view (e.g. a slice of the fused qkv projection)
  const long v_base = (long)b * S * v_sstride + (long)hkv * D_HEAD;

  // wave-striped q assignment: wave w owns rows {qbase + 8*i + w}, i=0..31
  // (causal kv ranges match across waves -> no idle compute waves)
  const int qbase = qblk * WG_QROWS;
  const int my_q = qbase + NWAVES * l31 + wid;     // this lane's q row (S^T col)

  // ---- load Q fragments: B-operand layout, 8 slices of d (16 each) ----
  bf16x8 qf[8];
  {
    const unsigned short* qrow = q + q_base + (long)my_q * q_sstride_in;
    bool valid = my_q < S;
#pragma unroll
    for (int s = 0; s < 8; ++s) {
      int d0 = s * 16 + hi * 8;
      if (valid) {
        qf[s] = *reinterpret_cast<const bf16x8*>(qrow + d0);
      } else {
        qf[s] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
      }
    }
    if (inv_freq != nullptr && valid) {
      // rope Q in-register: d = s*16 + hi*8 + j pairs with d+64 at s+4.
      // ~64 sincos per lane ONCE per workgroup pass — amortized over the
      // whole KV loop (the old rope kernel cost ~47 us/layer for Q)
      const float pos = (float)(pos0 + my_q);
#pragma unroll
      for (int s = 0; s < 4; ++s) {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int d = s * 16 + hi * 8 + j;
          float c, sn;
          __sincosf(pos * inv_freq[d], &sn, &c);
          const float a = (float)qf[s][j];
          const float b2 = (float)qf[s + 4][j];
          qf[s][j] = (__bf16)(a * c - b2 * sn);
          qf[s + 4][j] = (__bf16)(b2 * c + a * sn);
        }
      }
    }
  }

  // ---- online softmax state + O accumulators (4 d-tiles of 32) ----
  float m_run = -1e30f;
  float l_run = 0.f;
  f32x16 o_acc[4] = {};

  const int q_hi_wg = qbase + WG_QROWS - 1;  // max q row in WG
  int kv_end = S;
  if (causal) kv_end = min(S, q_hi_wg + 1);
  const int my_q_hi = qbase + NWAVES * 31 + wid;  // this wave's max q row

  // ---- double-buffered pipeline (guide T14 split + 2 LDS buffers):
  // per tile: write the pre-loaded NEXT tile into the other buffer, issue
  // global loads for the tile after it, compute the CURRENT tile, one
  // barrier. HBM latency hides under the MFMA phase.
  const int ntiles = (kv_end + KVBLK - 1) / KVBLK;

  ushort8v kreg[2], vreg[2];
  auto load_tile = [&](int tile) {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      int chunk = it * 512 + tid;
      int kv = chunk >> 4;
      int byte = (chunk & 15) * 16;
      int kvg = tile * KVBLK + kv;
      if (kvg < S)
        kreg[it] = *reinterpret_cast<const ushort8v*>(
            k + kv_base + (long)kvg * kv_sstride + byte / 2);
      else
        kreg[it] = ushort8v{0, 0, 0, 0, 0, 0, 0, 0};
      int d0 = (chunk & 15) * 8;
      if (kvg < S)
        vreg[it] = *reinterpret_cast<const ushort8v*>(
            v + v_base + (long)kvg * v_sstride + d0);
      else
        vreg[it] = ushort8v{0, 0, 0, 0, 0, 0, 0, 0};
    }
  };
  auto write_tile = [&](char* kbuf, char* vbuf) {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      int chunk = it * 512 + tid;
      int kv = chunk >> 4;
      int byte = (chunk & 15) * 16;
      *reinterpret_cast<ushort8v*>(kbuf + k_lds_off(kv, byte)) = kreg[it];
      int d0 = (chunk & 15) * 8;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        *reinterpret_cast<unsigned short*>(vbuf + vt_lds_off(d0 + j, kv * 2)) =
            vreg[it][j];
    }
  };

  if (ntiles > 0) {
    load_tile(0);
    write_tile(k_buf(0), v_buf(0));
    if (ntiles > 1) load_tile(1);
    __syncthreads();
  }

  for (int t = 0; t < ntiles; ++t) {
    const int kv0 = t * KVBLK;
    const int cur = t & 1, nxt = cur ^ 1;
    if (t + 1 < ntiles) {
      write_tile(k_buf(nxt), v_buf(nxt));  // regs hold tile t+1
      if (t + 2 < ntiles) load_tile(t + 2);  // issue early; lands next iter
    }
    char* kb = k_buf(cur);
    char* vb = v_buf(cur);

    bool compute = (!causal) || (kv0 <= my_q_hi);
    if constexpr ((VARIANT & 2) == 0) {
      // stage-only ablation: touch both buffers so staging isn't DCE'd
      if (compute) {
        bf16x8 keep0 = *reinterpret_cast<const bf16x8*>(kb + k_lds_off(l31, 0));
        bf16x8 keep1 = *reinterpret_cast<const bf16x8*>(vb + vt_lds_off(l31, 0));
        asm volatile("" ::"v"(keep0), "v"(keep1));
      }
      __syncthreads();
      continue;
    }
    if (compute) {
      // ---- QK^T (swapped): S^T[kv][q] in two 32-kv tiles ----
      f32x16 st[2] = {};
      __builtin_amdgcn_s_setprio(1);
      // ss outer / tt inner: the two accumulator chains alternate, so the
      // MFMA issue stream never stalls on its own dependent accumulator
      // (back-to-back issue 32 cyc/SIMD vs ~2x that for a serial chain)
#pragma unroll
      for (int ss = 0; ss < 8; ++ss) {
        int byte = (ss * 16 + hi * 8) * 2;
#pragma unroll
        for (int tt = 0; tt < 2; ++tt) {
          int kv = tt * 32 + l31;
          bf16x8 kf = *reinterpret_cast<const bf16x8*>(kb + k_lds_off(kv, byte));
          st[tt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[ss], st[tt], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);

      // ---- mask + online softmax (log2 domain; v_exp_f32 IS exp2) ----
      // Edge handling hoisted per tile: interior tiles skip the per-element
      // mask (guide trap 4c: never a per-element runtime select).
      const float sc2 = scale * 1.44269504f;  // fold log2(e) into the scale
      const bool edge = (kv0 + KVBLK > S) || (causal && (kv0 + KVBLK - 1) > my_q);
      float m_tile = -1e30f;
      if (edge) {
#pragma unroll
        for (int tt = 0; tt < 2; ++tt)
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            int kvg = kv0 + tt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
            float sv = st[tt][r] * sc2;
            if (kvg >= S || (causal && kvg > my_q)) sv = -1e30f;
            st[tt][r] = sv;
            m_tile = fmaxf(m_tile, sv);
          }
      } else {
#pragma unroll
        for (int tt = 0; tt < 2; ++tt)
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            float sv = st[tt][r] * sc2;
            st[tt][r] = sv;
            m_tile = fmaxf(m_tile, sv);
          }
      }
      m_tile = fmaxf(m_tile, __shfl_xor(m_tile, 32, WAVE));

      // defer-max (guide T13): when the tile max stays within THR2 of the
      // running max, keep m_run and skip the O rescale entirely.  P is then
      // bounded by 2^THR2; the f32 accumulator tolerates it.  Safe order:
      // the decision precedes this tile's exponentiation (textbook form).
      const float THR2 = 8.0f;
      bool defer = __builtin_amdgcn_wave_reduce_and_b32(
                       (m_run > -1e30f) && (m_tile - m_run <= THR2), 0) != 0;
      float m_new;
      if (defer) {
        m_new = m_run;
      } else {
        m_new = fmaxf(m_run, m_tile);
        float alpha = (m_run <= -1e30f) ? 0.f : __builtin_amdgcn_exp2f(m_run - m_new);
        if (m_new <= -1e30f) alpha = 1.f;
        float alpha_row[16];
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int qrow = (r & 3) + 8 * (r >> 2) + 4 * hi;
          alpha_row[r] = __shfl(alpha, qrow, WAVE);
        }
#pragma unroll
        for (int dt = 0; dt < 4; ++dt)
#pragma unroll
          for (int r = 0; r < 16; ++r) o_acc[dt][r] *= alpha_row[r];
        l_run *= alpha;
        m_run = m_new;
      }

      float p_sum = 0.f;
#pragma unroll
      for (int tt = 0; tt < 2; ++tt)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          float pv = (st[tt][r] <= -1e30f) ? 0.f
                                           : __builtin_amdgcn_exp2f(st[tt][r] - m_new);
          st[tt][r] = pv;
          p_sum += pv;
        }
      p_sum += __shfl_xor(p_sum, 32, WAVE);
      l_run += p_sum;

      if constexpr ((VARIANT & 4) == 0) {
        // no-PV ablation: keep the softmax results live
        asm volatile("" ::"v"(p_sum), "v"(st[0][0]), "v"(st[1][15]),
                     "v"(l_run), "v"(m_run));
        __syncthreads();
        continue;
      }
      // ---- P -> bf16 A fragments via permlane32_swap ----
      bf16x8 pa[4];
#pragma unroll
      for (int tt = 0; tt < 2; ++tt) {
        unsigned int pk[8];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          pk[j] = pack_bf16(st[tt][2 * j], st[tt][2 * j + 1]);
        {
          auto r0 = __builtin_amdgcn_permlane32_swap(pk[0], pk[2], false, false);
          auto r1 = __builtin_amdgcn_permlane32_swap(pk[1], pk[3], false, false);
          unsigned int w0 = r0[0], w2 = r0[1], w1 = r1[0], w3 = r1[1];
          pa[tt * 2] = __builtin_bit_cast(bf16x8, (uint4{w0, w1, w2, w3}));
        }
        {
          auto r0 = __builtin_amdgcn_permlane32_swap(pk[4], pk[6], false, false);
          auto r1 = __builtin_amdgcn_permlane32_swap(pk[5], pk[7], false, false);
          unsigned int w0 = r0[0], w2 = r0[1], w1 = r1[0], w3 = r1[1];
          pa[tt * 2 + 1] = __builtin_bit_cast(bf16x8, (uint4{w0, w1, w2, w3}));
        }
      }

      // ---- PV: O[q][d] += P @ V ----
      __builtin_amdgcn_s_setprio(1);
      // ss outer / dt inner: 4 independent O chains interleave (see QK^T)
#pragma unroll
      for (int ss = 0; ss < 4; ++ss) {
        int byte = (ss * 16 + hi * 8) * 2;
#pragma unroll
        for (int dt = 0; dt < 4; ++dt) {
          int d = dt * 32 + l31;
          bf16x8 vf = *reinterpret_cast<const bf16x8*>(vb + vt_lds_off(d, byte));
          o_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa[ss], vf, o_acc[dt], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();
  }

  // ---- epilogue: O /= l, store ----
  float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;
  float inv_for_row[16];
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int qrow = (r & 3) + 8 * (r >> 2) + 4 * hi;
    inv_for_row[r] = __shfl(inv_l, qrow, WAVE);  // same value in both halves
  }
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int qrow = (r & 3) + 8 * (r >> 2) + 4 * hi;
    int qg = qbase + NWAVES * qrow + wid;
    if (qg >= S) continue;
    unsigned short* orow = out + o_base + (long)qg * o_sstride;
#pragma unroll
    for (int dt = 0; dt < 4; ++dt)
      orow[dt * 32 + l31] = f2bf(o_acc[dt][r] * inv_for_row[r]);
  }
  // per-row softmax stats for cross-block merging (ring attention):
  // lane l31 of wave wid owns row my_q's (m_run, l_run); halves duplicate
  if (stats != nullptr && hi == 0 && my_q < S) {
    float2 ml{m_run, l_run};
    *reinterpret_cast<float2*>(stats + ((long)bh * S + my_q) * 2) = ml;
  }
}

extern "C" void launch_attn_prefill(void* out, const void* q, const void* k,
                                    const void* v, int B, int Hq, int Hkv,
                                    int S, long v_sstride, float scale,
                                    int causal, void* stats,
                                    hipStream_t stream) {
  int nqblk = (S + WG_QROWS - 1) / WG_QROWS;
  dim3 grid(B * Hq * nqblk), block(512);
  hipLaunchKernelGGL((attn_prefill_kernel<7>), grid, block, 0, stream,
                     (unsigned short*)out, (const unsigned short*)q,
                     (const unsigned short*)k, (const unsigned short*)v, B, Hq,
                     Hkv, S, v_sstride, (long)Hq * D_HEAD, (const float*)nullptr,
                     0, scale, causal, (float*)stats);
}

// q is a strided view into the fused qkv projection (seq stride
// q_sstride, head-contiguous); rope applied to Q rows on load using
// inv_freq [64] at positions pos0 + row — the rope kernel then only
// processes K and the Q rows never round-trip HBM
extern "C" void launch_attn_prefill_qrope(void* out, const void* q,
                                          const void* k, const void* v, int B,
                                          int Hq, int Hkv, int S,
                                          long v_sstride, long q_sstride,
                                          const void* inv_freq, int pos0,
                                          float scale, int causal,
                                          hipStream_t stream) {
  int nqblk = (S + WG_QROWS - 1) / WG_QROWS;
  dim3 grid(B * Hq * nqblk), block(512);
  hipLaunchKernelGGL((attn_prefill_kernel<7>), grid, block, 0, stream,
                     (unsigned short*)out, (const unsigned short*)q,
                     (const unsigned short*)k, (const unsigned short*)v, B, Hq,
                     Hkv, S, v_sstride, q_sstride, (const float*)inv_freq,
                     pos0, scale, causal, (float*)nullptr);
}

// ---------------------------------------------------------------------------
// Pipelined prefill: QK(t) MFMAs are ISSUED, then softmax+PV of tile t-1
// run in their shadow — the softmax VALU (47% of the lockstep kernel,
// profiles/attn_ablation_r01c.txt) overlaps the wave's own in-flight
// MFMAs instead of stalling the matrix pipe.  Costs: the score tile is
// double-buffered in registers (+32 VGPR) and V is TRIPLE-buffered in
// LDS (PV reads tile t-1 while tile t+1 stages), so occupancy is 1
// block/CU (2 waves/SIMD) — the overlap no longer depends on 4-wave
// co-residency.  K stays double-buffered.  Same fragment maps, masking,
// defer-max softmax and (m,l) stats as attn_prefill_kernel.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(512, 1) void attn_prefill_pipe_kernel(
    unsigned short* __restrict__ out,
    const unsigned short* __restrict__ q,
    const unsigned short* __restrict__ k,
    const unsigned short* __restrict__ v,
    int B, int Hq, int Hkv, int S, long v_sstride, float scale, int causal,
    float* __restrict__ stats) {
  __shared__ __attribute__((aligned(16))) char smem[2 * KVBLK * 256 + 3 * D_HEAD * 128];
  auto k_buf = [&](int i) -> char* { return smem + i * (KVBLK * 256); };
  auto v_buf = [&](int i) -> char* { return smem + 2 * KVBLK * 256 + i * (D_HEAD * 128); };

  const int wg = blockIdx.x;
  const int nqblk = (S + WG_QROWS - 1) / WG_QROWS;
  const int nbh = gridDim.x / nqblk;
  const int bh = wg % nbh;
  const int qblk = nqblk - 1 - (wg / nbh);
  const int b = bh / Hq;
  const int hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;
  const int hi = lane >> 5;
  const int l31 = lane & 31;

  const long q_sstride = (long)Hq * D_HEAD;
  const long kv_sstride = (long)Hkv * D_HEAD;
  const long q_base = (long)b * S * q_sstride + (long)hq * D_HEAD;
  const long kv_base = (long)b * S * kv_sstride + (long)hkv * D_HEAD;
  const long v_base = (long)b * S * v_sstride + (long)hkv * D_HEAD;

  const int qbase = qblk * WG_QROWS;
  const int my_q = qbase + NWAVES * l31 + wid;

  bf16x8 qf[8];
  {
    const unsigned short* qrow = q + q_base + (long)my_q * q_sstride;
    bool valid = my_q < S;
#pragma unroll
    for (int s = 0; s < 8; ++s) {
      int d0 = s * 16 + hi * 8;
      if (valid)
        qf[s] = *reinterpret_cast<const bf16x8*>(qrow + d0);
      else
        qf[s] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
    }
  }

  float m_run = -1e30f;
  float l_run = 0.f;
  f32x16 o_acc[4] = {};

  const int q_hi_wg = qbase + WG_QROWS - 1;
  int kv_end = S;
  if (causal) kv_end = min(S, q_hi_wg + 1);
  const int my_q_hi = qbase + NWAVES * 31 + wid;
  const int ntiles = (kv_end + KVBLK - 1) / KVBLK;

  ushort8v kreg[2], vreg[2];
  auto load_tile = [&](int tile) {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      int chunk = it * 512 + tid;
      int kv = chunk >> 4;
      int byte = (chunk & 15) * 16;
      int kvg = tile * KVBLK + kv;
      if (kvg < S)
        kreg[it] = *reinterpret_cast<const ushort8v*>(
            k + kv_base + (long)kvg * kv_sstride + byte / 2);
      else
        kreg[it] = ushort8v{0, 0, 0, 0, 0, 0, 0, 0};
      int d0 = (chunk & 15) * 8;
      if (kvg < S)
        vreg[it] = *reinterpret_cast<const ushort8v*>(
            v + v_base + (long)kvg * v_sstride + d0);
      else
        vreg[it] = ushort8v{0, 0, 0, 0, 0, 0, 0, 0};
    }
  };
  auto write_tile = [&](char* kbuf, char* vbuf) {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      int chunk = it * 512 + tid;
      int kv = chunk >> 4;
      int byte = (chunk & 15) * 16;
      *reinterpret_cast<ushort8v*>(kbuf + k_lds_off(kv, byte)) = kreg[it];
      int d0 = (chunk & 15) * 8;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        *reinterpret_cast<unsigned short*>(vbuf + vt_lds_off(d0 + j, kv * 2)) =
            vreg[it][j];
    }
  };

  if (ntiles > 0) {
    load_tile(0);
    write_tile(k_buf(0), v_buf(0));
    if (ntiles > 1) load_tile(1);
    __syncthreads();
  }

  f32x16 st_p[2];        // tile t-1 scores, softmaxed in tile t's shadow
  int pt = -1;           // which tile st_p holds (-1: none)
  const float sc2 = scale * 1.44269504f;

  // softmax(st_p for tile pt) + PV from v_buf(pt % 3); runs while the
  // CURRENT tile's QK MFMAs are in flight (st_p is complete: its chain
  // retired before the current issue began)
  auto softmax_pv = [&]() {
    const int kv0 = pt * KVBLK;
    const bool edge = (kv0 + KVBLK > S) || (causal && (kv0 + KVBLK - 1) > my_q);
    float m_tile = -1e30f;
    if (edge) {
#pragma unroll
      for (int tt = 0; tt < 2; ++tt)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int kvg = kv0 + tt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
          float sv = st_p[tt][r] * sc2;
          if (kvg >= S || (causal && kvg > my_q)) sv = -1e30f;
          st_p[tt][r] = sv;
          m_tile = fmaxf(m_tile, sv);
        }
    } else {
#pragma unroll
      for (int tt = 0; tt < 2; ++tt)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          float sv = st_p[tt][r] * sc2;
          st_p[tt][r] = sv;
          m_tile = fmaxf(m_tile, sv);
        }
    }
    m_tile = fmaxf(m_tile, __shfl_xor(m_tile, 32, WAVE));

    const float THR2 = 8.0f;
    bool defer = __builtin_amdgcn_wave_reduce_and_b32(
                     (m_run > -1e30f) && (m_tile - m_run <= THR2), 0) != 0;
    float m_new;
    if (defer) {
      m_new = m_run;
    } else {
      m_new = fmaxf(m_run, m_tile);
      float alpha = (m_run <= -1e30f) ? 0.f : __builtin_amdgcn_exp2f(m_run - m_new);
      if (m_new <= -1e30f) alpha = 1.f;
      float alpha_row[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int qrow = (r & 3) + 8 * (r >> 2) + 4 * hi;
        alpha_row[r] = __shfl(alpha, qrow, WAVE);
      }
#pragma unroll
      for (int dt = 0; dt < 4; ++dt)
#pragma unroll
        for (int r = 0; r < 16; ++r) o_acc[dt][r] *= alpha_row[r];
      l_run *= alpha;
      m_run = m_new;
    }

    float p_sum = 0.f;
#pragma unroll
    for (int tt = 0; tt < 2; ++tt)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float pv = (st_p[tt][r] <= -1e30f)
                       ? 0.f
                       : __builtin_amdgcn_exp2f(st_p[tt][r] - m_new);
        st_p[tt][r] = pv;
        p_sum += pv;
      }
    p_sum += __shfl_xor(p_sum, 32, WAVE);
    l_run += p_sum;

    bf16x8 pa[4];
#pragma unroll
    for (int tt = 0; tt < 2; ++tt) {
      unsigned int pk[8];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        pk[j] = pack_bf16(st_p[tt][2 * j], st_p[tt][2 * j + 1]);
      {
        auto r0 = __builtin_amdgcn_permlane32_swap(pk[0], pk[2], false, false);
        auto r1 = __builtin_amdgcn_permlane32_swap(pk[1], pk[3], false, false);
        unsigned int w0 = r0[0], w2 = r0[1], w1 = r1[0], w3 = r1[1];
        pa[tt * 2] = __builtin_bit_cast(bf16x8, (uint4{w0, w1, w2, w3}));
      }
      {
        auto r0 = __builtin_amdgcn_permlane32_swap(pk[4], pk[6], false, false);
        auto r1 = __builtin_amdgcn_permlane32_swap(pk[5], pk[7], false, false);
        unsigned int w0 = r0[0], w2 = r0[1], w1 = r1[0], w3 = r1[1];
        pa[tt * 2 + 1] = __builtin_bit_cast(bf16x8, (uint4{w0, w1, w2, w3}));
      }
    }

    const char* vb = v_buf(pt % 3);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ss = 0; ss < 4; ++ss) {
      int byte = (ss * 16 + hi * 8) * 2;
#pragma unroll
      for (int dt = 0; dt < 4; ++dt) {
        int d = dt * 32 + l31;
        bf16x8 vf = *reinterpret_cast<const bf16x8*>(vb + vt_lds_off(d, byte));
        o_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa[ss], vf,
                                                            o_acc[dt], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
  };

  for (int t = 0; t < ntiles; ++t) {
    const int kv0 = t * KVBLK;
    if (t + 1 < ntiles) {
      // stage K(t+1) into kb((t+1)&1) (last read: QK(t-1), before the
      // previous barrier) and V(t+1) into vb((t+1)%3) (last read:
      // PV(t-2), two barriers back)
      write_tile(k_buf((t + 1) & 1), v_buf((t + 1) % 3));
      if (t + 2 < ntiles) load_tile(t + 2);
    }

    const bool compute = (!causal) || (kv0 <= my_q_hi);
    if (compute) {
      const char* kb = k_buf(t & 1);
      f32x16 st_c[2] = {};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ss = 0; ss < 8; ++ss) {
        int byte = (ss * 16 + hi * 8) * 2;
#pragma unroll
        for (int tt = 0; tt < 2; ++tt) {
          int kv = tt * 32 + l31;
          bf16x8 kf = *reinterpret_cast<const bf16x8*>(kb + k_lds_off(kv, byte));
          st_c[tt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[ss], st_c[tt], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
      // tile t's MFMAs are now in flight; retire tile t-1 under them
      if (pt >= 0) softmax_pv();
      st_p[0] = st_c[0];
      st_p[1] = st_c[1];
      pt = t;
    } else if (pt >= 0) {
      softmax_pv();
      pt = -1;
    }
    __syncthreads();
  }
  if (pt >= 0) softmax_pv();  // flush: v_buf(pt%3) is untouched after
                              // the last stage (write_tile stops at
                              // ntiles-1, and (ntiles-1)%3 was staged
                              // two iterations before its PV here)

  float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;
  float inv_for_row[16];
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int qrow = (r & 3) + 8 * (r >> 2) + 4 * hi;
    inv_for_row[r] = __shfl(inv_l, qrow, WAVE);
  }
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int qrow = (r & 3) + 8 * (r >> 2) + 4 * hi;
    int qg = qbase + NWAVES * qrow + wid;
    if (qg >= S) continue;
    unsigned short* orow = out + q_base + (long)qg * q_sstride;
#pragma unroll
    for (int dt = 0; dt < 4; ++dt)
      orow[dt * 32 + l31] = f2bf(o_acc[dt][r] * inv_for_row[r]);
  }
  if (stats != nullptr && hi == 0 && my_q < S) {
    float2 ml{m_run, l_run};
    *reinterpret_cast<float2*>(stats + ((long)bh * S + my_q) * 2) = ml;
  }
}

extern "C" void launch_attn_prefill_pipe(void* out, const void* q,
                                         const void* k, const void* v, int B,
                                         int Hq, int Hkv, int S,
                                         long v_sstride, float scale,
                                         int causal, void* stats,
                                         hipStream_t stream) {
  int nqblk = (S + WG_QROWS - 1) / WG_QROWS;
  dim3 grid(B * Hq * nqblk), block(512);
  hipLaunchKernelGGL(attn_prefill_pipe_kernel, grid, block, 0, stream,
                     (unsigned short*)out, (const unsigned short*)q,
                     (const unsigned short*)k, (const unsigned short*)v, B, Hq,
                     Hkv, S, v_sstride, scale, causal, (float*)stats);
}

extern "C" void launch_attn_prefill_variant(int variant, void* out,
                                            const void* q, const void* k,
                                            const void* v, int B, int Hq,
                                            int Hkv, int S, float scale,
                                            int causal, hipStream_t stream) {
  int nqblk = (S + WG_QROWS - 1) / WG_QROWS;
  dim3 grid(B * Hq * nqblk), block(512);
  auto args = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, block, 0, stream, (unsigned short*)out,
                       (const unsigned short*)q, (const unsigned short*)k,
                       (const unsigned short*)v, B, Hq, Hkv, S,
                       (long)Hkv * D_HEAD, (long)Hq * D_HEAD,
                       (const float*)nullptr, 0, scale, causal,
                       (float*)nullptr);
  };
  if (variant == 1) args(attn_prefill_kernel<1>);
  else if (variant == 3) args(attn_prefill_kernel<3>);
  else args(attn_prefill_kernel<7>);
}

// ---------------------------------------------------------------------------
// Two-pass chunked decode attention (single token, GQA, KV cache) — the
// DEFAULT path when the single-pass grid (B*Hq WGs) cannot fill 256 CUs.
//   pass 1 (chunk kernel): persistent-WG grid (<=2048 WGs) walks
//     (b, hkv, chunk) tuples, skipping chunks beyond L in-register; per
//     tuple, lanes own KV rows (lane-per-row scores, no per-row
//     reductions), PV uses broadcast-coalesced V reads, and all GQA q
//     heads of the group share each K/V read; per-chunk online (m, l, O)
//     partials land in a workspace
//   pass 2 (combine kernel): grid (B x Hq) merges the live chunk partials
// Length L comes from host OR a device scalar (hipGraph-replayable); the
// grid is static under capture.  Measured vs the single-pass kernel at
// B=1..8 (Llama-3-8B shapes): 1.7-6x faster; above B*Hq>=512 the
// single-pass kernel wins (no workspace traffic) and the launcher routes
// there instead.
// ---------------------------------------------------------------------------

#define DEC_CHUNK 256
#define DEC_GMAX 8  // max GQA group size handled per WG

template <bool L_FROM_DEV>
__global__ __launch_bounds__(256, 4) void attn_decode_chunk_kernel(
    float* __restrict__ ws,  // [B][Hkv][NCHUNK][G][2+D] f32 (m, l, O)
    const unsigned short* __restrict__ q,   // [B][Hq][D]
    const unsigned short* __restrict__ kc,  // [B][Hkv][Smax][D]
    const unsigned short* __restrict__ vc,
    int B, int Hq, int Hkv, int Smax, int nchunk, int L,
    const int* __restrict__ L_dev, float scale) {
  if constexpr (L_FROM_DEV) L = *L_dev;
  const int G = Hq / Hkv;
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;

  // LDS: q rows (broadcast reads), per-lane p values, cross-wave merge
  __shared__ float q_lds[DEC_GMAX][D_HEAD];
  __shared__ float p_lds[4][DEC_GMAX][WAVE];
  __shared__ float sm[DEC_GMAX][4], sl[DEC_GMAX][4], so[DEC_GMAX][4][D_HEAD];

  // persistent-WG remap: the grid is sized for the DEVICE (>=256 WGs, <=2048),
  // not for B*Hkv*nchunk — each WG walks tuples with chunk fastest-varying and
  // skips dead chunks (kv0 >= L) in-register.  The cache is allocated at Smax
  // (8192) while decode typically runs at L~1k: a tuple-per-WG grid would be
  // ~90% dead launches, which is what made the first cut 2x SLOWER end-to-end
  // at b32 despite winning microbenchmarks.  Dead slots are never written;
  // the combine pass only reads live chunks (ceil(L/DEC_CHUNK)).
  const int total = B * Hkv * nchunk;
  for (int idx = blockIdx.x; idx < total; idx += gridDim.x) {
  const int chunk = idx % nchunk;
  const int hkv = (idx / nchunk) % Hkv;
  const int b = idx / (nchunk * Hkv);
  const int kv0 = chunk * DEC_CHUNK;
  if (kv0 >= L) continue;  // WG-uniform: L, idx uniform across the block
  float* slot = ws + ((((long)b * Hkv + hkv) * nchunk + chunk) * DEC_GMAX) * (2 + D_HEAD);
  const int kv_end = min(kv0 + DEC_CHUNK, L);

  for (int i = tid; i < G * D_HEAD; i += blockDim.x) {
    int g = i / D_HEAD, d = i % D_HEAD;
    q_lds[g][d] = bf2f(q[((long)b * Hq + hkv * G + g) * D_HEAD + d]);
  }
  __syncthreads();

  const long base = (((long)b * Hkv + hkv) * Smax) * D_HEAD;

  // phase A: lane-per-row scores (no per-row reductions)
  const int r = kv0 + wid * WAVE + lane;
  const bool valid = r < kv_end;
  const int r_safe = valid ? r : (kv_end - 1);
  float sacc[DEC_GMAX];
#pragma unroll
  for (int g = 0; g < DEC_GMAX; ++g) sacc[g] = 0.f;
  const unsigned short* krow = kc + base + (long)r_safe * D_HEAD;
#pragma unroll
  for (int d0 = 0; d0 < D_HEAD / 8; ++d0) {
    ushort8v k8 = *reinterpret_cast<const ushort8v*>(krow + d0 * 8);
    float kf[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) kf[j] = bf2f(k8[j]);
#pragma unroll
    for (int g = 0; g < DEC_GMAX; ++g) {
      if (g >= G) break;
#pragma unroll
      for (int j = 0; j < 8; ++j) sacc[g] += kf[j] * q_lds[g][d0 * 8 + j];
    }
  }
  float m_w[DEC_GMAX], l_w[DEC_GMAX];
#pragma unroll
  for (int g = 0; g < DEC_GMAX; ++g) {
    if (g >= G) break;
    float sv = valid ? sacc[g] * scale : -1e30f;
    float m = wave_reduce_max(sv);
    float pv = (sv <= -1e30f) ? 0.f : __expf(sv - m);
    p_lds[wid][g][lane] = pv;
    m_w[g] = m;
    l_w[g] = wave_reduce_sum(pv);
  }
  __syncthreads();

  // phase B: broadcast-PV — every lane reads the SAME row (coalesced V),
  // p broadcast from LDS; lane owns 2 output dims
  float o0[DEC_GMAX], o1[DEC_GMAX];
#pragma unroll
  for (int g = 0; g < DEC_GMAX; ++g) o0[g] = o1[g] = 0.f;
  const int rows = min(WAVE, kv_end - (kv0 + wid * WAVE));
  // 8-row batches: the per-row 4-B load is latency-bound when issued one
  // at a time (the runtime trip count blocks compiler unrolling); batching
  // 8 independent loads ahead of the FMAs keeps ~8 in flight and removes
  // the ~2-4x latency multiplier this loop carried at decode L
  const unsigned short* vrow0 = vc + base + (long)(kv0 + wid * WAVE) * D_HEAD + lane * 2;
  int rr = 0;
  for (; rr + 8 <= rows; rr += 8) {
    ushort2 v8[8];
#pragma unroll
    for (int u = 0; u < 8; ++u)
      v8[u] = *reinterpret_cast<const ushort2*>(vrow0 + (long)(rr + u) * D_HEAD);
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      float vx = bf2f(v8[u].x), vy = bf2f(v8[u].y);
#pragma unroll
      for (int g = 0; g < DEC_GMAX; ++g) {
        if (g >= G) break;
        float pv = p_lds[wid][g][rr + u];
        o0[g] += pv * vx;
        o1[g] += pv * vy;
      }
    }
  }
  for (; rr < rows; ++rr) {
    ushort2 v2 = *reinterpret_cast<const ushort2*>(vrow0 + (long)rr * D_HEAD);
    float vx = bf2f(v2.x), vy = bf2f(v2.y);
#pragma unroll
    for (int g = 0; g < DEC_GMAX; ++g) {
      if (g >= G) break;
      float pv = p_lds[wid][g][rr];
      o0[g] += pv * vx;
      o1[g] += pv * vy;
    }
  }

  // cross-wave merge (each wave covered disjoint rows)
#pragma unroll
  for (int g = 0; g < DEC_GMAX; ++g) {
    if (g >= G) break;
    sm[g][wid] = rows > 0 ? m_w[g] : -1e30f;
    sl[g][wid] = rows > 0 ? l_w[g] : 0.f;
    so[g][wid][lane * 2] = o0[g];
    so[g][wid][lane * 2 + 1] = o1[g];
  }
  __syncthreads();
  if (wid == 0) {
#pragma unroll
    for (int g = 0; g < DEC_GMAX; ++g) {
      if (g >= G) break;
      float m_g = fmaxf(fmaxf(sm[g][0], sm[g][1]), fmaxf(sm[g][2], sm[g][3]));
      float l_g = 0.f, a0 = 0.f, a1 = 0.f;
#pragma unroll
      for (int w = 0; w < 4; ++w) {
        float aw = (sm[g][w] <= -1e30f) ? 0.f : __expf(sm[g][w] - m_g);
        l_g += sl[g][w] * aw;
        a0 += so[g][w][lane * 2] * aw;
        a1 += so[g][w][lane * 2 + 1] * aw;
      }
      float* out = slot + g * (2 + D_HEAD);
      if (lane == 0) {
        out[0] = m_g;
        out[1] = l_g;
      }
      out[2 + lane * 2] = a0;
      out[2 + lane * 2 + 1] = a1;
    }
  }
  __syncthreads();  // protect q_lds/p_lds/merge LDS before the next tuple
  }  // tuple loop
}

template <bool L_FROM_DEV>
__global__ __launch_bounds__(64, 8) void attn_decode_combine_kernel(
    unsigned short* __restrict__ out,  // [B][Hq][D]
    const float* __restrict__ ws, int B, int Hq, int Hkv, int nchunk, int L,
    const int* __restrict__ L_dev) {
  if constexpr (L_FROM_DEV) L = *L_dev;
  const int G = Hq / Hkv;
  const int bh = blockIdx.x;
  const int b = bh / Hq;
  const int hq = bh % Hq;
  const int hkv = hq / G, g = hq % G;
  const int lane = threadIdx.x & (WAVE - 1);
  const int live = (L + DEC_CHUNK - 1) / DEC_CHUNK;
  float m_g = -1e30f;
  for (int c = 0; c < live; ++c) {
    const float* slot = ws + ((((long)b * Hkv + hkv) * nchunk + c) * DEC_GMAX + g) * (2 + D_HEAD);
    m_g = fmaxf(m_g, slot[0]);
  }
  float l_g = 0.f, a0 = 0.f, a1 = 0.f;
  for (int c = 0; c < live; ++c) {
    const float* slot = ws + ((((long)b * Hkv + hkv) * nchunk + c) * DEC_GMAX + g) * (2 + D_HEAD);
    float aw = (slot[0] <= -1e30f) ? 0.f : __expf(slot[0] - m_g);
    l_g += slot[1] * aw;
    a0 += slot[2 + lane * 2] * aw;
    a1 += slot[2 + lane * 2 + 1] * aw;
  }
  float inv = (l_g > 0.f) ? 1.f / l_g : 0.f;
  unsigned short* orow = out + ((long)b * Hq + hq) * D_HEAD;
  orow[lane * 2] = f2bf(a0 * inv);
  orow[lane * 2 + 1] = f2bf(a1 * inv);
}

// ---------------------------------------------------------------------------
// FUSED small-batch decode attention: ONE kernel per step instead of the
// two-pass chunk+combine.  One workgroup per (b,hkv); the 4 waves stride
// 64-row KV sub-blocks with a per-wave ONLINE softmax merge (rescale o by
// exp(m_old - m_new) when a later sub-block raises the max), then one
// cross-wave LDS merge writes the normalized output directly — no
// workspace round-trip, no combine launch.  Routed at B*Hkv <= 16 (the
// b1/b2 serving case, where the two-pass form spent ~23 us/layer of
// fixed cost); larger batches keep the chunked form (more WGs = more
// CUs at work).  L may be a device scalar (hipGraph-replayable); the
// sub-block loop handles any L without re-capture.
// ---------------------------------------------------------------------------
template <bool L_FROM_DEV>
__global__ __launch_bounds__(256, 4) void attn_decode_fused_kernel(
    unsigned short* __restrict__ out,       // [B][Hq][D]
    const unsigned short* __restrict__ q,   // [B][Hq][D]
    const unsigned short* __restrict__ kc,  // [B][Hkv][Smax][D]
    const unsigned short* __restrict__ vc,
    int B, int Hq, int Hkv, int Smax, int L,
    const int* __restrict__ L_dev, float scale) {
  if constexpr (L_FROM_DEV) L = *L_dev;
  const int G = Hq / Hkv;
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;
  const int hkv = blockIdx.x % Hkv;
  const int b = blockIdx.x / Hkv;

  __shared__ float q_lds[DEC_GMAX][D_HEAD];
  __shared__ float p_lds[4][DEC_GMAX][WAVE];
  __shared__ float sm[DEC_GMAX][4], sl[DEC_GMAX][4], so[DEC_GMAX][4][D_HEAD];

  for (int i = tid; i < G * D_HEAD; i += blockDim.x) {
    int g = i / D_HEAD, d = i % D_HEAD;
    q_lds[g][d] = bf2f(q[((long)b * Hq + hkv * G + g) * D_HEAD + d]);
  }
  __syncthreads();
  const long base = (((long)b * Hkv + hkv) * Smax) * D_HEAD;

  float m_w[DEC_GMAX], l_w[DEC_GMAX], o0[DEC_GMAX], o1[DEC_GMAX];
#pragma unroll
  for (int g = 0; g < DEC_GMAX; ++g) {
    m_w[g] = -1e30f;
    l_w[g] = 0.f;
    o0[g] = o1[g] = 0.f;
  }

  for (int r0 = wid * WAVE; r0 < L; r0 += 4 * WAVE) {
    const int r = r0 + lane;
    const bool valid = r < L;
    const int r_safe = valid ? r : L - 1;
    // phase A: lane-per-row scores
    float sacc[DEC_GMAX];
#pragma unroll
    for (int g = 0; g < DEC_GMAX; ++g) sacc[g] = 0.f;
    const unsigned short* krow = kc + base + (long)r_safe * D_HEAD;
#pragma unroll
    for (int d0 = 0; d0 < D_HEAD / 8; ++d0) {
      ushort8v k8 = *reinterpret_cast<const ushort8v*>(krow + d0 * 8);
      float kf[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) kf[j] = bf2f(k8[j]);
#pragma unroll
      for (int g = 0; g < DEC_GMAX; ++g) {
        if (g >= G) break;
#pragma unroll
        for (int j = 0; j < 8; ++j) sacc[g] += kf[j] * q_lds[g][d0 * 8 + j];
      }
    }
    // online merge of this sub-block into the wave's running (m, l, o)
#pragma unroll
    for (int g = 0; g < DEC_GMAX; ++g) {
      if (g >= G) break;
      float sv = valid ? sacc[g] * scale : -1e30f;
      float m_t = wave_reduce_max(sv);
      float m_new = fmaxf(m_w[g], m_t);
      float alpha = (m_w[g] <= -1e30f) ? 0.f : __expf(m_w[g] - m_new);
      if (m_new <= -1e30f) alpha = 1.f;
      float pv = (sv <= -1e30f) ? 0.f : __expf(sv - m_new);
      p_lds[wid][g][lane] = pv;
      l_w[g] = l_w[g] * alpha + wave_reduce_sum(pv);
      o0[g] *= alpha;
      o1[g] *= alpha;
      m_w[g] = m_new;
    }
    // phase B: broadcast-PV accumulation, 8-row batches (see the chunk
    // kernel note on load-latency serialization)
    const int rows = min(WAVE, L - r0);
    const unsigned short* vrow0 = vc + base + (long)r0 * D_HEAD + lane * 2;
    int rr = 0;
    for (; rr + 8 <= rows; rr += 8) {
      ushort2 v8[8];
#pragma unroll
      for (int u = 0; u < 8; ++u)
        v8[u] = *reinterpret_cast<const ushort2*>(vrow0 + (long)(rr + u) * D_HEAD);
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        float vx = bf2f(v8[u].x), vy = bf2f(v8[u].y);
#pragma unroll
        for (int g = 0; g < DEC_GMAX; ++g) {
          if (g >= G) break;
          float pv = p_lds[wid][g][rr + u];
          o0[g] += pv * vx;
          o1[g] += pv * vy;
        }
      }
    }
    for (; rr < rows; ++rr) {
      ushort2 v2 = *reinterpret_cast<const ushort2*>(vrow0 + (long)rr * D_HEAD);
      float vx = bf2f(v2.x), vy = bf2f(v2.y);
#pragma unroll
      for (int g = 0; g < DEC_GMAX; ++g) {
        if (g >= G) break;
        float pv = p_lds[wid][g][rr];
        o0[g] += pv * vx;
        o1[g] += pv * vy;
      }
    }
  }

  // cross-wave merge (waves covered disjoint sub-blocks) + direct output
#pragma unroll
  for (int g = 0; g < DEC_GMAX; ++g) {
    if (g >= G) break;
    sm[g][wid] = m_w[g];
    sl[g][wid] = l_w[g];
    so[g][wid][lane * 2] = o0[g];
    so[g][wid][lane * 2 + 1] = o1[g];
  }
  __syncthreads();
  if (wid == 0) {
#pragma unroll
    for (int g = 0; g < DEC_GMAX; ++g) {
      if (g >= G) break;
      float m_g = fmaxf(fmaxf(sm[g][0], sm[g][1]), fmaxf(sm[g][2], sm[g][3]));
      float l_g = 0.f, a0 = 0.f, a1 = 0.f;
#pragma unroll
      for (int w = 0; w < 4; ++w) {
        float aw = (sm[g][w] <= -1e30f) ? 0.f : __expf(sm[g][w] - m_g);
        l_g += sl[g][w] * aw;
        a0 += so[g][w][lane * 2] * aw;
        a1 += so[g][w][lane * 2 + 1] * aw;
      }
      float inv = (l_g > 0.f) ? 1.f / l_g : 0.f;
      unsigned short* orow = out + ((long)b * Hq + hkv * G + g) * D_HEAD;
      orow[lane * 2] = f2bf(a0 * inv);
      orow[lane * 2 + 1] = f2bf(a1 * inv);
    }
  }
}

// ---------------------------------------------------------------------------
// Decode attention (single new token, GQA, KV cache).
//   q: [B,Hq,D] bf16; kc/vc: [B,Hkv,Smax,D] bf16; out: [B,Hq,D]
// One workgroup (4 waves) per (b,hq); waves stride the kv length; memory-
// bound — coalesced 256 B K/V row reads, online softmax per wave, cross-
// wave merge through LDS (guide §B "attention decode").
// ---------------------------------------------------------------------------

template <bool L_FROM_DEV>
__global__ __launch_bounds__(256, 2) void attn_decode_kernel(
    unsigned short* __restrict__ out, const unsigned short* __restrict__ q,
    const unsigned short* __restrict__ kc, const unsigned short* __restrict__ vc,
    int B, int Hq, int Hkv, int Smax, int L, const int* __restrict__ L_dev,
    float scale) {
  if constexpr (L_FROM_DEV) L = *L_dev;  // hipGraph-replayable length
  const int bh = blockIdx.x;
  const int b = bh / Hq;
  const int hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;

  const unsigned short* qrow = q + ((long)b * Hq + hq) * D_HEAD;
  const long cache_base = (((long)b * Hkv + hkv) * Smax) * D_HEAD;

  // each lane owns 2 d positions
  float q0 = bf2f(qrow[lane * 2]);
  float q1 = bf2f(qrow[lane * 2 + 1]);

  float m_w = -1e30f, l_w = 0.f, o0 = 0.f, o1 = 0.f;
  for (int kv = wid; kv < L; kv += 4) {
    const unsigned short* krow = kc + cache_base + (long)kv * D_HEAD;
    float2 kf;
    {
      ushort2 kv2 = *reinterpret_cast<const ushort2*>(krow + lane * 2);
      kf.x = bf2f(kv2.x);
      kf.y = bf2f(kv2.y);
    }
    float s = q0 * kf.x + q1 * kf.y;
    s = wave_reduce_sum(s) * scale;
    float m_new = fmaxf(m_w, s);
    float alpha = __expf(m_w - m_new);
    float p = __expf(s - m_new);
    if (m_w <= -1e30f) alpha = 0.f;
    const unsigned short* vrow = vc + cache_base + (long)kv * D_HEAD;
    ushort2 vv = *reinterpret_cast<const ushort2*>(vrow + lane * 2);
    o0 = o0 * alpha + p * bf2f(vv.x);
    o1 = o1 * alpha + p * bf2f(vv.y);
    l_w = l_w * alpha + p;
    m_w = m_new;
  }

  // cross-wave merge
  __shared__ float sm[4], sl[4], so[4][D_HEAD];
  sm[wid] = m_w;
  sl[wid] = l_w;
  so[wid][lane * 2] = o0;
  so[wid][lane * 2 + 1] = o1;
  __syncthreads();
  if (wid == 0) {
    float m_g = fmaxf(fmaxf(sm[0], sm[1]), fmaxf(sm[2], sm[3]));
    float l_g = 0.f, a0 = 0.f, a1 = 0.f;
#pragma unroll
    for (int w = 0; w < 4; ++w) {
      float aw = (sm[w] <= -1e30f) ? 0.f : __expf(sm[w] - m_g);
      l_g += sl[w] * aw;
      a0 += so[w][lane * 2] * aw;
      a1 += so[w][lane * 2 + 1] * aw;
    }
    float inv = (l_g > 0.f) ? 1.f / l_g : 0.f;
    unsigned short* orow = out + ((long)b * Hq + hq) * D_HEAD;
    orow[lane * 2] = f2bf(a0 * inv);
    orow[lane * 2 + 1] = f2bf(a1 * inv);
  }
}

extern "C" void launch_attn_decode(void* out, void* workspace, const void* q,
                                   const void* kc, const void* vc, int B,
                                   int Hq, int Hkv, int Smax, int L,
                                   const void* L_dev, float scale,
                                   hipStream_t stream) {
  const int G = Hq / Hkv;
  // Two-pass chunked decode exists to MAKE parallelism when B*Hq workgroups
  // cannot fill 256 CUs (B<=8: 1.7-6x over single-pass).  Once the
  // single-pass grid alone fills the chip (B*Hq >= 512 WGs) the chunked
  // form only adds workspace traffic + a combine pass and measures ~2x
  // SLOWER end-to-end (b32 Llama-3-8B: 9.1 vs 18.3 ms/step) — so route
  // large batches to the single-pass kernel.
  // (A fused one-kernel variant — one WG per (b,hkv), online per-wave
  // merge, no workspace/combine — measured STRICTLY worse at every L:
  // 8 WGs cannot generate the read bandwidth the chunked grid gets from
  // 16-256 CUs (37.6 vs 22.3 us at L=256; 787 vs 45.5 at L=8192; decode
  // e2e 6.53 vs 3.96 ms/tok).  The two-pass "fixed cost" IS the
  // parallelism floor, not kernel overhead — docs/PERF.md late round 2.)
  // routing override for A/B probes: BOBRA_DEC_ATTN=chunk|single|fused.
  // The fused kernel is NEVER auto-routed: even at B*Hkv >= 64 (full
  // grid, KV read once) its per-WG serial sub-block walk loses to the
  // massively-parallel chunk/single grids (b8 L1024: 117 vs 61 us;
  // decode b8 9.6 -> 10.5 ms when it was routed) — kept as a probe.
  int route = 0;  // 0 auto, 1 force chunk, 2 force single, 3 force fused
  if (const char* e = getenv("BOBRA_DEC_ATTN"))
    route = e[0] == 'c' ? 1 : e[0] == 's' ? 2 : e[0] == 'f' ? 3 : 0;
  if (route == 3 && G <= DEC_GMAX) {
    dim3 g0(B * Hkv), b0(256);
    if (L_dev != nullptr)
      hipLaunchKernelGGL((attn_decode_fused_kernel<true>), g0, b0, 0, stream,
                         (unsigned short*)out, (const unsigned short*)q,
                         (const unsigned short*)kc, (const unsigned short*)vc,
                         B, Hq, Hkv, Smax, 0, (const int*)L_dev, scale);
    else
      hipLaunchKernelGGL((attn_decode_fused_kernel<false>), g0, b0, 0, stream,
                         (unsigned short*)out, (const unsigned short*)q,
                         (const unsigned short*)kc, (const unsigned short*)vc,
                         B, Hq, Hkv, Smax, L, (const int*)nullptr, scale);
    return;
  }
  const bool want_chunk =
      route == 1 || (route == 0 && B * Hq < 512);
  if (workspace != nullptr && G <= DEC_GMAX && want_chunk) {
    const int nchunk = (Smax + DEC_CHUNK - 1) / DEC_CHUNK;
    const int total = B * Hkv * nchunk;
    dim3 g1(total < 2048 ? total : 2048), b1(256);
    dim3 g2(B * Hq), b2(64);
    if (L_dev != nullptr) {
      hipLaunchKernelGGL((attn_decode_chunk_kernel<true>), g1, b1, 0, stream,
                         (float*)workspace, (const unsigned short*)q,
                         (const unsigned short*)kc, (const unsigned short*)vc,
                         B, Hq, Hkv, Smax, nchunk, 0, (const int*)L_dev, scale);
      hipLaunchKernelGGL((attn_decode_combine_kernel<true>), g2, b2, 0, stream,
                         (unsigned short*)out, (const float*)workspace, B, Hq,
                         Hkv, nchunk, 0, (const int*)L_dev);
    } else {
      hipLaunchKernelGGL((attn_decode_chunk_kernel<false>), g1, b1, 0, stream,
                         (float*)workspace, (const unsigned short*)q,
                         (const unsigned short*)kc, (const unsigned short*)vc,
                         B, Hq, Hkv, Smax, nchunk, L, (const int*)nullptr, scale);
      hipLaunchKernelGGL((attn_decode_combine_kernel<false>), g2, b2, 0, stream,
                         (unsigned short*)out, (const float*)workspace, B, Hq,
                         Hkv, nchunk, L, (const int*)nullptr);
    }
    return;
  }
  // fallback: single-pass kernel (large GQA groups / no workspace)
  dim3 grid(B * Hq), block(256);
  if (L_dev != nullptr)
    hipLaunchKernelGGL((attn_decode_kernel<true>), grid, block, 0, stream,
                       (unsigned short*)out, (const unsigned short*)q,
                       (const unsigned short*)kc, (const unsigned short*)vc, B,
                       Hq, Hkv, Smax, 0, (const int*)L_dev, scale);
  else
    hipLaunchKernelGGL((attn_decode_kernel<false>), grid, block, 0, stream,
                       (unsigned short*)out, (const unsigned short*)q,
                       (const unsigned short*)kc, (const unsigned short*)vc, B,
                       Hq, Hkv, Smax, L, (const int*)nullptr, scale);
}
