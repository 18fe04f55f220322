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
#include "common.h"

#define D_HEAD 128
#define QBLK 32
#define KVBLK 64
#define WG_QROWS 128  // 4 waves * QBLK

typedef float f32x16 __attribute__((ext_vector_type(16)));

__device__ __forceinline__ unsigned int pack_bf16(float lo, float hi) {
  return (unsigned int)f2bf(lo) | ((unsigned int)f2bf(hi) << 16);
}

// K image swizzle: row stride 256 B, XOR byte bits 4..7 with kv&15 so the
// 16-lane ds_read_b128 groups land on distinct 16-B slots (guide T2).
__device__ __forceinline__ int k_lds_off(int kv, int byte_in_row) {
  return kv * 256 + (byte_in_row ^ ((kv & 15) << 4));
}

// V^T image: [d][kv] rows of 128 B, XOR byte bits 4..6 with d&7.
__device__ __forceinline__ int vt_lds_off(int d, int byte_in_row) {
  return d * 128 + (byte_in_row ^ ((d & 7) << 4));
}

__global__ __launch_bounds__(256, 2) void attn_prefill_kernel(
    unsigned short* __restrict__ out,      // [B,S,Hq,D]
    const unsigned short* __restrict__ q,  // [B,S,Hq,D]
    const unsigned short* __restrict__ k,  // [B,S,Hkv,D]
    const unsigned short* __restrict__ v,  // [B,S,Hkv,D]
    int B, int Hq, int Hkv, int S, float scale, int causal) {
  __shared__ __attribute__((aligned(16))) char smem[KVBLK * 256 + D_HEAD * 128];
  char* k_lds = smem;                  // swizzled K tile [64][128] bf16
  char* vt_lds = smem + KVBLK * 256;   // swizzled V^T tile [128][64] bf16

  const int wg = blockIdx.x;
  const int nqblk = (S + WG_QROWS - 1) / WG_QROWS;
  const int bh = wg / nqblk;
  // deepest q-blocks first: under causal masking the last q-block has the
  // most KV tiles — schedule it first so the tail is short workgroups
  const int qblk = nqblk - 1 - (wg % nqblk);
  const int b = bh / Hq;
  const int hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;
  const int hi = lane >> 5;    // half-wave
  const int l31 = lane & 31;

  // BSHD strides: consecutive sequence positions are H*D elements apart
  const long q_sstride = (long)Hq * D_HEAD;
  const long kv_sstride = (long)Hkv * D_HEAD;
  const long q_base = (long)b * S * q_sstride + (long)hq * D_HEAD;
  const long kv_base = (long)b * S * kv_sstride + (long)hkv * D_HEAD;

  // wave-striped q assignment: wave w owns rows {qbase + 4*i + w}, i=0..31
  // (causal kv ranges match across waves -> no idle compute waves)
  const int qbase = qblk * WG_QROWS;
  const int my_q = qbase + 4 * l31 + wid;          // this lane's q row (S^T col)

  // ---- load Q fragments: B-operand layout, 8 slices of d (16 each) ----
  bf16x8 qf[8];
  {
    const unsigned short* qrow = q + q_base + (long)my_q * q_sstride;
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
  }

  // ---- online softmax state + O accumulators (4 d-tiles of 32) ----
  float m_run = -1e30f;
  float l_run = 0.f;
  f32x16 o_acc[4] = {};

  const int q_hi_wg = qbase + WG_QROWS - 1;  // max q row in WG
  int kv_end = S;
  if (causal) kv_end = min(S, q_hi_wg + 1);
  const int my_q_hi = qbase + 4 * 31 + wid;  // this wave's max q row

  for (int kv0 = 0; kv0 < kv_end; kv0 += KVBLK) {
    // ---- cooperative staging ----
    // K tile: 64 rows x 256 B; 256 threads x 4 chunks of 16 B
    {
#pragma unroll
      for (int it = 0; it < 4; ++it) {
        int chunk = it * 256 + tid;         // 1024 chunks of 16 B
        int kv = chunk >> 4;                // 16 chunks per row
        int byte = (chunk & 15) * 16;
        int kvg = kv0 + kv;
        ushort8v val;
        if (kvg < S)
          val = *reinterpret_cast<const ushort8v*>(k + kv_base + (long)kvg * kv_sstride + byte / 2);
        else
          val = ushort8v{0, 0, 0, 0, 0, 0, 0, 0};
        *reinterpret_cast<ushort8v*>(k_lds + k_lds_off(kv, byte)) = val;
      }
      // V tile transposed: thread reads 8 contiguous d of one kv row,
      // scatter-writes them to VT rows
#pragma unroll
      for (int it = 0; it < 4; ++it) {
        int chunk = it * 256 + tid;
        int kv = chunk >> 4;
        int d0 = (chunk & 15) * 8;
        int kvg = kv0 + kv;
        ushort8v val;
        if (kvg < S)
          val = *reinterpret_cast<const ushort8v*>(v + kv_base + (long)kvg * kv_sstride + d0);
        else
          val = ushort8v{0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
        for (int j = 0; j < 8; ++j)
          *reinterpret_cast<unsigned short*>(vt_lds + vt_lds_off(d0 + j, kv * 2)) = val[j];
      }
    }
    __syncthreads();

    bool compute = (!causal) || (kv0 <= my_q_hi);
    if (compute) {
      // ---- QK^T (swapped): S^T[kv][q] in two 32-kv tiles ----
      f32x16 st[2] = {};
#pragma unroll
      for (int t = 0; t < 2; ++t) {
#pragma unroll
        for (int s = 0; s < 8; ++s) {
          int byte = (s * 16 + hi * 8) * 2;
          int kv = t * 32 + l31;
          bf16x8 kf = *reinterpret_cast<const bf16x8*>(k_lds + k_lds_off(kv, byte));
          st[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[s], st[t], 0, 0, 0);
        }
      }

      // ---- mask + online softmax ----
      float m_tile = -1e30f;
#pragma unroll
      for (int t = 0; t < 2; ++t) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int kvg = kv0 + t * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
          float sv = st[t][r] * scale;
          if (kvg >= S || (causal && kvg > my_q)) sv = -1e30f;
          st[t][r] = sv;
          m_tile = fmaxf(m_tile, sv);
        }
      }
      m_tile = fmaxf(m_tile, __shfl_xor(m_tile, 32, WAVE));
      float m_new = fmaxf(m_run, m_tile);
      // all-masked tile guard (fully OOB rows keep m_new = -1e30)
      float alpha = (m_run <= -1e30f) ? 0.f : __expf(m_run - m_new);
      if (m_new <= -1e30f) alpha = 1.f;

      float p_sum = 0.f;
#pragma unroll
      for (int t = 0; t < 2; ++t)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          float pv = (st[t][r] <= -1e30f) ? 0.f : __expf(st[t][r] - m_new);
          st[t][r] = pv;
          p_sum += pv;
        }
      p_sum += __shfl_xor(p_sum, 32, WAVE);
      l_run = l_run * alpha + p_sum;
      m_run = m_new;
      // O rows are q-indexed by the C-layout REGISTER pattern, not by the
      // lane (alpha lives at lane q = l&31): redistribute before rescaling
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

      // ---- P -> bf16 A fragments via permlane32_swap ----
      // per 32-kv tile: 8 packs -> 2 swaps x2 -> A slices (16 kv each)
      bf16x8 pa[4];  // 4 slices of 16 kv covering the 64-kv tile
#pragma unroll
      for (int t = 0; t < 2; ++t) {
        unsigned int pk[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) pk[j] = pack_bf16(st[t][2 * j], st[t][2 * j + 1]);
        // slice 0 of this tile (kv t*32 + 0..15)
        {
          auto r0 = __builtin_amdgcn_permlane32_swap(pk[0], pk[2], false, false);
          auto r1 = __builtin_amdgcn_permlane32_swap(pk[1], pk[3], false, false);
          unsigned int w0 = r0[0], w2 = r0[1], w1 = r1[0], w3 = r1[1];
          pa[t * 2] = __builtin_bit_cast(bf16x8, (uint4{w0, w1, w2, w3}));
        }
        // slice 1 (kv t*32 + 16..31)
        {
          auto r0 = __builtin_amdgcn_permlane32_swap(pk[4], pk[6], false, false);
          auto r1 = __builtin_amdgcn_permlane32_swap(pk[5], pk[7], false, false);
          unsigned int w0 = r0[0], w2 = r0[1], w1 = r1[0], w3 = r1[1];
          pa[t * 2 + 1] = __builtin_bit_cast(bf16x8, (uint4{w0, w1, w2, w3}));
        }
      }

      // ---- PV: O[q][d] += P @ V ----
#pragma unroll
      for (int dt = 0; dt < 4; ++dt) {
#pragma unroll
        for (int s = 0; s < 4; ++s) {
          // B operand: V[kv = s*16 + hi*8 + jj][d = dt*32 + l31]
          int d = dt * 32 + l31;
          int byte = (s * 16 + hi * 8) * 2;
          bf16x8 vf = *reinterpret_cast<const bf16x8*>(vt_lds + vt_lds_off(d, byte));
          o_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa[s], vf, o_acc[dt], 0, 0, 0);
        }
      }
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
    int qg = qbase + 4 * qrow + wid;
    if (qg >= S) continue;
    unsigned short* orow = out + q_base + (long)qg * q_sstride;
#pragma unroll
    for (int dt = 0; dt < 4; ++dt)
      orow[dt * 32 + l31] = f2bf(o_acc[dt][r] * inv_for_row[r]);
  }
}

extern "C" void launch_attn_prefill(void* out, const void* q, const void* k,
                                    const void* v, int B, int Hq, int Hkv,
                                    int S, float scale, int causal,
                                    hipStream_t stream) {
  int nqblk = (S + WG_QROWS - 1) / WG_QROWS;
  dim3 grid(B * Hq * nqblk), block(256);
  hipLaunchKernelGGL(attn_prefill_kernel, grid, block, 0, stream,
                     (unsigned short*)out, (const unsigned short*)q,
                     (const unsigned short*)k, (const unsigned short*)v, B, Hq,
                     Hkv, S, scale, causal);
}

// ---------------------------------------------------------------------------
// Decode attention (single new token, GQA, KV cache).
//   q: [B,Hq,D] bf16; kc/vc: [B,Hkv,Smax,D] bf16; out: [B,Hq,D]
// One workgroup (4 waves) per (b,hq); waves stride the kv length; memory-
// bound — coalesced 256 B K/V row reads, online softmax per wave, cross-
// wave merge through LDS (guide §B "attention decode").
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256, 2) void attn_decode_kernel(
    unsigned short* __restrict__ out, const unsigned short* __restrict__ q,
    const unsigned short* __restrict__ kc, const unsigned short* __restrict__ vc,
    int B, int Hq, int Hkv, int Smax, int L, float scale) {
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

extern "C" void launch_attn_decode(void* out, const void* q, const void* kc,
                                   const void* vc, int B, int Hq, int Hkv,
                                   int Smax, int L, float scale,
                                   hipStream_t stream) {
  dim3 grid(B * Hq), block(256);
  hipLaunchKernelGGL(attn_decode_kernel, grid, block, 0, stream,
                     (unsigned short*)out, (const unsigned short*)q,
                     (const unsigned short*)kc, (const unsigned short*)vc, B,
                     Hq, Hkv, Smax, L, scale);
}
