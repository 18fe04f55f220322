// MFMA layout probes (debug-only): isolate the v_mfma_f32_32x32x16_bf16
// fragment-layout assumptions used by attention.hip.
#include "common.h"

typedef float f32x16d __attribute__((ext_vector_type(16)));

// C = A(32x16) @ B(16x32), one wave. A/B row-major f32 in, C f32 out.
// Verifies the A/B/C lane maps in one shot.
__global__ __launch_bounds__(64) void dbg_mfma_kernel(
    float* __restrict__ C, const float* __restrict__ A,
    const float* __restrict__ B) {
  int lane = threadIdx.x & 63;
  int hi = lane >> 5, l31 = lane & 31;
  bf16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int k = 8 * hi + j;
    a[j] = (__bf16)A[l31 * 16 + k];   // A[i=l31][k]
    b[j] = (__bf16)B[k * 32 + l31];   // B[k][j=l31]
  }
  f32x16d c = {};
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
    C[row * 32 + l31] = c[r];
  }
}

extern "C" void launch_dbg_mfma(void* C, const void* A, const void* B,
                                hipStream_t s) {
  hipLaunchKernelGGL(dbg_mfma_kernel, dim3(1), dim3(64), 0, s, (float*)C,
                     (const float*)A, (const float*)B);
}

// One wave: S^T = K(64x128) @ Q^T  then  O = P @ V with P = relu(S)+1 (a
// fixed elementwise stand-in for softmax so the permlane pack path is
// exercised without the online state).  Direct global loads, no LDS.
//   Q: [32][128] bf16, K,V: [64][128] bf16; S_out: [64][32] f32 (S^T),
//   O_out: [32][128] f32
__global__ __launch_bounds__(64) void dbg_attn_core_kernel(
    float* __restrict__ S_out, float* __restrict__ O_out,
    const unsigned short* __restrict__ Q, const unsigned short* __restrict__ K,
    const unsigned short* __restrict__ V) {
  int lane = threadIdx.x & 63;
  int hi = lane >> 5, l31 = lane & 31;

  // Q fragments: B operand, 8 d-slices
  bf16x8 qf[8];
#pragma unroll
  for (int s = 0; s < 8; ++s) {
    const unsigned short* p = Q + l31 * 128 + s * 16 + hi * 8;
#pragma unroll
    for (int j = 0; j < 8; ++j) qf[s][j] = (__bf16)bf2f(p[j]);
  }

  f32x16d st[2] = {};
#pragma unroll
  for (int t = 0; t < 2; ++t) {
#pragma unroll
    for (int s = 0; s < 8; ++s) {
      bf16x8 kf;
      const unsigned short* p = K + (t * 32 + l31) * 128 + s * 16 + hi * 8;
#pragma unroll
      for (int j = 0; j < 8; ++j) kf[j] = (__bf16)bf2f(p[j]);
      st[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[s], st[t], 0, 0, 0);
    }
  }
  // write S^T
#pragma unroll
  for (int t = 0; t < 2; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int kv = t * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
      S_out[kv * 32 + l31] = st[t][r];
    }

  // P = relu(S)+1 elementwise, then pack into A fragments
#pragma unroll
  for (int t = 0; t < 2; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) st[t][r] = fmaxf(st[t][r], 0.f) + 1.f;

  bf16x8 pa[4];
#pragma unroll
  for (int t = 0; t < 2; ++t) {
    unsigned int pk[8];
#pragma unroll
    for (int j = 0; j < 8; ++j)
      pk[j] = (unsigned int)f2bf(st[t][2 * j]) |
              ((unsigned int)f2bf(st[t][2 * j + 1]) << 16);
    {
      auto r0 = __builtin_amdgcn_permlane32_swap(pk[0], pk[2], false, false);
      auto r1 = __builtin_amdgcn_permlane32_swap(pk[1], pk[3], false, false);
      unsigned int w0 = r0[0], w2 = r0[1], w1 = r1[0], w3 = r1[1];
      pa[t * 2] = __builtin_bit_cast(bf16x8, (uint4{w0, w1, w2, w3}));
    }
    {
      auto r0 = __builtin_amdgcn_permlane32_swap(pk[4], pk[6], false, false);
      auto r1 = __builtin_amdgcn_permlane32_swap(pk[5], pk[7], false, false);
      unsigned int w0 = r0[0], w2 = r0[1], w1 = r1[0], w3 = r1[1];
      pa[t * 2 + 1] = __builtin_bit_cast(bf16x8, (uint4{w0, w1, w2, w3}));
    }
  }

  // O = P @ V
#pragma unroll
  for (int dt = 0; dt < 4; ++dt) {
    f32x16d o = {};
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      bf16x8 vf;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int kv = s * 16 + 8 * hi + j;
        vf[j] = (__bf16)bf2f(V[kv * 128 + dt * 32 + l31]);
      }
      o = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa[s], vf, o, 0, 0, 0);
    }
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int qrow = (r & 3) + 8 * (r >> 2) + 4 * hi;
      O_out[qrow * 128 + dt * 32 + l31] = o[r];
    }
  }
}

extern "C" void launch_dbg_attn_core(void* S_out, void* O_out, const void* Q,
                                     const void* K, const void* V,
                                     hipStream_t s) {
  hipLaunchKernelGGL(dbg_attn_core_kernel, dim3(1), dim3(64), 0, s,
                     (float*)S_out, (float*)O_out, (const unsigned short*)Q,
                     (const unsigned short*)K, (const unsigned short*)V);
}
