// Common helpers for bobrapet_amd CDNA4 (gfx950) kernels.
//
// Design rules (see /opt/skills/guides/cdna_hip_programming.md):
//  - wave = 64 lanes, hard-coded
//  - bf16 loads vectorized as ushort4/ushort8 (8-16 B/lane)
//  - per-wave shuffle reductions over width 64
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64

typedef __hip_bfloat16 bf16_t;

typedef unsigned short ushort4v __attribute__((ext_vector_type(4)));
typedef unsigned short ushort8v __attribute__((ext_vector_type(8)));
typedef float float4v __attribute__((ext_vector_type(4)));
typedef float float16v __attribute__((ext_vector_type(16)));
typedef short short8v __attribute__((ext_vector_type(8)));
// MFMA fragment types (gfx950): bf16 A/B = 4 VGPRs (8 bf16), C/D f32
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));

__device__ __forceinline__ float bf2f(unsigned short u) {
  union {
    float f;
    unsigned int i;
  } cv;
  cv.i = ((unsigned int)u) << 16;
  return cv.f;
}

__device__ __forceinline__ unsigned short f2bf(float f) {
  union {
    float f;
    unsigned int i;
  } cv;
  cv.f = f;
  // round-to-nearest-even
  unsigned int lsb = (cv.i >> 16) & 1;
  cv.i += 0x7fffu + lsb;
  return (unsigned short)(cv.i >> 16);
}

// full-wave (64-lane) reductions
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// XCD-aware bijective blockIdx remap (guide §5: 8 XCDs, each with its own
// L2; contiguous grid chunks per XCD recover L2 locality on tiled ops).
__device__ __forceinline__ unsigned int xcd_swizzle(unsigned int wgid,
                                                    unsigned int nwg) {
  const unsigned int NXCD = 8;
  if (nwg < NXCD) return wgid;
  unsigned int q = nwg / NXCD, r = nwg % NXCD;
  unsigned int xcd = wgid % NXCD, idx = wgid / NXCD;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

#define HIP_CHECK_KERNEL()                                                     \
  do {                                                                         \
  } while (0)
