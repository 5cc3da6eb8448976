// Shared device helpers for sparkdl CDNA4 (gfx950) kernels.
//
// Conventions (per the CDNA4 HIP programming model):
//   - wavefront = 64 lanes; all cross-lane idioms use 64-wide shuffles
//   - bf16 travels as short8 (16 B/lane vector loads — hipcc does not
//     auto-vectorize scalar bf16 loads)
//   - fp32 accumulation everywhere (no fp32-input MFMA on CDNA4; norms /
//     optimizers are memory-bound vector code)
#pragma once

#include <hip/hip_runtime.h>

#define WAVE 64

typedef short short8 __attribute__((ext_vector_type(8)));
typedef short short4v __attribute__((ext_vector_type(4)));
typedef float float4v __attribute__((ext_vector_type(4)));

// bf16 <-> fp32 bit conversions. f2bf rounds to nearest-even.
__device__ __forceinline__ float bf2f(short s) {
  union { float f; unsigned u; } c;
  c.u = ((unsigned)(unsigned short)s) << 16;
  return c.f;
}

__device__ __forceinline__ short f2bf(float f) {
  union { float f; unsigned u; } c;
  c.f = f;
  if ((c.u & 0x7fffffffu) > 0x7f800000u) return (short)0x7fc0;  // NaN
  unsigned lsb = (c.u >> 16) & 1u;
  c.u += 0x7fffu + lsb;
  return (short)(c.u >> 16);
}

// Full-wave butterfly sum: every lane ends with the 64-lane total.
__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 1; off < WAVE; off <<= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

// Block-level sum across waves (LDS staging). `scratch` needs
// blockDim.x/WAVE floats. Every thread returns the block total.
__device__ __forceinline__ float block_sum(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int nw = blockDim.x / WAVE;
  v = wave_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float total = 0.f;
#pragma unroll 4
  for (int i = 0; i < nw; ++i) total += scratch[i];
  __syncthreads();
  return total;
}
