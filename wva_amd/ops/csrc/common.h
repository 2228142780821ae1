// Common device helpers for wva_amd MI355X (gfx950) kernels.
//
// CDNA4: wavefront = 64 lanes; SIMD-32 units issue wave64 VALU in 2 cycles.
// All reductions here are wave64-shaped (not 32-wide warp idioms).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE_SIZE 64

using bf16 = __hip_bfloat16;
using bf16x2 = __hip_bfloat162;

__device__ __forceinline__ float bf2f(bf16 v) { return __bfloat162float(v); }
__device__ __forceinline__ bf16 f2bf(float v) { return __float2bfloat16(v); }

// Full wave64 reduction (sum). 6 xor-shuffle steps across 64 lanes.
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int offset = 32; offset > 0; offset >>= 1) {
    v += __shfl_xor(v, offset, WAVE_SIZE);
  }
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int offset = 32; offset > 0; offset >>= 1) {
    v = fmaxf(v, __shfl_xor(v, offset, WAVE_SIZE));
  }
  return v;
}

// Block (workgroup) reduction over up to 16 waves via LDS.
// `lds` must hold >= blockDim.x / WAVE_SIZE floats; result valid in all lanes.
__device__ __forceinline__ float block_reduce_sum(float v, float* lds) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  const int num_waves = blockDim.x / WAVE_SIZE;
  v = wave_reduce_sum(v);
  if (lane == 0) lds[wave] = v;
  __syncthreads();
  float out = (lane < num_waves) ? lds[lane] : 0.0f;
  out = wave_reduce_sum(out);  // only first num_waves lanes carry data
  return __shfl(out, 0, WAVE_SIZE);
}
