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

// --- fp8 (OCP e4m3fn) KV-cache helpers, gfx950 hardware converts ---
// The fp8 KV path stores K/V as 1-byte e4m3 (half the HBM traffic and
// double the KV capacity of bf16); compute stays bf16/fp32 — fragments
// are up-converted at build time through these.
using fp8_t = unsigned char;
typedef __attribute__((ext_vector_type(2))) float float2_vt;

__device__ __forceinline__ fp8_t f2fp8(float v) {
  return (fp8_t)(__builtin_amdgcn_cvt_pk_fp8_f32(v, 0.0f, 0u, false) & 0xffu);
}

// pack two floats -> two adjacent e4m3 bytes
__device__ __forceinline__ unsigned short pk2_fp8(float a, float b) {
  return (unsigned short)(__builtin_amdgcn_cvt_pk_fp8_f32(a, b, 0u, false) &
                          0xffffu);
}

// two adjacent e4m3 bytes -> two floats
__device__ __forceinline__ float2_vt unpk2_fp8(unsigned short v) {
  return __builtin_amdgcn_cvt_pk_f32_fp8((unsigned int)v, false);
}

__device__ __forceinline__ float fp8_to_f32(fp8_t v) {
  const float2_vt f = __builtin_amdgcn_cvt_pk_f32_fp8((unsigned int)v, false);
  return f[0];
}
