#include "hip/hip_runtime.h"
// Fused rotary position embedding (NeoX/Llama style, non-interleaved
// half-rotation) for q and k in one launch — MI355X (gfx950).
//
// Decode shape: q [T, Hq, D], k [T, Hk, D], positions [T]; D = head_dim,
// rotated pairs are (d, d + D/2). cos/sin computed on the fly from
// rope_theta (fp32) — cheaper than streaming a cache for decode batches
// and bit-matched against the fp32 torch reference in tests.
//
// Grid: one workgroup per token, 256 threads cover all heads × D/2 pairs.

#include "common.h"

__global__ void rope_kernel(
    bf16* __restrict__ q,        // [T, Hq * D]
    bf16* __restrict__ k,        // [T, Hk * D]
    const int* __restrict__ positions,  // [T]
    const int num_q_heads,
    const int num_k_heads,
    const int head_dim,
    const float theta) {
  const int token = blockIdx.x;
  const int pos = positions[token];
  const int half = head_dim / 2;
  const int total = (num_q_heads + num_k_heads) * half;

  for (int idx = threadIdx.x; idx < total; idx += blockDim.x) {
    const int head = idx / half;
    const int d = idx % half;
    const bool is_q = head < num_q_heads;
    bf16* base = is_q
        ? q + (long)token * num_q_heads * head_dim + (long)head * head_dim
        : k + (long)token * num_k_heads * head_dim +
              (long)(head - num_q_heads) * head_dim;

    const float inv_freq = __powf(theta, -2.0f * (float)d / (float)head_dim);
    const float angle = (float)pos * inv_freq;
    float c, s;
    __sincosf(angle, &s, &c);

    const float x1 = bf2f(base[d]);
    const float x2 = bf2f(base[d + half]);
    base[d] = f2bf(x1 * c - x2 * s);
    base[d + half] = f2bf(x2 * c + x1 * s);
  }
}

extern "C" void launch_rope(
    void* q, void* k, const int* positions, int tokens, int num_q_heads,
    int num_k_heads, int head_dim, float theta, hipStream_t stream) {
  dim3 grid(tokens);
  dim3 block(256);
  hipLaunchKernelGGL(rope_kernel, grid, block, 0, stream, (bf16*)q, (bf16*)k,
                     positions, num_q_heads, num_k_heads, head_dim, theta);
}
