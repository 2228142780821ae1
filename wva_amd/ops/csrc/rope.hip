// Fused rotary position embedding (NeoX/Llama style, non-interleaved
// half-rotation) — MI355X (gfx950).
//
// Two entry points:
//  * rope_kernel: in-place RoPE on contiguous q [T,Hq,D] / k [T,Hk,D]
//    (kept for tests and generic use).
//  * rope_append_kv_kernel: the decode fast path. Reads the UN-SLICED
//    qkv GEMM output row [B, (Hq+2Hk)·D] directly (no .contiguous()
//    copies), applies RoPE to q and k, writes q to a contiguous output
//    and APPENDS rotated k and raw v into the KV cache at each sequence's
//    current position — replacing 2 slice-copies + rope + 2 cache
//    index-writes per layer with one launch
//    (profiles/: ~13% of decode step time was this glue).
//
// Grid: one workgroup per token; 256 threads cover heads × D/2 pairs.

#include "common.h"

__device__ __forceinline__ void rotate_pair(
    const bf16* src, bf16* dst, int d, int half, int head_dim, float pos,
    float theta) {
  const float inv_freq = __powf(theta, -2.0f * (float)d / (float)head_dim);
  const float angle = pos * inv_freq;
  float c, s;
  __sincosf(angle, &s, &c);
  const float x1 = bf2f(src[d]);
  const float x2 = bf2f(src[d + half]);
  dst[d] = f2bf(x1 * c - x2 * s);
  dst[d + half] = f2bf(x2 * c + x1 * s);
}

__global__ void rope_kernel(
    bf16* __restrict__ q,        // [T, Hq * D]
    bf16* __restrict__ k,        // [T, Hk * D]
    const int* __restrict__ positions,  // [T]
    const int num_q_heads,
    const int num_k_heads,
    const int head_dim,
    const float theta) {
  const int token = blockIdx.x;
  const float pos = (float)positions[token];
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
    rotate_pair(base, base, d, half, head_dim, pos, theta);
  }
}

__global__ void rope_append_kv_kernel(
    const bf16* __restrict__ qkv,   // [B, (Hq + 2·Hk) · D]
    bf16* __restrict__ q_out,       // [B, Hq, D]
    bf16* __restrict__ k_cache,     // [B, Hk, S_max, D] (head-major)
    bf16* __restrict__ v_cache,     // [B, Hk, S_max, D]
    const int* __restrict__ positions,  // [B] append position per sequence
    const int num_q_heads,
    const int num_kv_heads,
    const int head_dim,
    const int max_seq,
    const float theta) {
  const int b = blockIdx.x;
  const int pos_i = positions[b];
  const float pos = (float)pos_i;
  const int half = head_dim / 2;
  const long qkv_row = (long)b * (num_q_heads + 2 * num_kv_heads) * head_dim;

  // q heads + k heads: rotate; v heads: straight copy (by 2-elem pairs)
  const int rot_total = (num_q_heads + num_kv_heads) * half;
  for (int idx = threadIdx.x; idx < rot_total; idx += blockDim.x) {
    const int head = idx / half;
    const int d = idx % half;
    if (head < num_q_heads) {
      const bf16* src = qkv + qkv_row + (long)head * head_dim;
      bf16* dst = q_out + ((long)b * num_q_heads + head) * head_dim;
      rotate_pair(src, dst, d, half, head_dim, pos, theta);
    } else {
      const int kh = head - num_q_heads;
      const bf16* src = qkv + qkv_row + ((long)num_q_heads + kh) * head_dim;
      bf16* dst = k_cache +
          (((long)b * num_kv_heads + kh) * max_seq + pos_i) * head_dim;
      rotate_pair(src, dst, d, half, head_dim, pos, theta);
    }
  }
  // v copy: vectorized bf16x2
  const int v_total = num_kv_heads * half;
  const bf16x2* vsrc = reinterpret_cast<const bf16x2*>(
      qkv + qkv_row + (long)(num_q_heads + num_kv_heads) * head_dim);
  for (int idx = threadIdx.x; idx < v_total; idx += blockDim.x) {
    const int kh = idx / half;
    const int d2 = idx % half;
    bf16x2* dst = reinterpret_cast<bf16x2*>(
        v_cache + (((long)b * num_kv_heads + kh) * max_seq + pos_i) * head_dim);
    dst[d2] = vsrc[(long)kh * half + d2];
  }
}

// fp8 (e4m3) cache variant of the decode append: q stays bf16; rotated
// k and copied v are down-converted to 1-byte e4m3 on store (half the
// cache bytes, double the KV capacity; see common.h fp8 helpers).
__global__ void rope_append_kv_fp8_kernel(
    const bf16* __restrict__ qkv,   // [B, (Hq + 2·Hk) · D]
    bf16* __restrict__ q_out,       // [B, Hq, D]
    fp8_t* __restrict__ k_cache,    // [B, Hk, S_max, D] e4m3
    fp8_t* __restrict__ v_cache,    // [B, Hk, S_max, D]
    const int* __restrict__ positions,
    const int num_q_heads,
    const int num_kv_heads,
    const int head_dim,
    const int max_seq,
    const float theta) {
  const int b = blockIdx.x;
  const int pos_i = positions[b];
  const float pos = (float)pos_i;
  const int half = head_dim / 2;
  const long qkv_row = (long)b * (num_q_heads + 2 * num_kv_heads) * head_dim;

  const int rot_total = (num_q_heads + num_kv_heads) * half;
  for (int idx = threadIdx.x; idx < rot_total; idx += blockDim.x) {
    const int head = idx / half;
    const int d = idx % half;
    if (head < num_q_heads) {
      const bf16* src = qkv + qkv_row + (long)head * head_dim;
      bf16* dst = q_out + ((long)b * num_q_heads + head) * head_dim;
      rotate_pair(src, dst, d, half, head_dim, pos, theta);
    } else {
      const int kh = head - num_q_heads;
      const bf16* src = qkv + qkv_row + ((long)num_q_heads + kh) * head_dim;
      const float inv_freq =
          __powf(theta, -2.0f * (float)d / (float)head_dim);
      const float angle = pos * inv_freq;
      float c, s;
      __sincosf(angle, &s, &c);
      const float x1 = bf2f(src[d]);
      const float x2 = bf2f(src[d + half]);
      fp8_t* dst = k_cache +
          (((long)b * num_kv_heads + kh) * max_seq + pos_i) * head_dim;
      dst[d] = f2fp8(x1 * c - x2 * s);
      dst[d + half] = f2fp8(x2 * c + x1 * s);
    }
  }
  // v copy: bf16x2 -> packed 2×e4m3 (adjacent dims), u16 stores
  const int v_total = num_kv_heads * half;
  const bf16x2* vsrc = reinterpret_cast<const bf16x2*>(
      qkv + qkv_row + (long)(num_q_heads + num_kv_heads) * head_dim);
  for (int idx = threadIdx.x; idx < v_total; idx += blockDim.x) {
    const int kh = idx / half;
    const int d2 = idx % half;
    const bf16x2 v2 = vsrc[(long)kh * half + d2];
    unsigned short* dst = reinterpret_cast<unsigned short*>(
        v_cache + (((long)b * num_kv_heads + kh) * max_seq + pos_i) * head_dim);
    dst[d2] = pk2_fp8(bf2f(v2.x), bf2f(v2.y));
  }
}

extern "C" void launch_rope_append_kv_fp8(
    const void* qkv, void* q_out, void* k_cache, void* v_cache,
    const int* positions, int batch, int num_q_heads, int num_kv_heads,
    int head_dim, int max_seq, float theta, hipStream_t stream) {
  dim3 grid(batch);
  dim3 block(256);
  hipLaunchKernelGGL(rope_append_kv_fp8_kernel, grid, block, 0, stream,
                     (const bf16*)qkv, (bf16*)q_out, (fp8_t*)k_cache,
                     (fp8_t*)v_cache, positions, num_q_heads, num_kv_heads,
                     head_dim, max_seq, theta);
}

extern "C" void launch_rope(
    void* q, void* k, const int* positions, int tokens, int num_q_heads,
    int num_k_heads, int head_dim, float theta, hipStream_t stream) {
  dim3 grid(tokens);
  dim3 block(256);
  hipLaunchKernelGGL(rope_kernel, grid, block, 0, stream, (bf16*)q, (bf16*)k,
                     positions, num_q_heads, num_k_heads, head_dim, theta);
}

extern "C" void launch_rope_append_kv(
    const void* qkv, void* q_out, void* k_cache, void* v_cache,
    const int* positions, int batch, int num_q_heads, int num_kv_heads,
    int head_dim, int max_seq, float theta, hipStream_t stream) {
  dim3 grid(batch);
  dim3 block(256);
  hipLaunchKernelGGL(rope_append_kv_kernel, grid, block, 0, stream,
                     (const bf16*)qkv, (bf16*)q_out, (bf16*)k_cache,
                     (bf16*)v_cache, positions, num_q_heads, num_kv_heads,
                     head_dim, max_seq, theta);
}
