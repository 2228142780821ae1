// GQA decode attention (single query token per sequence) over a
// contiguous KV cache — MI355X (gfx950).
//
// Shape contract (Llama-3 family): head_dim = 128, G = Hq/Hk ≤ 8 query
// heads share one KV head. One 4-wave workgroup per (batch, kv_head);
// the G query heads of the group are processed together so K/V stream
// from HBM exactly once per group (the op is HBM-bound: 2·ctx·D·2 bytes
// per (b,kv_head)). Lanes hold 2 contiguous elements (lane l → elements
// 2l, 2l+1) so every K/V row is one coalesced 256 B wave read (dword per
// lane). Waves take interleaved position tiles; partial (max, sum, acc)
// merge across waves flash-decoding style through LDS. fp32 softmax and
// accumulation throughout.
//
// Layouts: q [B, Hq, 128], k/v cache [B, S_max, Hk, 128], out [B, Hq, 128].

#include "common.h"

#define HEAD_DIM 128
#define MAX_G 8
#define TILE 8  // positions per wave inner iteration

__global__ __launch_bounds__(256) void gqa_decode_attn_kernel(
    bf16* __restrict__ out,            // [B, Hq, 128]
    const bf16* __restrict__ q,        // [B, Hq, 128]
    const bf16* __restrict__ k_cache,  // [B, S_max, Hk, 128]
    const bf16* __restrict__ v_cache,  // [B, S_max, Hk, 128]
    const int* __restrict__ context_lens,  // [B]
    const int num_q_heads,
    const int num_kv_heads,
    const int max_seq,
    const float scale) {
  const int b = blockIdx.x;
  const int kvh = blockIdx.y;
  const int G = num_q_heads / num_kv_heads;
  const int ctx = context_lens[b];

  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  const int num_waves = blockDim.x / WAVE_SIZE;

  // Per-lane fragment of each query head in the group (2 elements).
  float qf[MAX_G][2];
  for (int g = 0; g < G; ++g) {
    const bf16x2* qrow = reinterpret_cast<const bf16x2*>(
        q + ((long)b * num_q_heads + kvh * G + g) * HEAD_DIM);
    bf16x2 v = qrow[lane];
    qf[g][0] = bf2f(v.x) * scale;
    qf[g][1] = bf2f(v.y) * scale;
  }

  float m[MAX_G], s[MAX_G], acc[MAX_G][2];
  for (int g = 0; g < G; ++g) {
    m[g] = -INFINITY;
    s[g] = 0.0f;
    acc[g][0] = acc[g][1] = 0.0f;
  }

  const long kv_row_stride = (long)num_kv_heads * HEAD_DIM;
  const bf16x2* k_base = reinterpret_cast<const bf16x2*>(
      k_cache + (long)b * max_seq * kv_row_stride + (long)kvh * HEAD_DIM);
  const bf16x2* v_base = reinterpret_cast<const bf16x2*>(
      v_cache + (long)b * max_seq * kv_row_stride + (long)kvh * HEAD_DIM);
  const long row2 = kv_row_stride / 2;  // bf16x2 stride between positions

  // Interleaved tiles: wave w takes tiles w, w+num_waves, ...
  for (int t0 = wave * TILE; t0 < ctx; t0 += num_waves * TILE) {
    const int tn = min(TILE, ctx - t0);
#pragma unroll
    for (int ti = 0; ti < TILE; ++ti) {
      if (ti >= tn) break;
      const int t = t0 + ti;
      const bf16x2 kv = k_base[(long)t * row2 + lane];
      const float k0 = bf2f(kv.x), k1 = bf2f(kv.y);
      const bf16x2 vv = v_base[(long)t * row2 + lane];
      const float v0 = bf2f(vv.x), v1 = bf2f(vv.y);
#pragma unroll
      for (int g = 0; g < MAX_G; ++g) {
        if (g >= G) break;
        float partial = fmaf(qf[g][0], k0, qf[g][1] * k1);
        const float score = wave_reduce_sum(partial);
        // online softmax update
        const float m_new = fmaxf(m[g], score);
        const float corr = __expf(m[g] - m_new);
        const float p = __expf(score - m_new);
        s[g] = s[g] * corr + p;
        acc[g][0] = fmaf(acc[g][0], corr, p * v0);
        acc[g][1] = fmaf(acc[g][1], corr, p * v1);
        m[g] = m_new;
      }
    }
  }

  // Merge partials across waves through LDS.
  // Layout per (wave, g): [m, s] scalars + 128 acc floats.
  __shared__ float lds_ms[4][MAX_G][2];
  __shared__ float lds_acc[4][MAX_G][HEAD_DIM];
  for (int g = 0; g < G; ++g) {
    if (lane == 0) {
      lds_ms[wave][g][0] = m[g];
      lds_ms[wave][g][1] = s[g];
    }
    lds_acc[wave][g][2 * lane] = acc[g][0];
    lds_acc[wave][g][2 * lane + 1] = acc[g][1];
  }
  __syncthreads();

  if (wave == 0) {
    for (int g = 0; g < G; ++g) {
      float M = -INFINITY;
      for (int w = 0; w < num_waves; ++w) M = fmaxf(M, lds_ms[w][g][0]);
      float S = 0.0f, o0 = 0.0f, o1 = 0.0f;
      for (int w = 0; w < num_waves; ++w) {
        const float mw = lds_ms[w][g][0];
        if (mw == -INFINITY) continue;
        const float f = __expf(mw - M);
        S += lds_ms[w][g][1] * f;
        o0 = fmaf(lds_acc[w][g][2 * lane], f, o0);
        o1 = fmaf(lds_acc[w][g][2 * lane + 1], f, o1);
      }
      const float inv = S > 0.0f ? 1.0f / S : 0.0f;
      bf16x2* orow = reinterpret_cast<bf16x2*>(
          out + ((long)b * num_q_heads + kvh * G + g) * HEAD_DIM);
      orow[lane] = bf16x2{f2bf(o0 * inv), f2bf(o1 * inv)};
    }
  }
}

extern "C" void launch_gqa_decode_attn(
    void* out, const void* q, const void* k_cache, const void* v_cache,
    const int* context_lens, int batch, int num_q_heads, int num_kv_heads,
    int max_seq, float scale, hipStream_t stream) {
  dim3 grid(batch, num_kv_heads);
  dim3 block(256);
  hipLaunchKernelGGL(gqa_decode_attn_kernel, grid, block, 0, stream,
                     (bf16*)out, (const bf16*)q, (const bf16*)k_cache,
                     (const bf16*)v_cache, context_lens, num_q_heads,
                     num_kv_heads, max_seq, scale);
}
