// GQA decode attention (single query token per sequence) over a
// contiguous head-major KV cache — MI355X (gfx950).
//
// Shape contract (Llama-3 family): head_dim = 128, G = Hq/Hk ≤ 8 query
// heads share one KV head; caches [B, Hk, S_max, 128] in bf16 or fp8
// e4m3 (KV_FP8 template — up-converted at fragment build). One 4-wave
// workgroup per (batch, kv_head, split).
//
// Three kernel generations, all kept (measured dispatch in ops.cpp):
//   v3  shared-LDS-tile, VALU scores + VALU PV (head-per-wave softmax
//       ownership; regression baseline)
//   v4  MFMA scores: K streams HBM→VGPR fragments (no K staging), V via
//       LDS, VALU PV — best for G < 8 with split-KV
//   v5  v4 + MFMA PV (P cast to bf16, corr/s broadcast via LDS) — best
//       for G = 8 or unsplit grids
// plus split-KV partial/merge kernels shared by all three. The
// optimization ladder with per-step measurements (13.04 → 6.26 ms for
// the 8B decode step) and the negative results along the way are in
// docs/mi355x-kernels.md; MFMA fragment conventions were probe-verified
// on hardware (scripts/mfma_probe.hip): identical lane→k bijections for
// A and B suffice, C/D is col = lane&15, row = (lane>>4)·4 + reg.
//
// LDS notes: 66-dword row stride (64 + 2 pad) keeps every b64 access
// aligned AND bank-conflict-free; a 65-dword pad is conflict-free but
// leaves odd rows 4-byte-aligned, silently splitting vector LDS ops.
//
#include "common.h"

#define HEAD_DIM 128
#define MAX_G 8
#define TILE 64            // positions per shared tile (= wave width)
#define NUM_WAVES 4
// LDS row stride in dwords: 66 = 64 + 2 pad. Even stride keeps every
// 8-byte access provably aligned (ds_read/write_b64) and banking stays
// conflict-free: b64 groups see banks (2·row + 2·d) mod 64 — distinct
// per lane; b32 PV reads see (2·t + lane) mod 32 — distinct per lane.
// (A 65-dword pad is conflict-free too but leaves every odd row 4-byte
// aligned, so the compiler silently splits vector LDS accesses.)
#define ROW_DW 66

typedef __attribute__((ext_vector_type(4))) unsigned int uint4_t;

// Split-KV: when the (B × Hk) grid is too small to fill 256 CUs (small
// batch / TP-8 70B decode), the context is additionally partitioned across
// blockIdx.z splits; each split writes (m, s, acc[128]) partials to a
// workspace [B, Hk, G, splits, 2 + 128] and a second kernel merges.
// splits == 1 takes the direct-store fast path.
__global__ __launch_bounds__(256, 2) void gqa_decode_attn_kernel(
    bf16* __restrict__ out,            // [B, Hq, 128]
    float* __restrict__ workspace,     // [B, Hk, G, splits, 2+128] or null
    const bf16* __restrict__ q,        // [B, Hq, 128]
    const bf16* __restrict__ k_cache,  // [B, Hk, S_max, 128]
    const bf16* __restrict__ v_cache,  // [B, Hk, S_max, 128]
    const int* __restrict__ context_lens,  // [B]
    const int num_q_heads,
    const int num_kv_heads,
    const int max_seq,
    const float scale) {
  const int b = blockIdx.x;
  const int kvh = blockIdx.y;
  const int split = blockIdx.z;
  const int num_splits = gridDim.z;
  const int G = num_q_heads / num_kv_heads;
  const int ctx = context_lens[b];

  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;

  __shared__ float q_smem[MAX_G][HEAD_DIM];
  __shared__ unsigned int k_smem[TILE * ROW_DW];  // bf16x2-packed rows
  __shared__ unsigned int v_smem[TILE * ROW_DW];
  __shared__ float p_smem[MAX_G][TILE];

  // Stage scaled q into LDS (fp32).
  for (int idx = threadIdx.x; idx < G * (HEAD_DIM / 2); idx += blockDim.x) {
    const int g = idx / (HEAD_DIM / 2);
    const int d2 = idx % (HEAD_DIM / 2);
    const bf16x2 v2 = reinterpret_cast<const bf16x2*>(
        q + ((long)b * num_q_heads + kvh * G + g) * HEAD_DIM)[d2];
    q_smem[g][2 * d2] = bf2f(v2.x) * scale;
    q_smem[g][2 * d2 + 1] = bf2f(v2.y) * scale;
  }

  // Per-wave running softmax state for its owned heads (g = wave + j*4).
  const int heads_mine = (G > wave) ? (G - wave + NUM_WAVES - 1) / NUM_WAVES : 0;
  float m[2], s[2], acc[2][2];  // at most 2 heads per wave (G ≤ 8)
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    m[j] = -INFINITY;
    s[j] = 0.0f;
    acc[j][0] = acc[j][1] = 0.0f;
  }

  // head-major cache: this (b, kvh)'s positions are contiguous
  const long kv_row_dw = HEAD_DIM / 2;  // 64 dwords, rows back-to-back
  const unsigned int* k_base = reinterpret_cast<const unsigned int*>(
      k_cache + ((long)b * num_kv_heads + kvh) * max_seq * HEAD_DIM);
  const unsigned int* v_base = reinterpret_cast<const unsigned int*>(
      v_cache + ((long)b * num_kv_heads + kvh) * max_seq * HEAD_DIM);

  __syncthreads();  // q_smem visible

  // Interleave tiles across splits: split s takes tiles s, s+S, s+2S, ...
  for (int t0 = split * TILE; t0 < ctx; t0 += num_splits * TILE) {
    const int tn = min(TILE, ctx - t0);

    // --- cooperative staging over the CONTIGUOUS head-major tile:
    // the tile is tn·64 sequential dwords per cache; each wave streams
    // dwordx2 chunks (one wave instruction = 512 B sequential) ---
    {
      typedef __attribute__((ext_vector_type(2))) unsigned int uint2_t;
      const int tile_u2 = tn * (HEAD_DIM / 4);  // uint2 elements per cache
      const unsigned int* k_src = &k_base[(long)t0 * kv_row_dw];
      const unsigned int* v_src = &v_base[(long)t0 * kv_row_dw];
      for (int idx = wave * WAVE_SIZE + lane; idx < 2 * tile_u2;
           idx += NUM_WAVES * WAVE_SIZE) {
        const bool is_v = idx >= tile_u2;
        const int u2 = is_v ? idx - tile_u2 : idx;
        const int row = u2 >> 5;         // 32 uint2 per 64-dword row
        const int d2 = (u2 & 31) * 2;    // dword offset in the row
        const uint2_t val = *reinterpret_cast<const uint2_t*>(
            &(is_v ? v_src : k_src)[u2 * 2]);
        unsigned int* dst = (is_v ? v_smem : k_smem) + row * ROW_DW + d2;
        *reinterpret_cast<uint2_t*>(dst) = val;
      }
    }
    __syncthreads();

    // --- scores + softmax + p staging, per owned head ---
    const bool live = lane < tn;
    for (int j = 0; j < heads_mine; ++j) {
      const int g = wave + j * NUM_WAVES;
      float score = -INFINITY;
      if (live) {
        score = 0.0f;
        const unsigned int* krow = &k_smem[lane * ROW_DW];
        const float* qg = q_smem[g];
        typedef __attribute__((ext_vector_type(2))) unsigned int uint2_t;
#pragma unroll 8
        for (int d2 = 0; d2 < HEAD_DIM / 4; ++d2) {
          // aligned ds_read_b64: 4 bf16 K elements
          const uint2_t kv2 = *reinterpret_cast<const uint2_t*>(&krow[d2 * 2]);
          unsigned int kw[2] = {kv2[0], kv2[1]};
          float partial = 0.0f;
#pragma unroll
          for (int w2 = 0; w2 < 2; ++w2) {
            const bf16x2 p2 = *reinterpret_cast<const bf16x2*>(&kw[w2]);
            partial = fmaf(bf2f(p2.x), qg[d2 * 4 + 2 * w2], partial);
            partial = fmaf(bf2f(p2.y), qg[d2 * 4 + 2 * w2 + 1], partial);
          }
          score += partial;
        }
      }
      const float tile_max = wave_reduce_max(score);
      const float m_new = fmaxf(m[j], tile_max);
      const float p = live ? __expf(score - m_new) : 0.0f;
      const float tile_sum = wave_reduce_sum(p);
      const float corr = (m[j] == -INFINITY) ? 0.0f : __expf(m[j] - m_new);
      s[j] = s[j] * corr + tile_sum;
      acc[j][0] *= corr;
      acc[j][1] *= corr;
      m[j] = m_new;
      p_smem[g][lane] = p;  // lane ↔ position, in-bounds (TILE == 64)
    }
    // Waves read only their OWN heads' p rows and their own K/V-view; the
    // barrier below orders the NEXT tile's staging against this compute.

    // --- PV: lane ↔ output dword, sweep tile positions ---
    for (int j = 0; j < heads_mine; ++j) {
      const int g = wave + j * NUM_WAVES;
      const float* pg = p_smem[g];
      float a0 = acc[j][0], a1 = acc[j][1];
      for (int t = 0; t < tn; ++t) {
        const unsigned int vw = v_smem[t * ROW_DW + lane];
        const bf16x2 vv = *reinterpret_cast<const bf16x2*>(&vw);
        const float p = pg[t];  // broadcast
        a0 = fmaf(p, bf2f(vv.x), a0);
        a1 = fmaf(p, bf2f(vv.y), a1);
      }
      acc[j][0] = a0;
      acc[j][1] = a1;
    }
    __syncthreads();  // compute done before next tile overwrites LDS
  }

  // --- finalize ---
  if (num_splits == 1) {
    // fast path: owning wave writes its heads directly
    for (int j = 0; j < heads_mine; ++j) {
      const int g = wave + j * NUM_WAVES;
      const float inv = s[j] > 0.0f ? 1.0f / s[j] : 0.0f;
      bf16x2* orow = reinterpret_cast<bf16x2*>(
          out + ((long)b * num_q_heads + kvh * G + g) * HEAD_DIM);
      orow[lane] = bf16x2{f2bf(acc[j][0] * inv), f2bf(acc[j][1] * inv)};
    }
  } else {
    // partials → workspace [B, Hk, G, splits, 2+128]
    for (int j = 0; j < heads_mine; ++j) {
      const int g = wave + j * NUM_WAVES;
      float* wsp = workspace +
          ((((long)b * num_kv_heads + kvh) * G + g) * num_splits + split) *
              (2 + HEAD_DIM);
      if (lane == 0) {
        wsp[0] = m[j];
        wsp[1] = s[j];
      }
      wsp[2 + 2 * lane] = acc[j][0];
      wsp[2 + 2 * lane + 1] = acc[j][1];
    }
  }
}

// ---------------------------------------------------------------------------
// v4: MFMA scores. Differences from v3:
//   * K is NOT staged through LDS at all: each wave loads its 16 tile
//     positions straight from HBM into mfma_f32_16x16x32_bf16 A
//     fragments (dwordx4 per lane per k-subtile) and computes the
//     64-position × 16-head score tile in 4 MFMAs on the matrix cores.
//     Scores land in LDS (s_smem) for the cross-wave softmax; only V
//     still goes through LDS (its position-major PV sweep needs the
//     transpose LDS provides). LDS drops ~17 KiB → higher residency.
//   * A/B fragments use the same lane→k bijection for K and Q, which is
//     sufficient for correctness (the k-sum is permutation-invariant; the
//     C/D mapping col=lane&15, row=(lane>>4)·4+reg was probe-verified on
//     MI355X — scripts/mfma_probe.hip).
//   * Softmax state, p staging and the vector PV sweep are v3 verbatim.
// ---------------------------------------------------------------------------

typedef __attribute__((ext_vector_type(8))) short bf16x8_frag;
typedef __attribute__((ext_vector_type(4))) float f32x4_frag;

// fp32 -> bf16 raw bits (round-to-nearest-even via hardware convert)
__device__ __forceinline__ short f2bf_bits(float v) {
  const bf16 b = f2bf(v);
  return *reinterpret_cast<const short*>(&b);
}

// Pack 8 floats (scaled into e4m3 range) into the fp8 MFMA operand.
__device__ __forceinline__ long pack8_fp8(const float* v, float inv_scale) {
  unsigned int lo = 0, hi = 0;
  lo = __builtin_amdgcn_cvt_pk_fp8_f32(v[0] * inv_scale, v[1] * inv_scale,
                                       lo, false);
  lo = __builtin_amdgcn_cvt_pk_fp8_f32(v[2] * inv_scale, v[3] * inv_scale,
                                       lo, true);
  hi = __builtin_amdgcn_cvt_pk_fp8_f32(v[4] * inv_scale, v[5] * inv_scale,
                                       hi, false);
  hi = __builtin_amdgcn_cvt_pk_fp8_f32(v[6] * inv_scale, v[7] * inv_scale,
                                       hi, true);
  return (long)(((unsigned long)hi << 32) | lo);
}

#define S_ROW 65  // s_smem row stride (floats): 64 + 1 pad

template <bool KV_FP8>
__global__ __launch_bounds__(256, 2) void gqa_decode_attn_v4_kernel(
    bf16* __restrict__ out,
    float* __restrict__ workspace,
    const bf16* __restrict__ q,
    const void* __restrict__ k_cache,  // bf16 or e4m3 (KV_FP8)
    const void* __restrict__ v_cache,
    const int* __restrict__ context_lens,
    const int num_q_heads,
    const int num_kv_heads,
    const int max_seq,
    const float scale) {
  const int b = blockIdx.x;
  const int kvh = blockIdx.y;
  const int split = blockIdx.z;
  const int num_splits = gridDim.z;
  const int G = num_q_heads / num_kv_heads;
  const int ctx = context_lens[b];

  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;

  __shared__ float s_smem[MAX_G][S_ROW];      // score tile [head][pos]
  __shared__ unsigned int v_smem[TILE * ROW_DW];
  __shared__ float p_smem[MAX_G][TILE];

  // --- Q fragments (persistent): B operand, col = head = lane%16,
  // k = 32·kk + 8·(lane/16) + i. Heads ≥ G are zero. In fp8-KV mode Q
  // is quantized ONCE per workgroup to e4m3 (s_q folded into the
  // softmax scale) so scores run on the fp8 MFMA with K bytes fed RAW —
  // no per-fragment up-conversion on the hot path. ---
  bf16x8_frag q_frag[4];
  long q8_frag[4] = {0, 0, 0, 0};
  float q_scale = 1.0f;
  {
    const int g = lane % 16;
    const int k0 = 8 * (lane / 16);
    float qv[4][8];
#pragma unroll
    for (int kk = 0; kk < 4; ++kk)
#pragma unroll
      for (int i = 0; i < 8; ++i) qv[kk][i] = 0.0f;
    if (g < G) {
      const bf16* qrow = q + ((long)b * num_q_heads + kvh * G + g) * HEAD_DIM;
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) {
        const short* src = reinterpret_cast<const short*>(qrow + 32 * kk + k0);
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          q_frag[kk][i] = src[i];
          const bf16 bv = *reinterpret_cast<const bf16*>(&src[i]);
          qv[kk][i] = bf2f(bv);
        }
      }
    } else {
#pragma unroll
      for (int kk = 0; kk < 4; ++kk)
#pragma unroll
        for (int i = 0; i < 8; ++i) q_frag[kk][i] = 0;
    }
    if constexpr (KV_FP8) {
      float amax = 0.0f;
#pragma unroll
      for (int kk = 0; kk < 4; ++kk)
#pragma unroll
        for (int i = 0; i < 8; ++i) amax = fmaxf(amax, fabsf(qv[kk][i]));
      amax = wave_reduce_max(amax);
      // cross-wave max via LDS (p_smem pane reused pre-loop)
      __shared__ float qmax_smem[NUM_WAVES];
      if (lane == 0) qmax_smem[wave] = amax;
      __syncthreads();
      float m0 = qmax_smem[0];
#pragma unroll
      for (int w2 = 1; w2 < NUM_WAVES; ++w2) m0 = fmaxf(m0, qmax_smem[w2]);
      q_scale = fmaxf(m0 / 448.0f, 1e-12f);
      const float inv = 1.0f / q_scale;
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) q8_frag[kk] = pack8_fp8(qv[kk], inv);
      __syncthreads();  // qmax_smem consumed before any reuse
    }
  }

  const int heads_mine = (G > wave) ? (G - wave + NUM_WAVES - 1) / NUM_WAVES : 0;
  float m[2], s[2], acc[2][2];
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    m[j] = -INFINITY;
    s[j] = 0.0f;
    acc[j][0] = acc[j][1] = 0.0f;
  }

  const long kv_row_dw = HEAD_DIM / 2;
  const long slab_off = ((long)b * num_kv_heads + kvh) * max_seq * HEAD_DIM;
  const bf16* k_slab = reinterpret_cast<const bf16*>(k_cache) + slab_off;
  const fp8_t* k_slab8 = reinterpret_cast<const fp8_t*>(k_cache) + slab_off;
  const unsigned int* v_base = reinterpret_cast<const unsigned int*>(
      reinterpret_cast<const bf16*>(v_cache) + slab_off);
  const fp8_t* v_base8 = reinterpret_cast<const fp8_t*>(v_cache) + slab_off;

  for (int t0 = split * TILE; t0 < ctx; t0 += num_splits * TILE) {
    const int tn = min(TILE, ctx - t0);

    // --- stage V only; the fp8 path up-converts during staging so the
    // LDS layout and PV sweep are identical in both modes ---
    if constexpr (KV_FP8) {
      const int tile_g8 = tn * (HEAD_DIM / 8);
      const fp8_t* v_src = v_base8 + (long)t0 * HEAD_DIM;
      for (int idx = wave * WAVE_SIZE + lane; idx < tile_g8;
           idx += NUM_WAVES * WAVE_SIZE) {
        const int row = idx >> 4;          // 16 groups of 8 per row
        const int d8 = (idx & 15) * 8;
        const unsigned short* src =
            reinterpret_cast<const unsigned short*>(
                v_src + (long)row * HEAD_DIM + d8);
        unsigned int* dst = &v_smem[row * ROW_DW + d8 / 2];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const float2_vt f = unpk2_fp8(src[j]);
          const bf16x2 packed{f2bf(f[0]), f2bf(f[1])};
          dst[j] = *reinterpret_cast<const unsigned int*>(&packed);
        }
      }
    } else {
      typedef __attribute__((ext_vector_type(2))) unsigned int uint2_t;
      const int tile_u2 = tn * (HEAD_DIM / 4);
      const unsigned int* v_src = &v_base[(long)t0 * kv_row_dw];
      for (int idx = wave * WAVE_SIZE + lane; idx < tile_u2;
           idx += NUM_WAVES * WAVE_SIZE) {
        const int row = idx >> 5;
        const int d2 = (idx & 31) * 2;
        const uint2_t val = *reinterpret_cast<const uint2_t*>(&v_src[idx * 2]);
        *reinterpret_cast<uint2_t*>(&v_smem[row * ROW_DW + d2]) = val;
      }
    }

    // --- MFMA scores: wave w owns tile positions 16w..16w+15 ---
    {
      const int prow = 16 * wave + (lane % 16);       // A row = position
      const int pos = t0 + prow;
      const int pos_c = pos < max_seq ? pos : max_seq - 1;  // clamped load
      const int k0 = 8 * (lane / 16);
      f32x4_frag sf = {0.f, 0.f, 0.f, 0.f};
      if constexpr (KV_FP8) {
        // fp8×fp8 MFMA: K bytes fed raw (8 per lane, one b64 load);
        // Q was pre-quantized with scale q_scale
        const long* ksrc = reinterpret_cast<const long*>(
            k_slab8 + (long)pos_c * HEAD_DIM);
        const int l0 = k0 / 8;  // which 8-byte group this lane reads
#pragma unroll
        for (int kk = 0; kk < 4; ++kk) {
          const long a8 = ksrc[kk * 4 + l0];
          sf = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
              a8, q8_frag[kk], sf, 0, 0, 0);
        }
      } else {
#pragma unroll
        for (int kk = 0; kk < 4; ++kk) {
          bf16x8_frag a_frag;
          const short* src = reinterpret_cast<const short*>(
              k_slab + (long)pos_c * HEAD_DIM) + 32 * kk + k0;
#pragma unroll
          for (int i = 0; i < 8; ++i) a_frag[i] = src[i];
          sf = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, q_frag[kk],
                                                       sf, 0, 0, 0);
        }
      }
      // C/D: col = lane&15 (head), row = (lane>>4)·4 + i (position);
      // fp8 path folds the Q quantization scale back in here
      const float s_eff = KV_FP8 ? scale * q_scale : scale;
      const int g = lane & 15;
      if (g < G) {
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          const int t = 16 * wave + (lane >> 4) * 4 + i;
          s_smem[g][t] = (t < tn) ? sf[i] * s_eff : -INFINITY;
        }
      }
    }
    __syncthreads();  // V + scores visible

    // --- softmax per owned head (v3, scores from s_smem) ---
    const bool live = lane < tn;
    for (int j = 0; j < heads_mine; ++j) {
      const int g = wave + j * NUM_WAVES;
      const float score = live ? s_smem[g][lane] : -INFINITY;
      const float tile_max = wave_reduce_max(score);
      const float m_new = fmaxf(m[j], tile_max);
      const float p = live ? __expf(score - m_new) : 0.0f;
      const float tile_sum = wave_reduce_sum(p);
      const float corr = (m[j] == -INFINITY) ? 0.0f : __expf(m[j] - m_new);
      s[j] = s[j] * corr + tile_sum;
      acc[j][0] *= corr;
      acc[j][1] *= corr;
      m[j] = m_new;
      p_smem[g][lane] = p;
    }

    // --- PV (v3 verbatim) ---
    for (int j = 0; j < heads_mine; ++j) {
      const int g = wave + j * NUM_WAVES;
      const float* pg = p_smem[g];
      float a0 = acc[j][0], a1 = acc[j][1];
      for (int t = 0; t < tn; ++t) {
        const unsigned int vw = v_smem[t * ROW_DW + lane];
        const bf16x2 vv = *reinterpret_cast<const bf16x2*>(&vw);
        const float p = pg[t];
        a0 = fmaf(p, bf2f(vv.x), a0);
        a1 = fmaf(p, bf2f(vv.y), a1);
      }
      acc[j][0] = a0;
      acc[j][1] = a1;
    }
    __syncthreads();
  }

  if (num_splits == 1) {
    for (int j = 0; j < heads_mine; ++j) {
      const int g = wave + j * NUM_WAVES;
      const float inv = s[j] > 0.0f ? 1.0f / s[j] : 0.0f;
      bf16x2* orow = reinterpret_cast<bf16x2*>(
          out + ((long)b * num_q_heads + kvh * G + g) * HEAD_DIM);
      orow[lane] = bf16x2{f2bf(acc[j][0] * inv), f2bf(acc[j][1] * inv)};
    }
  } else {
    for (int j = 0; j < heads_mine; ++j) {
      const int g = wave + j * NUM_WAVES;
      float* wsp = workspace +
          ((((long)b * num_kv_heads + kvh) * G + g) * num_splits + split) *
              (2 + HEAD_DIM);
      if (lane == 0) {
        wsp[0] = m[j];
        wsp[1] = s[j];
      }
      wsp[2 + 2 * lane] = acc[j][0];
      wsp[2 + 2 * lane + 1] = acc[j][1];
    }
  }
}

// ---------------------------------------------------------------------------
// v5: v4 + MFMA PV. PMC counters showed v4 issue-bound on VALU (4.07 G
// VALU vs 30 M MFMA instructions — the vector PV sweep, 64 positions ×
// 2 fma × heads per wave per tile). v5 computes O += P·V on the matrix
// cores too:
//   * P [16 heads × 64 pos] (zero-padded heads) as A fragments from
//     p_smem, cast fp32→bf16 at fragment build (FA2-standard precision);
//   * V as B fragments straight from v_smem u16 reads (no conversion —
//     V is bf16 in LDS);
//   * wave w owns head_dim n-subtiles {w, w+4} → 2 persistent f32x4
//     accumulators; online-softmax rescale multiplies each accumulator
//     register by corr[head(reg)] from LDS before the tile's 4 MFMAs.
// Needs one extra barrier per tile (every wave now reads every head's
// p row) and LDS broadcast of corr/s; softmax state stays wave-owned.
// ---------------------------------------------------------------------------

template <bool KV_FP8>
__global__ __launch_bounds__(256, 2) void gqa_decode_attn_v5_kernel(
    bf16* __restrict__ out,
    float* __restrict__ workspace,
    const bf16* __restrict__ q,
    const void* __restrict__ k_cache,  // bf16 or e4m3 (KV_FP8)
    const void* __restrict__ v_cache,
    const int* __restrict__ context_lens,
    const int num_q_heads,
    const int num_kv_heads,
    const int max_seq,
    const float scale) {
  const int b = blockIdx.x;
  const int kvh = blockIdx.y;
  const int split = blockIdx.z;
  const int num_splits = gridDim.z;
  const int G = num_q_heads / num_kv_heads;
  const int ctx = context_lens[b];

  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;

  __shared__ float s_smem[MAX_G][S_ROW];
  __shared__ unsigned int v_smem[TILE * ROW_DW];
  __shared__ float p_smem[MAX_G][TILE];
  __shared__ float corr_smem[MAX_G];
  __shared__ float sum_smem[MAX_G];
  __shared__ float m_smem[MAX_G];

  // Q fragments (same as v4; fp8 mode pre-quantizes once per WG)
  bf16x8_frag q_frag[4];
  long q8_frag[4] = {0, 0, 0, 0};
  float q_scale = 1.0f;
  {
    const int g = lane % 16;
    const int k0 = 8 * (lane / 16);
    float qv[4][8];
#pragma unroll
    for (int kk = 0; kk < 4; ++kk)
#pragma unroll
      for (int i = 0; i < 8; ++i) qv[kk][i] = 0.0f;
    if (g < G) {
      const bf16* qrow = q + ((long)b * num_q_heads + kvh * G + g) * HEAD_DIM;
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) {
        const short* src = reinterpret_cast<const short*>(qrow + 32 * kk + k0);
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          q_frag[kk][i] = src[i];
          const bf16 bv = *reinterpret_cast<const bf16*>(&src[i]);
          qv[kk][i] = bf2f(bv);
        }
      }
    } else {
#pragma unroll
      for (int kk = 0; kk < 4; ++kk)
#pragma unroll
        for (int i = 0; i < 8; ++i) q_frag[kk][i] = 0;
    }
    if constexpr (KV_FP8) {
      float amax = 0.0f;
#pragma unroll
      for (int kk = 0; kk < 4; ++kk)
#pragma unroll
        for (int i = 0; i < 8; ++i) amax = fmaxf(amax, fabsf(qv[kk][i]));
      amax = wave_reduce_max(amax);
      __shared__ float qmax_smem5[NUM_WAVES];
      if (lane == 0) qmax_smem5[wave] = amax;
      __syncthreads();
      float m0 = qmax_smem5[0];
#pragma unroll
      for (int w2 = 1; w2 < NUM_WAVES; ++w2) m0 = fmaxf(m0, qmax_smem5[w2]);
      q_scale = fmaxf(m0 / 448.0f, 1e-12f);
      const float inv = 1.0f / q_scale;
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) q8_frag[kk] = pack8_fp8(qv[kk], inv);
      __syncthreads();
    }
  }

  const int heads_mine = (G > wave) ? (G - wave + NUM_WAVES - 1) / NUM_WAVES : 0;
  float m[2], s[2];
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    m[j] = -INFINITY;
    s[j] = 0.0f;
  }
  // PV accumulators: wave owns n-subtiles wave and wave+4 of head_dim;
  // lane holds (head = (lane>>4)*4 + reg, dim = 16*nsub + lane&15).
  f32x4_frag acc_pv[2] = {f32x4_frag{0.f, 0.f, 0.f, 0.f},
                          f32x4_frag{0.f, 0.f, 0.f, 0.f}};

  const long kv_row_dw = HEAD_DIM / 2;
  const long slab_off = ((long)b * num_kv_heads + kvh) * max_seq * HEAD_DIM;
  const bf16* k_slab = reinterpret_cast<const bf16*>(k_cache) + slab_off;
  const fp8_t* k_slab8 = reinterpret_cast<const fp8_t*>(k_cache) + slab_off;
  const unsigned int* v_base = reinterpret_cast<const unsigned int*>(
      reinterpret_cast<const bf16*>(v_cache) + slab_off);
  const fp8_t* v_base8 = reinterpret_cast<const fp8_t*>(v_cache) + slab_off;

  for (int t0 = split * TILE; t0 < ctx; t0 += num_splits * TILE) {
    const int tn = min(TILE, ctx - t0);

    // --- stage V (fp8 up-converts during staging; LDS layout fixed) ---
    if constexpr (KV_FP8) {
      const int tile_g8 = tn * (HEAD_DIM / 8);
      const fp8_t* v_src = v_base8 + (long)t0 * HEAD_DIM;
      for (int idx = wave * WAVE_SIZE + lane; idx < tile_g8;
           idx += NUM_WAVES * WAVE_SIZE) {
        const int row = idx >> 4;
        const int d8 = (idx & 15) * 8;
        const unsigned short* src =
            reinterpret_cast<const unsigned short*>(
                v_src + (long)row * HEAD_DIM + d8);
        unsigned int* dst = &v_smem[row * ROW_DW + d8 / 2];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const float2_vt f = unpk2_fp8(src[j]);
          const bf16x2 packed{f2bf(f[0]), f2bf(f[1])};
          dst[j] = *reinterpret_cast<const unsigned int*>(&packed);
        }
      }
    } else {
      typedef __attribute__((ext_vector_type(2))) unsigned int uint2_t;
      const int tile_u2 = tn * (HEAD_DIM / 4);
      const unsigned int* v_src = &v_base[(long)t0 * kv_row_dw];
      for (int idx = wave * WAVE_SIZE + lane; idx < tile_u2;
           idx += NUM_WAVES * WAVE_SIZE) {
        const int row = idx >> 5;
        const int d2 = (idx & 31) * 2;
        const uint2_t val = *reinterpret_cast<const uint2_t*>(&v_src[idx * 2]);
        *reinterpret_cast<uint2_t*>(&v_smem[row * ROW_DW + d2]) = val;
      }
    }

    // --- MFMA scores (v4) ---
    {
      const int prow = 16 * wave + (lane % 16);
      const int pos = t0 + prow;
      const int pos_c = pos < max_seq ? pos : max_seq - 1;
      const int k0 = 8 * (lane / 16);
      f32x4_frag sf = {0.f, 0.f, 0.f, 0.f};
      if constexpr (KV_FP8) {
        const long* ksrc = reinterpret_cast<const long*>(
            k_slab8 + (long)pos_c * HEAD_DIM);
        const int l0 = k0 / 8;
#pragma unroll
        for (int kk = 0; kk < 4; ++kk) {
          const long a8 = ksrc[kk * 4 + l0];
          sf = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
              a8, q8_frag[kk], sf, 0, 0, 0);
        }
      } else {
#pragma unroll
        for (int kk = 0; kk < 4; ++kk) {
          bf16x8_frag a_frag;
          const short* src = reinterpret_cast<const short*>(
              k_slab + (long)pos_c * HEAD_DIM) + 32 * kk + k0;
#pragma unroll
          for (int i = 0; i < 8; ++i) a_frag[i] = src[i];
          sf = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, q_frag[kk],
                                                       sf, 0, 0, 0);
        }
      }
      const float s_eff = KV_FP8 ? scale * q_scale : scale;
      const int g = lane & 15;
      if (g < G) {
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          const int t = 16 * wave + (lane >> 4) * 4 + i;
          s_smem[g][t] = (t < tn) ? sf[i] * s_eff : -INFINITY;
        }
      }
    }
    __syncthreads();  // V + scores visible

    // --- softmax per owned head (v4) + publish p, corr ---
    const bool live = lane < tn;
    for (int j = 0; j < heads_mine; ++j) {
      const int g = wave + j * NUM_WAVES;
      const float score = live ? s_smem[g][lane] : -INFINITY;
      const float tile_max = wave_reduce_max(score);
      const float m_new = fmaxf(m[j], tile_max);
      const float p = live ? __expf(score - m_new) : 0.0f;
      const float tile_sum = wave_reduce_sum(p);
      const float corr = (m[j] == -INFINITY) ? 0.0f : __expf(m[j] - m_new);
      s[j] = s[j] * corr + tile_sum;
      m[j] = m_new;
      p_smem[g][lane] = p;
      if (lane == 0) corr_smem[g] = corr;
    }
    // dead heads (g >= G) keep corr undefined; PV zero-pads their rows
    __syncthreads();  // p + corr visible to all waves

    // --- MFMA PV: O[16 heads × dims] += P[16×64] · V[64×dims] ---
    {
      // A fragments (P): shared by this wave's two n-subtiles
      bf16x8_frag p_frag[2];
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        const int g = lane % 16;
        const int pos0 = 32 * kk + 8 * (lane / 16);
        if (g < G) {
#pragma unroll
          for (int i = 0; i < 8; ++i)
            p_frag[kk][i] = f2bf_bits(p_smem[g][pos0 + i]);
        } else {
#pragma unroll
          for (int i = 0; i < 8; ++i) p_frag[kk][i] = 0;
        }
      }
      const unsigned short* v_u16 =
          reinterpret_cast<const unsigned short*>(v_smem);
#pragma unroll
      for (int nsub = 0; nsub < 2; ++nsub) {
        const int dim = 16 * (wave + 4 * nsub) + (lane % 16);
        // rescale accumulator by corr[head(reg)]
        f32x4_frag d = acc_pv[nsub];
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          const int h = (lane >> 4) * 4 + i;
          d[i] *= (h < G) ? corr_smem[h] : 0.0f;
        }
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
          bf16x8_frag v_frag;
          const int pos0 = 32 * kk + 8 * (lane / 16);
#pragma unroll
          for (int i = 0; i < 8; ++i) {
            const int t = pos0 + i;
            // t >= tn rows are unstaged (stale LDS can decode to NaN,
            // and 0·NaN = NaN even though p[t] = 0) — zero them
            v_frag[i] = (t < tn)
                ? (short)v_u16[(t * ROW_DW + (dim >> 1)) * 2 + (dim & 1)]
                : (short)0;
          }
          d = __builtin_amdgcn_mfma_f32_16x16x32_bf16(p_frag[kk], v_frag, d,
                                                      0, 0, 0);
        }
        acc_pv[nsub] = d;
      }
    }
    __syncthreads();  // PV consumed V/p before next tile overwrites
  }

  // publish per-head softmax sums (and maxima for the split path)
  for (int j = 0; j < heads_mine; ++j) {
    const int g = wave + j * NUM_WAVES;
    if (lane == 0) {
      sum_smem[g] = s[j];
      m_smem[g] = m[j];
    }
  }
  __syncthreads();

  if (num_splits == 1) {
    // lane holds (head = (lane>>4)*4+reg, dim = 16*(wave+4*nsub)+lane&15)
#pragma unroll
    for (int nsub = 0; nsub < 2; ++nsub) {
      const int dim = 16 * (wave + 4 * nsub) + (lane & 15);
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int h = (lane >> 4) * 4 + i;
        if (h >= G) continue;
        const float sv = sum_smem[h];
        const float inv = sv > 0.0f ? 1.0f / sv : 0.0f;
        out[((long)b * num_q_heads + kvh * G + h) * HEAD_DIM + dim] =
            f2bf(acc_pv[nsub][i] * inv);
      }
    }
  } else {
#pragma unroll
    for (int nsub = 0; nsub < 2; ++nsub) {
      const int dim = 16 * (wave + 4 * nsub) + (lane & 15);
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int h = (lane >> 4) * 4 + i;
        if (h >= G) continue;
        float* wsp = workspace +
            ((((long)b * num_kv_heads + kvh) * G + h) * num_splits + split) *
                (2 + HEAD_DIM);
        if (wave == 0 && nsub == 0 && (lane & 15) == 0) {
          wsp[0] = m_smem[h];
          wsp[1] = sum_smem[h];
        }
        wsp[2 + dim] = acc_pv[nsub][i];
      }
    }
  }
}

// Merge split-KV partials: one 64-thread wave per (b, q_head).
// 4 waves per (b, h) row: each wave reduces a strided subset of splits
// into a partial (m, s, acc[128]), then wave 0 combines the 4 partials
// through LDS. A single-wave merge looping over ~224 splits serially
// was HALF the long-context step time (profiles/longctx_kernel_stats
// .txt pre-fix: 101 µs vs the 115 µs attention sweep).
#define MERGE_WAVES 8
__global__ __launch_bounds__(64 * MERGE_WAVES) void gqa_decode_attn_merge_kernel(
    bf16* __restrict__ out,          // [B, Hq, 128]
    const float* __restrict__ workspace,  // [B, Hk, G, splits, 2+128]
    const int num_q_heads,
    const int num_kv_heads,
    const int num_splits) {
  const int b = blockIdx.x;
  const int h = blockIdx.y;  // query head
  const int G = num_q_heads / num_kv_heads;
  const int kvh = h / G;
  const int g = h % G;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;

  __shared__ float sm_m[MERGE_WAVES];
  __shared__ float sm_s[MERGE_WAVES];
  __shared__ float sm_o[MERGE_WAVES][HEAD_DIM];

  const float* base = workspace +
      ((((long)b * num_kv_heads + kvh) * G + g) * (long)num_splits) *
          (2 + HEAD_DIM);
  // wave-local pass over its strided split subset
  float M = -INFINITY;
  for (int sp = wave; sp < num_splits; sp += MERGE_WAVES) {
    M = fmaxf(M, base[sp * (2 + HEAD_DIM)]);
  }
  float S = 0.0f, o0 = 0.0f, o1 = 0.0f;
  for (int sp = wave; sp < num_splits; sp += MERGE_WAVES) {
    const float* wsp = base + sp * (2 + HEAD_DIM);
    const float mw = wsp[0];
    if (mw == -INFINITY) continue;
    const float f = __expf(mw - M);
    S += wsp[1] * f;
    o0 = fmaf(wsp[2 + 2 * lane], f, o0);
    o1 = fmaf(wsp[2 + 2 * lane + 1], f, o1);
  }
  if (lane == 0) {
    sm_m[wave] = M;
    sm_s[wave] = S;
  }
  sm_o[wave][2 * lane] = o0;
  sm_o[wave][2 * lane + 1] = o1;
  __syncthreads();

  if (wave != 0) return;
  // combine the wave partials (online-softmax merge across <=4 terms)
  float Mg = -INFINITY;
  for (int w = 0; w < MERGE_WAVES; ++w) Mg = fmaxf(Mg, sm_m[w]);
  float Sg = 0.0f;
  float g0 = 0.0f, g1 = 0.0f;
  for (int w = 0; w < MERGE_WAVES; ++w) {
    if (sm_m[w] == -INFINITY) continue;
    const float f = __expf(sm_m[w] - Mg);
    Sg += sm_s[w] * f;
    g0 = fmaf(sm_o[w][2 * lane], f, g0);
    g1 = fmaf(sm_o[w][2 * lane + 1], f, g1);
  }
  const float inv = Sg > 0.0f ? 1.0f / Sg : 0.0f;
  bf16x2* orow = reinterpret_cast<bf16x2*>(
      out + ((long)b * num_q_heads + h) * HEAD_DIM);
  orow[lane] = bf16x2{f2bf(g0 * inv), f2bf(g1 * inv)};
}

extern "C" void launch_gqa_decode_attn_v4_ex(
    void* out, void* workspace, const void* q, const void* k_cache,
    const void* v_cache, const int* context_lens, int batch, int num_q_heads,
    int num_kv_heads, int max_seq, int num_splits, float scale, int kv_fp8,
    hipStream_t stream);
extern "C" void launch_gqa_decode_attn_v5_ex(
    void* out, void* workspace, const void* q, const void* k_cache,
    const void* v_cache, const int* context_lens, int batch, int num_q_heads,
    int num_kv_heads, int max_seq, int num_splits, float scale, int kv_fp8,
    hipStream_t stream);

extern "C" int gqa_decode_attn_num_splits(int batch, int num_kv_heads,
                                          int max_ctx_hint) {
  // Fill the v4 kernel's resident workgroups: 70 VGPRs / ~21 KiB LDS
  // admit 7 WGs/CU (compiler occupancy report) → target ~1792 WGs over
  // 256 CUs; cap so each split still gets >= 2 tiles of work.
  const int base = batch * num_kv_heads;
  if (base >= 1792) return 1;
  int splits = 1792 / base;
  // Split cap: 16 for moderate contexts (diminishing returns once each
  // split's tile count is small), but LONG contexts lift it — at B=1
  // ctx=128k the 16-split grid is 128 WGs on 256 CUs (7%% of resident
  // capacity) and the KV sweep ran at 0.43 TB/s (profiles/
  // longctx_itl.json, pre-fix). Allow as many splits as keep >= 8
  // tiles of work per split; the merge kernel's cost is O(splits) per
  // (b,h) row and stays negligible.
  static int min_tiles = [] {
    const char* e = getenv("WVA_ATTN_MIN_TILES");  // tuning knob
    int v = e ? atoi(e) : 8;
    return v > 0 ? v : 8;
  }();
  int cap = 16;
  if (max_ctx_hint > 0) {
    const int by_work = max_ctx_hint / (min_tiles * TILE);
    if (by_work > cap) cap = by_work;
  }
  if (splits > cap) splits = cap;
  const int max_useful = max_ctx_hint > 0 ? (max_ctx_hint + 2 * TILE - 1) / (2 * TILE) : splits;
  if (splits > max_useful && max_useful >= 1) splits = max_useful;
  return splits < 1 ? 1 : splits;
}

extern "C" void launch_gqa_decode_attn(
    void* out, void* workspace, const void* q, const void* k_cache,
    const void* v_cache, const int* context_lens, int batch, int num_q_heads,
    int num_kv_heads, int max_seq, int num_splits, float scale,
    hipStream_t stream) {
  dim3 grid(batch, num_kv_heads, num_splits);
  dim3 block(256);
  hipLaunchKernelGGL(gqa_decode_attn_kernel, grid, block, 0, stream,
                     (bf16*)out, (float*)workspace, (const bf16*)q,
                     (const bf16*)k_cache, (const bf16*)v_cache, context_lens,
                     num_q_heads, num_kv_heads, max_seq, scale);
  if (num_splits > 1) {
    dim3 mgrid(batch, num_q_heads);
    hipLaunchKernelGGL(gqa_decode_attn_merge_kernel, mgrid, dim3(64 * MERGE_WAVES), 0,
                       stream, (bf16*)out, (const float*)workspace,
                       num_q_heads, num_kv_heads, num_splits);
  }
}

extern "C" void launch_gqa_decode_attn_v5(
    void* out, void* workspace, const void* q, const void* k_cache,
    const void* v_cache, const int* context_lens, int batch, int num_q_heads,
    int num_kv_heads, int max_seq, int num_splits, float scale,
    hipStream_t stream) {
  launch_gqa_decode_attn_v5_ex(out, workspace, q, k_cache, v_cache,
                               context_lens, batch, num_q_heads, num_kv_heads,
                               max_seq, num_splits, scale, 0, stream);
}

extern "C" void launch_gqa_decode_attn_v5_ex(
    void* out, void* workspace, const void* q, const void* k_cache,
    const void* v_cache, const int* context_lens, int batch, int num_q_heads,
    int num_kv_heads, int max_seq, int num_splits, float scale, int kv_fp8,
    hipStream_t stream) {
  dim3 grid(batch, num_kv_heads, num_splits);
  dim3 block(256);
  if (kv_fp8) {
    hipLaunchKernelGGL(gqa_decode_attn_v5_kernel<true>, grid, block, 0,
                       stream, (bf16*)out, (float*)workspace, (const bf16*)q,
                       k_cache, v_cache, context_lens, num_q_heads,
                       num_kv_heads, max_seq, scale);
  } else {
    hipLaunchKernelGGL(gqa_decode_attn_v5_kernel<false>, grid, block, 0,
                       stream, (bf16*)out, (float*)workspace, (const bf16*)q,
                       k_cache, v_cache, context_lens, num_q_heads,
                       num_kv_heads, max_seq, scale);
  }
  if (num_splits > 1) {
    dim3 mgrid(batch, num_q_heads);
    hipLaunchKernelGGL(gqa_decode_attn_merge_kernel, mgrid, dim3(64 * MERGE_WAVES), 0,
                       stream, (bf16*)out, (const float*)workspace,
                       num_q_heads, num_kv_heads, num_splits);
  }
}

extern "C" void launch_gqa_decode_attn_v4(
    void* out, void* workspace, const void* q, const void* k_cache,
    const void* v_cache, const int* context_lens, int batch, int num_q_heads,
    int num_kv_heads, int max_seq, int num_splits, float scale,
    hipStream_t stream) {
  launch_gqa_decode_attn_v4_ex(out, workspace, q, k_cache, v_cache,
                               context_lens, batch, num_q_heads, num_kv_heads,
                               max_seq, num_splits, scale, 0, stream);
}

extern "C" void launch_gqa_decode_attn_v4_ex(
    void* out, void* workspace, const void* q, const void* k_cache,
    const void* v_cache, const int* context_lens, int batch, int num_q_heads,
    int num_kv_heads, int max_seq, int num_splits, float scale, int kv_fp8,
    hipStream_t stream) {
  dim3 grid(batch, num_kv_heads, num_splits);
  dim3 block(256);
  if (kv_fp8) {
    hipLaunchKernelGGL(gqa_decode_attn_v4_kernel<true>, grid, block, 0,
                       stream, (bf16*)out, (float*)workspace, (const bf16*)q,
                       k_cache, v_cache, context_lens, num_q_heads,
                       num_kv_heads, max_seq, scale);
  } else {
    hipLaunchKernelGGL(gqa_decode_attn_v4_kernel<false>, grid, block, 0,
                       stream, (bf16*)out, (float*)workspace, (const bf16*)q,
                       k_cache, v_cache, context_lens, num_q_heads,
                       num_kv_heads, max_seq, scale);
  }
  if (num_splits > 1) {
    dim3 mgrid(batch, num_q_heads);
    hipLaunchKernelGGL(gqa_decode_attn_merge_kernel, mgrid, dim3(64 * MERGE_WAVES), 0,
                       stream, (bf16*)out, (const float*)workspace,
                       num_q_heads, num_kv_heads, num_splits);
  }
}
