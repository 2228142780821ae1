#include "hip/hip_runtime.h"
// Causal flash-attention forward for PREFILL — MI355X (gfx950).
//
// Shape contract: head_dim = 128, GQA G = Hq/Hk ≤ 8, one query TILE of
// 64 positions per workgroup: grid = (B·Hq, ceil(S/64)). Each of the 4
// waves OWNS 16 query rows (one 16-row m-tile), so the online softmax
// needs no cross-wave communication at all; K/V tiles are shared
// through L2 (K) and LDS (V).
//
// Per (q-tile i, kv-tile j) iteration, per wave:
//   * S_ij [16q × 64k]: 4 n-tiles × 4 k-subtiles of
//     mfma_f32_16x16x32_bf16 — A = Q fragments (persistent, loaded once
//     per workgroup), B = K fragments streamed straight from the
//     head-major cache (HBM→VGPR, the v4-scores pattern; the G-way
//     reuse across q-head workgroups is served by L2).
//   * causal mask + online softmax on the accumulator fragments: row
//     max/sum are 16-lane shuffle reductions (a C/D row lives in one
//     16-lane group, one register).
//   * P staged to LDS (bf16), V staged to LDS (cooperative, like v3/v5);
//     O_i [16q × 128d] += P·V as 2 k-subtiles × 8 dim-tiles of MFMA with
//     B = V fragments from LDS u16 reads (the v5-PV pattern).
// Causality at tile granularity: kv-tile j is processed only while
// j·64 ≤ i·64 + 63; the diagonal tile applies the per-element mask.
//
// Layouts match the decode path: q [T, Hq, 128] (T = B·S, row-major
// tokens), k/v head-major caches [B, Hk, S_max, 128] (ALREADY populated
// by the caller for positions < S), out [T, Hq, 128].
//
// Fragment conventions probe-verified in scripts/mfma_probe.hip; the
// C/D mapping is col = lane&15, row = (lane>>4)·4 + reg.

#include "common.h"

#define HEAD_DIM 128
#define QT 64              // query rows per workgroup
#define KT_POS 64          // kv positions per tile
#define NUM_WAVES 4
#define ROW_DW 66          // LDS dword stride for V rows (64 + 2 pad)
#define P_ROW 66           // LDS stride for P rows (64 positions + pad)

typedef __attribute__((ext_vector_type(8))) short bf16x8_frag;
typedef __attribute__((ext_vector_type(4))) float f32x4_frag;

__device__ __forceinline__ short pf_f2bf_bits(float v) {
  const bf16 b = f2bf(v);
  return *reinterpret_cast<const short*>(&b);
}

__device__ __forceinline__ float group16_max(float v) {
  // max across the 16-lane group holding one C/D row
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) {
    v = fmaxf(v, __shfl_xor(v, off, WAVE_SIZE));
  }
  return v;
}

__device__ __forceinline__ float group16_sum(float v) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) {
    v += __shfl_xor(v, off, WAVE_SIZE);
  }
  return v;
}

template <bool KV_FP8>
__global__ __launch_bounds__(256, 2) void prefill_attn_kernel(
    bf16* __restrict__ out,            // [T, Hq, 128]
    const bf16* __restrict__ q,        // [T, Hq, 128]
    const void* __restrict__ k_cache,  // [B, Hk, S_max, 128] bf16|e4m3
    const void* __restrict__ v_cache,
    const int B,
    const int S,
    const int num_q_heads,
    const int num_kv_heads,
    const int max_seq,
    const float scale) {
  const int bh = blockIdx.x;           // b * Hq + h
  const int b = bh / num_q_heads;
  const int h = bh % num_q_heads;
  const int kvh = h / (num_q_heads / num_kv_heads);
  const int qt0 = blockIdx.y * QT;     // first query position of the tile
  if (qt0 >= S) return;

  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;

  // V staged TRANSPOSED: [dim][pos] u16 rows (stride 68 = 64 + 4 pad:
  // 2·row mod 32 is distinct for the 16 rows of a fragment read, so the
  // b64 pairs are bank-conflict-free). A B-fragment (k = 8 consecutive
  // positions at fixed dim) is then one contiguous 16 B row segment
  // instead of 8 scattered u16 reads.
#define VT_ROW 68
  __shared__ unsigned short v_smem_t[HEAD_DIM * VT_ROW];
  __shared__ float p_smem[NUM_WAVES][16][P_ROW];

  const long slab_off = ((long)b * num_kv_heads + kvh) * max_seq * HEAD_DIM;
  const bf16* k_slab = reinterpret_cast<const bf16*>(k_cache) + slab_off;
  const fp8_t* k_slab8 = reinterpret_cast<const fp8_t*>(k_cache) + slab_off;
  const unsigned int* v_base = reinterpret_cast<const unsigned int*>(
      reinterpret_cast<const bf16*>(v_cache) + slab_off);
  const fp8_t* v_base8 = reinterpret_cast<const fp8_t*>(v_cache) + slab_off;

  // --- persistent Q fragments for this wave's 16 rows ---
  // A-frag: lane row = q-row (l%16), k = head-dim 8·(l/16)+i, 4 subtiles
  const int my_q = qt0 + 16 * wave + (lane % 16);  // global q position
  const bool q_live = my_q < S;
  bf16x8_frag q_frag[4];
  {
    const int col0 = 8 * (lane / 16);
    if (q_live) {
      const short* qrow = reinterpret_cast<const short*>(
          q + ((long)(b * S + my_q) * num_q_heads + h) * HEAD_DIM);
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) {
        q_frag[kk] =
            *reinterpret_cast<const bf16x8_frag*>(qrow + 32 * kk + col0);
      }
    } else {
#pragma unroll
      for (int kk = 0; kk < 4; ++kk)
#pragma unroll
        for (int i = 0; i < 8; ++i) q_frag[kk][i] = 0;
    }
  }

  // online-softmax state per owned q-row: the row for C/D reg i is
  // r = (lane>>4)·4 + i; every lane in a 16-lane group carries the same
  // 4 rows, so m/s are per (reg) and uniform across the group after
  // reductions.
  float m_run[4], s_run[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    m_run[i] = -INFINITY;
    s_run[i] = 0.0f;
  }
  // O accumulators: 8 dim-tiles × f32x4 (C/D row = q-row, col = dim)
  f32x4_frag o_acc[8];
#pragma unroll
  for (int n = 0; n < 8; ++n) o_acc[n] = f32x4_frag{0.f, 0.f, 0.f, 0.f};

  const int kv_end = qt0 + QT < S ? qt0 + QT : S;  // causal upper bound

  for (int kt0 = 0; kt0 < kv_end; kt0 += KT_POS) {
    const int kn = min(KT_POS, S - kt0);

    // --- stage V tile transposed (cooperative, all waves): each
    // thread reads one dim-pair at one position (coalesced along the
    // HBM row) and scatters two bf16 u16s into dim-major rows; the fp8
    // path up-converts here so downstream fragments are unchanged ---
    if constexpr (KV_FP8) {
      const fp8_t* v_src = v_base8 + (long)kt0 * HEAD_DIM;
      const int total = kn * (HEAD_DIM / 2);  // byte-pairs in the tile
      for (int idx = wave * WAVE_SIZE + lane; idx < total;
           idx += NUM_WAVES * WAVE_SIZE) {
        const int pos = idx >> 6;          // 64 pairs per position row
        const int d2 = idx & 63;
        const unsigned short packed = *reinterpret_cast<const unsigned short*>(
            v_src + (long)pos * HEAD_DIM + 2 * d2);
        const float2_vt f = unpk2_fp8(packed);
        const bf16 b0 = f2bf(f[0]);
        const bf16 b1 = f2bf(f[1]);
        v_smem_t[(2 * d2) * VT_ROW + pos] =
            *reinterpret_cast<const unsigned short*>(&b0);
        v_smem_t[(2 * d2 + 1) * VT_ROW + pos] =
            *reinterpret_cast<const unsigned short*>(&b1);
      }
    } else {
      const unsigned int* v_src = &v_base[(long)kt0 * (HEAD_DIM / 2)];
      const int total = kn * (HEAD_DIM / 2);  // dwords in the tile
      for (int idx = wave * WAVE_SIZE + lane; idx < total;
           idx += NUM_WAVES * WAVE_SIZE) {
        const int pos = idx >> 6;          // 64 dwords per position row
        const int d2 = idx & 63;           // dim pair
        const unsigned int val = v_src[idx];
        v_smem_t[(2 * d2) * VT_ROW + pos] = (unsigned short)(val & 0xffffu);
        v_smem_t[(2 * d2 + 1) * VT_ROW + pos] =
            (unsigned short)(val >> 16);
      }
    }

    // --- scores S_ij: 4 n-tiles × 4 k-subtiles ---
    f32x4_frag s_frag[4];
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) s_frag[nt] = f32x4_frag{0.f, 0.f, 0.f, 0.f};
    {
      const int col0 = 8 * (lane / 16);
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        const int kpos = kt0 + 16 * nt + (lane % 16);
        const int kpos_c = kpos < max_seq ? kpos : max_seq - 1;
#pragma unroll
        for (int kk = 0; kk < 4; ++kk) {
          bf16x8_frag k_frag;
          if constexpr (KV_FP8) {
            const unsigned short* src =
                reinterpret_cast<const unsigned short*>(
                    k_slab8 + (long)kpos_c * HEAD_DIM + 32 * kk + col0);
#pragma unroll
            for (int j = 0; j < 4; ++j) {
              const float2_vt f = unpk2_fp8(src[j]);
              k_frag[2 * j] = pf_f2bf_bits(f[0]);
              k_frag[2 * j + 1] = pf_f2bf_bits(f[1]);
            }
          } else {
            const short* krow = reinterpret_cast<const short*>(
                k_slab + (long)kpos_c * HEAD_DIM);
            k_frag =
                *reinterpret_cast<const bf16x8_frag*>(krow + 32 * kk + col0);
          }
          s_frag[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              q_frag[kk], k_frag, s_frag[nt], 0, 0, 0);
        }
      }
    }

    // --- causal mask + online softmax on the fragments ---
    // element (reg i, n-tile nt): q-row r = qt0 + 16·wave + (lane>>4)·4+i,
    // k-pos = kt0 + 16·nt + (lane&15)
    float tile_max[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) tile_max[i] = -INFINITY;
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int qr = qt0 + 16 * wave + (lane >> 4) * 4 + i;
        const int kp = kt0 + 16 * nt + (lane & 15);
        float v = s_frag[nt][i] * scale;
        if (kp > qr || kp >= kn + kt0 || qr >= S) v = -INFINITY;
        s_frag[nt][i] = v;
        tile_max[i] = fmaxf(tile_max[i], v);
      }
    }
#pragma unroll
    for (int i = 0; i < 4; ++i) tile_max[i] = group16_max(tile_max[i]);

    float corr[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const float m_new = fmaxf(m_run[i], tile_max[i]);
      corr[i] = (m_run[i] == -INFINITY) ? 0.0f : __expf(m_run[i] - m_new);
      m_run[i] = m_new;
    }
    float tile_sum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const float p = (s_frag[nt][i] == -INFINITY)
                            ? 0.0f
                            : __expf(s_frag[nt][i] - m_run[i]);
        s_frag[nt][i] = p;
        tile_sum[i] += p;
      }
    }
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      s_run[i] = s_run[i] * corr[i] + group16_sum(tile_sum[i]);
    }

    // rescale O by corr (per q-row = per reg)
#pragma unroll
    for (int n = 0; n < 8; ++n)
#pragma unroll
      for (int i = 0; i < 4; ++i) o_acc[n][i] *= corr[i];

    // --- stage P to this wave's LDS pane (bf16 would do; fp32 keeps
    // the write simple and the A-frag build cheap) ---
    // element (reg i, nt) → row (lane>>4)·4+i, col 16·nt + (lane&15)
#pragma unroll
    for (int nt = 0; nt < 4; ++nt)
#pragma unroll
      for (int i = 0; i < 4; ++i)
        p_smem[wave][(lane >> 4) * 4 + i][16 * nt + (lane & 15)] =
            s_frag[nt][i];

    __syncthreads();  // V staged + P visible (own pane only, but V needs it)

    // --- O += P·V: A = P (row = q-row, k = positions), B = V from the
    // transposed LDS tile (one contiguous row segment per fragment) ---
    {
      const bool full = (kn == KT_POS);
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {  // 2 position subtiles of 32
        bf16x8_frag p_frag;
        {
          const int row = lane % 16;
          const int p0 = 32 * kk + 8 * (lane / 16);
#pragma unroll
          for (int i = 0; i < 8; ++i)
            p_frag[i] = pf_f2bf_bits(p_smem[wave][row][p0 + i]);
        }
#pragma unroll
        for (int n = 0; n < 8; ++n) {
          const int dim = 16 * n + (lane % 16);
          const int p0 = 32 * kk + 8 * (lane / 16);
          bf16x8_frag v_frag;
          const unsigned short* vrow = &v_smem_t[dim * VT_ROW + p0];
          if (full) {
#pragma unroll
            for (int i = 0; i < 8; ++i) v_frag[i] = (short)vrow[i];
          } else {
#pragma unroll
            for (int i = 0; i < 8; ++i)
              v_frag[i] = (p0 + i < kn) ? (short)vrow[i] : (short)0;
          }
          o_acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              p_frag, v_frag, o_acc[n], 0, 0, 0);
        }
      }
    }
    __syncthreads();  // done with V before the next tile restages
  }

  // --- write O / s normalization: element (reg i, dim-tile n):
  // q-row = qt0 + 16·wave + (lane>>4)·4 + i, dim = 16·n + (lane&15) ---
#pragma unroll
  for (int n = 0; n < 8; ++n) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int qr = qt0 + 16 * wave + (lane >> 4) * 4 + i;
      if (qr >= S) continue;
      const float inv = s_run[i] > 0.0f ? 1.0f / s_run[i] : 0.0f;
      out[((long)(b * S + qr) * num_q_heads + h) * HEAD_DIM + 16 * n +
          (lane & 15)] = f2bf(o_acc[n][i] * inv);
    }
  }
}

extern "C" void launch_prefill_attn_ex(
    void* out, const void* q, const void* k_cache, const void* v_cache,
    int batch, int seq, int num_q_heads, int num_kv_heads, int max_seq,
    float scale, int kv_fp8, hipStream_t stream);

extern "C" void launch_prefill_attn(
    void* out, const void* q, const void* k_cache, const void* v_cache,
    int batch, int seq, int num_q_heads, int num_kv_heads, int max_seq,
    float scale, hipStream_t stream) {
  launch_prefill_attn_ex(out, q, k_cache, v_cache, batch, seq, num_q_heads,
                         num_kv_heads, max_seq, scale, 0, stream);
}

extern "C" void launch_prefill_attn_ex(
    void* out, const void* q, const void* k_cache, const void* v_cache,
    int batch, int seq, int num_q_heads, int num_kv_heads, int max_seq,
    float scale, int kv_fp8, hipStream_t stream) {
  dim3 grid(batch * num_q_heads, (seq + QT - 1) / QT);
  dim3 block(256);
  if (kv_fp8) {
    hipLaunchKernelGGL(prefill_attn_kernel<true>, grid, block, 0, stream,
                       (bf16*)out, (const bf16*)q, k_cache, v_cache, batch,
                       seq, num_q_heads, num_kv_heads, max_seq, scale);
  } else {
    hipLaunchKernelGGL(prefill_attn_kernel<false>, grid, block, 0, stream,
                       (bf16*)out, (const bf16*)q, k_cache, v_cache, batch,
                       seq, num_q_heads, num_kv_heads, max_seq, scale);
  }
}
