#include "hip/hip_runtime.h"
// Weight-streaming skinny GEMM for decode: C[M,N] = A[M,K] · W[N,K]^T,
// bf16 in / bf16 out, fp32 MFMA accumulation — MI355X (gfx950).
//
// Decode-step linears have M = batch ≤ 64 while W is 25-235 MB: the op
// is pure weight streaming (arith intensity < 1 FLOP/byte of W), yet
// hipBLASLt's tile kernels measure only 2.3-4.6 TB/s on these shapes
// (profiles/decode8b_v4_breakdown.txt). This kernel streams each W row
// exactly once at HBM rate:
//   * grid tiles N (NT rows of W per workgroup) × split-K; every
//     workgroup streams its W[NT, K/SK] slice ONCE through LDS
//     (double-buffered KT-chunks) and re-reads A from L2 (A is ≤ 512 KB
//     — resident after the first workgroups touch it).
//   * compute: mfma_f32_16x16x32_bf16; wave-level (m-tile, n-tile)
//     pairs; A fragments straight from global (L2), W fragments from
//     LDS as aligned b128 reads (66-dword row pad — same provably
//     conflict-free layout as attention.hip).
//   * split-K (small N): fp32 partials to workspace[SK, M, N]; a tiny
//     merge kernel reduces and casts. SK chosen so the grid fills the
//     7-WG/CU residency (≈1792 workgroups).
//
// Fragment convention: identical lane→k bijection for A and B operands
// (probe-verified sufficient, scripts/mfma_probe.hip); C/D mapping
// col = lane&15 (n), row = (lane>>4)·4 + reg (m).

#include "common.h"

#define KT 128          // K elements per staged chunk (64 dwords)
#define W_ROW_DW 66     // LDS row stride in dwords: 64 + 2 pad
#define MAX_MT 4        // M ≤ 64 → ≤ 4 m-tiles of 16
#define NUM_WAVES 4

typedef __attribute__((ext_vector_type(8))) short bf16x8_frag;
typedef __attribute__((ext_vector_type(4))) float f32x4_frag;
typedef __attribute__((ext_vector_type(4))) unsigned int uint4_vec;

// NT = W rows per workgroup (32 for M>16, 64 for M<=16 so every wave has
// (m,n) pairs to own). 256 threads.
template <int NT>
__global__ __launch_bounds__(256, 2) void skinny_gemm_kernel(
    bf16* __restrict__ c,          // [M, N] (null when split-K)
    float* __restrict__ ws,        // [SK, M, N] fp32 partials (or null)
    const bf16* __restrict__ a,    // [M, K]
    const bf16* __restrict__ w,    // [N, K] row-major (torch linear weight)
    const int M,
    const int N,
    const int K) {
  const int n_base = blockIdx.x * NT;
  const int sk = blockIdx.y;          // split-K index
  const int SK = gridDim.y;
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;

  // this split's K range (multiple of KT by launcher contract)
  const int k_chunks_total = K / KT;
  const int chunks_per_split = k_chunks_total / SK;
  const int k_begin = sk * chunks_per_split * KT;

  __shared__ unsigned int w_smem[2][NT * W_ROW_DW];

  // (m-tile, n-tile) pair ownership: pairs = mt × (NT/16), wave takes
  // every 4th. With NT=32: M=64 → 8 pairs, 2/wave; NT=64: M≤16 → 4
  // pairs, 1/wave.
  const int mt = (M + 15) / 16;
  const int nt = NT / 16;
  const int num_pairs = mt * nt;
  // each wave owns at most 4 pairs (16 accumulator VGPRs)
  f32x4_frag acc[4];
#pragma unroll
  for (int p = 0; p < 4; ++p) acc[p] = f32x4_frag{0.f, 0.f, 0.f, 0.f};

  // --- stage one KT chunk of W into LDS buffer `buf` ---
  // chunk layout: w_smem[buf][row * W_ROW_DW + d], row ∈ [0, NT),
  // d ∈ [0, 64) dwords (= 128 bf16 of that W row's chunk).
  // uint2 (8 B) granularity: every row offset (66 dwords = 264 B) is
  // 8 B-aligned, so both the global loads and the LDS stores are true
  // b64 ops on every row (a 16 B granule would be misaligned on odd
  // rows and silently split).
  typedef __attribute__((ext_vector_type(2))) unsigned int uint2_vec;
  auto stage = [&](int buf, int k0) {
    const int total_u2 = NT * 32;  // 32 uint2 per 64-dword row
    for (int idx = threadIdx.x; idx < total_u2; idx += 256) {
      const int row = idx / 32;
      const int d2 = idx % 32;
      const int n = n_base + row;
      uint2_vec val = {0u, 0u};
      if (n < N) {
        val = *reinterpret_cast<const uint2_vec*>(
            w + (long)n * K + k0 + d2 * 4);
      }
      *reinterpret_cast<uint2_vec*>(&w_smem[buf][row * W_ROW_DW + d2 * 2]) =
          val;
    }
  };

  stage(0, k_begin);
  __syncthreads();

  for (int ch = 0; ch < chunks_per_split; ++ch) {
    const int k0 = k_begin + ch * KT;
    const int cur = ch & 1;
    // prefetch next chunk into the other buffer (no barrier yet: the
    // writes target the buffer nobody reads this iteration)
    if (ch + 1 < chunks_per_split) stage(cur ^ 1, k0 + KT);

    // --- compute on the current chunk ---
    // All of a wave's pairs share the same n-tile (pn = (wave + 4j) % nt
    // is constant: nt | 4), so B loads hoist out of the pair loop; A
    // fragments are single aligned b128 global loads (L2-resident), B
    // fragments two b64 LDS reads (odd 66-dword rows are 8 B-aligned).
    const int a_col0 = 8 * (lane / 16);  // k offset of this lane's frag
    const int pn_w = wave % nt;
    const int b_row = pn_w * 16 + (lane % 16);
    const unsigned int* b_base = &w_smem[cur][b_row * W_ROW_DW];
#pragma unroll
    for (int kk = 0; kk < KT / 32; ++kk) {
      bf16x8_frag b_frag;
      {
        typedef __attribute__((ext_vector_type(2))) unsigned int u2;
        const u2* src = reinterpret_cast<const u2*>(
            &b_base[kk * 16 + a_col0 / 2]);
        const u2 lo = src[0];
        const u2 hi = src[1];
        unsigned int words[4] = {lo[0], lo[1], hi[0], hi[1]};
        b_frag = *reinterpret_cast<const bf16x8_frag*>(words);
      }
      for (int p = wave, slot = 0; p < num_pairs && slot < 4;
           p += NUM_WAVES, ++slot) {
        const int pm = p / nt;
        bf16x8_frag a_frag;
        const int m = pm * 16 + (lane % 16);
        if (m < M) {
          a_frag = *reinterpret_cast<const bf16x8_frag*>(
              a + (long)m * K + k0 + kk * 32 + a_col0);
        } else {
#pragma unroll
          for (int i = 0; i < 8; ++i) a_frag[i] = 0;
        }
        acc[slot] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag, b_frag, acc[slot], 0, 0, 0);
      }
    }
    __syncthreads();  // everyone done with `cur` before it is re-staged
  }

  // --- store ---
  for (int p = wave, slot = 0; p < num_pairs && slot < 4;
       p += NUM_WAVES, ++slot) {
    const int pm = p / nt;
    const int pn = p % nt;
    const int n = n_base + pn * 16 + (lane & 15);
    if (n >= N) continue;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int m = pm * 16 + (lane >> 4) * 4 + i;
      if (m >= M) continue;
      if (ws != nullptr) {
        ws[((long)sk * M + m) * N + n] = acc[slot][i];
      } else {
        c[(long)m * N + n] = f2bf(acc[slot][i]);
      }
    }
  }
}

// Reduce split-K partials and cast: C[m][n] = bf16(Σ_sk ws[sk][m][n]).
__global__ __launch_bounds__(256) void skinny_gemm_merge_kernel(
    bf16* __restrict__ c, const float* __restrict__ ws, const int SK,
    const long MN) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= MN) return;
  float acc = 0.f;
  for (int sk = 0; sk < SK; ++sk) acc += ws[sk * MN + i];
  c[i] = f2bf(acc);
}

// SK heuristic: fill ≈1792 workgroups; SK must divide K/KT and is a
// power of two ≤ 8.
extern "C" int skinny_gemm_num_splits(int N, int K, int nt) {
  const int n_wgs = (N + nt - 1) / nt;
  const int k_chunks = K / KT;
  int sk = 1;
  while (sk < 8 && n_wgs * sk * 2 <= 1792 && (k_chunks % (sk * 2)) == 0) {
    sk *= 2;
  }
  return sk;
}

extern "C" void launch_skinny_gemm(void* c, void* ws, const void* a,
                                   const void* w, int M, int N, int K,
                                   int num_splits, hipStream_t stream) {
  const int nt = (M > 16) ? 32 : 64;
  dim3 grid((N + nt - 1) / nt, num_splits);
  dim3 block(256);
  float* ws_ptr = num_splits > 1 ? (float*)ws : nullptr;
  bf16* c_ptr = num_splits > 1 ? nullptr : (bf16*)c;
  if (nt == 32) {
    hipLaunchKernelGGL(skinny_gemm_kernel<32>, grid, block, 0, stream, c_ptr,
                       ws_ptr, (const bf16*)a, (const bf16*)w, M, N, K);
  } else {
    hipLaunchKernelGGL(skinny_gemm_kernel<64>, grid, block, 0, stream, c_ptr,
                       ws_ptr, (const bf16*)a, (const bf16*)w, M, N, K);
  }
  if (num_splits > 1) {
    const long MN = (long)M * N;
    dim3 mgrid((MN + 255) / 256);
    hipLaunchKernelGGL(skinny_gemm_merge_kernel, mgrid, block, 0, stream,
                       (bf16*)c, (const float*)ws, num_splits, MN);
  }
}

extern "C" int skinny_gemm_tile_n(int M) { return (M > 16) ? 32 : 64; }
