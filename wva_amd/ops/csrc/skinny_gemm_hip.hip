#include "hip/hip_runtime.h"
// Weight-streaming skinny GEMM for decode: C[M,N] = A[M,K] · W[N,K]^T,
// bf16 in / bf16 out, fp32 MFMA accumulation — MI355X (gfx950).
//
// Decode-step linears have M = batch ≤ 64 while W is 25-1050 MB: the op
// is pure weight streaming. hipBLASLt is near-roofline when N is large
// (its N/MT-workgroup grid fills the chip) but reaches only 1.8-2.6 TB/s
// on the small-N shapes (o-proj N=4096 → ~64 workgroups on 256 CUs);
// profiles/skinny_gemm_ab.txt. This kernel fills the chip with split-K.
//
// Structure — NO LDS, NO barriers:
//   * wave w of each workgroup owns n-rows [n_base + 16w, +16) and
//     streams its W slice straight from HBM into MFMA B fragments
//     (one b128 per lane per k-subtile). Each W byte is read exactly
//     once chip-wide; there is nothing to stage because inter-wave reuse
//     is zero by construction (an LDS-staged variant measured 1.8 TB/s —
//     the stage→barrier round trip serialized the stream; see
//     docs/mi355x-kernels.md negative results).
//   * A fragments load from global per (m-tile, k-subtile); A is ≤ 512 KB
//     and every XCD keeps it L2-resident (the only reuse, 4×, is served
//     by L2 — cheaper than an LDS round trip at this size).
//   * grid = (ceil(N/64), SK): split-K partials (fp32) to
//     workspace[SK, M, N], merged by a cast kernel. SK fills ≈1792
//     workgroups (7 WGs/CU × 256 CUs).
//
// Fragment convention: identical lane→k bijection for A and B operands
// (probe-verified sufficient, scripts/mfma_probe.hip); C/D mapping
// col = lane&15 (n), row = (lane>>4)·4 + reg (m).

#include "common.h"

#define KT 128          // K elements per unrolled block
#define NT 64           // W rows per workgroup (16 per wave)
#define NUM_WAVES 4

typedef __attribute__((ext_vector_type(8))) short bf16x8_frag;
typedef __attribute__((ext_vector_type(4))) float f32x4_frag;

// NSUB = n-subtiles (16-row groups) per wave: 1 for M <= 16, 2 for
// larger M — the second subtile reuses the same A fragments, halving
// the L2 A-read : HBM W-read ratio that bounds the M=64 case.
// W_FP8: weight-only e4m3 quantization (W8A16): B fragments up-convert
// 8 weight bytes per load (halving the HBM stream this kernel is bound
// by); per-channel scales are folded into the output store, which is
// exact because a row scale commutes with the k-sum.
template <int NSUB, bool W_FP8>
__global__ __launch_bounds__(256, 2) void skinny_gemm_kernel(
    bf16* __restrict__ c,          // [M, N] (null when split-K)
    float* __restrict__ ws,        // [SK, M, N] fp32 partials (or null)
    const bf16* __restrict__ a,    // [M, K]
    const void* __restrict__ w,    // [N, K] row-major, bf16 or e4m3
    const float* __restrict__ w_scale,  // [N] per-channel (W_FP8 only)
    const int M,
    const int N,
    const int K) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  const int sk = blockIdx.y;
  const int SK = gridDim.y;

  // this wave's NSUB×16 W rows; dead rows fold into zero fragments
  const int wg_rows = NT * NSUB;
  const int n_row0 = blockIdx.x * wg_rows + 16 * NSUB * wave + (lane % 16);
  const bf16* w_rows[NSUB];
  const fp8_t* w_rows8[NSUB];
  bool n_live[NSUB];
#pragma unroll
  for (int ns = 0; ns < NSUB; ++ns) {
    const int n_row = n_row0 + 16 * ns;
    n_live[ns] = n_row < N;
    const long off = (long)(n_live[ns] ? n_row : 0) * K;
    w_rows[ns] = reinterpret_cast<const bf16*>(w) + off;
    w_rows8[ns] = reinterpret_cast<const fp8_t*>(w) + off;
  }

  const int mt = (M + 15) / 16;          // ≤ 4 m-tiles
  const int m_row = lane % 16;           // within an m-tile
  const int col0 = 8 * (lane / 16);      // this lane's k offset in a frag

  const int k_chunks = K / KT;
  const int chunks_per_split = k_chunks / SK;
  const int k_begin = sk * chunks_per_split * KT;
  const int k_end = k_begin + chunks_per_split * KT;

  f32x4_frag acc[4][NSUB];
#pragma unroll
  for (int p = 0; p < 4; ++p)
#pragma unroll
    for (int ns = 0; ns < NSUB; ++ns)
      acc[p][ns] = f32x4_frag{0.f, 0.f, 0.f, 0.f};

#pragma unroll 2
  for (int k0 = k_begin; k0 < k_end; k0 += KT) {
#pragma unroll
    for (int kk = 0; kk < KT / 32; ++kk) {
      // lane→k bijection chosen for CONTIGUITY: lane's four k-subtile
      // windows are adjacent 8-element runs (32 B fp8 / 64 B bf16 per
      // lane per K-block), so the unrolled loads merge into wide ones.
      // Any bijection shared by A and B is exact (probe-verified).
      const int k = k0 + 4 * col0 + 8 * kk;
      bf16x8_frag b_frag[NSUB];
#pragma unroll
      for (int ns = 0; ns < NSUB; ++ns) {
        if (n_live[ns]) {
          if constexpr (W_FP8) {
            const unsigned short* src =
                reinterpret_cast<const unsigned short*>(w_rows8[ns] + k);
#pragma unroll
            for (int j = 0; j < 4; ++j) {
              const float2_vt f = unpk2_fp8(src[j]);
              const bf16 b0 = f2bf(f[0]);
              const bf16 b1 = f2bf(f[1]);
              b_frag[ns][2 * j] = *reinterpret_cast<const short*>(&b0);
              b_frag[ns][2 * j + 1] = *reinterpret_cast<const short*>(&b1);
            }
          } else {
            b_frag[ns] =
                *reinterpret_cast<const bf16x8_frag*>(w_rows[ns] + k);
          }
        } else {
#pragma unroll
          for (int i = 0; i < 8; ++i) b_frag[ns][i] = 0;
        }
      }
#pragma unroll
      for (int pm = 0; pm < 4; ++pm) {
        if (pm >= mt) break;
        const int m = pm * 16 + m_row;
        bf16x8_frag a_frag;
        if (m < M) {
          a_frag = *reinterpret_cast<const bf16x8_frag*>(a + (long)m * K + k);
        } else {
#pragma unroll
          for (int i = 0; i < 8; ++i) a_frag[i] = 0;
        }
#pragma unroll
        for (int ns = 0; ns < NSUB; ++ns) {
          acc[pm][ns] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag, b_frag[ns], acc[pm][ns], 0, 0, 0);
        }
      }
    }
  }

  // --- store: lane holds (m = (lane>>4)*4 + i, n = ... + lane&15) ---
#pragma unroll
  for (int ns = 0; ns < NSUB; ++ns) {
    const int n_out =
        blockIdx.x * wg_rows + 16 * NSUB * wave + 16 * ns + (lane & 15);
    if (n_out >= N) continue;
#pragma unroll
    for (int pm = 0; pm < 4; ++pm) {
      if (pm >= mt) break;
      const float scl = W_FP8 ? w_scale[n_out] : 1.0f;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int m = pm * 16 + (lane >> 4) * 4 + i;
        if (m >= M) continue;
        if (ws != nullptr) {
          ws[((long)sk * M + m) * N + n_out] = acc[pm][ns][i] * scl;
        } else {
          c[(long)m * N + n_out] = f2bf(acc[pm][ns][i] * scl);
        }
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

// SK heuristic: fill ≈1792 workgroups; SK must divide K/KT, power of two
// ≤ 16. `nt` is the workgroup N-tile width (64 or 128 by M).
extern "C" int skinny_gemm_num_splits(int N, int K, int nt) {
  const int n_wgs = (N + nt - 1) / nt;
  const int k_chunks = K / KT;
  int sk = 1;
  while (sk < 16 && n_wgs * sk * 2 <= 1792 && (k_chunks % (sk * 2)) == 0) {
    sk *= 2;
  }
  return sk;
}

extern "C" void launch_skinny_gemm_ex(void* c, void* ws, const void* a,
                                      const void* w, const float* w_scale,
                                      int M, int N, int K, int num_splits,
                                      int w_fp8, hipStream_t stream) {
  const int nt = (M > 16) ? 2 * NT : NT;
  dim3 grid((N + nt - 1) / nt, num_splits);
  dim3 block(256);
  float* ws_ptr = num_splits > 1 ? (float*)ws : nullptr;
  bf16* c_ptr = num_splits > 1 ? nullptr : (bf16*)c;
  if (nt == NT) {
    if (w_fp8) {
      hipLaunchKernelGGL((skinny_gemm_kernel<1, true>), grid, block, 0,
                         stream, c_ptr, ws_ptr, (const bf16*)a, w, w_scale,
                         M, N, K);
    } else {
      hipLaunchKernelGGL((skinny_gemm_kernel<1, false>), grid, block, 0,
                         stream, c_ptr, ws_ptr, (const bf16*)a, w, w_scale,
                         M, N, K);
    }
  } else {
    if (w_fp8) {
      hipLaunchKernelGGL((skinny_gemm_kernel<2, true>), grid, block, 0,
                         stream, c_ptr, ws_ptr, (const bf16*)a, w, w_scale,
                         M, N, K);
    } else {
      hipLaunchKernelGGL((skinny_gemm_kernel<2, false>), grid, block, 0,
                         stream, c_ptr, ws_ptr, (const bf16*)a, w, w_scale,
                         M, N, K);
    }
  }
  if (num_splits > 1) {
    const long MN = (long)M * N;
    dim3 mgrid((MN + 255) / 256);
    hipLaunchKernelGGL(skinny_gemm_merge_kernel, mgrid, block, 0, stream,
                       (bf16*)c, (const float*)ws, num_splits, MN);
  }
}

extern "C" void launch_skinny_gemm(void* c, void* ws, const void* a,
                                   const void* w, int M, int N, int K,
                                   int num_splits, hipStream_t stream) {
  launch_skinny_gemm_ex(c, ws, a, w, nullptr, M, N, K, num_splits, 0,
                        stream);
}

extern "C" int skinny_gemm_tile_n(int M) { return (M > 16) ? 2 * NT : NT; }
