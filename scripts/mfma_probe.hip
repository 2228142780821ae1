// MFMA fragment-layout probe for gfx950 (MI355X).
//
// Empirically verifies the lane→(row,k) mapping for
// __builtin_amdgcn_mfma_f32_16x16x32_bf16 A/B fragments by computing
// D = A·B for asymmetric integer matrices under each candidate mapping
// and comparing against a CPU reference (guide: always probe with
// asymmetric B; exact small integers are bf16-representable).
//
// Hypotheses for A (16×32, row-major semantics D[m][n] = Σk A[m][k]B[k][n]):
//   H1 (contiguous): lane l, reg i → row = l%16, k = 8·(l/16) + i
//   H2 (split-K):    lane l, reg i → row = l%16,
//                    k = 4·(l/16) + (i%4) + 16·(i/4)
// B mirrors A with col = l%16.
// C/D (documented): col = lane&15, row = (lane>>4)·4 + reg.
//
// Build & run on the GPU box:
//   hipcc --offload-arch=gfx950 scripts/mfma_probe.hip -o /tmp/mfma_probe
//   /tmp/mfma_probe

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>
#include <cstdlib>

typedef __hip_bfloat16 bf16;
typedef __attribute__((ext_vector_type(8))) short bf16x8_frag;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__device__ __forceinline__ short bf_bits(float f) {
  union { float f; unsigned int u; } c;
  c.f = f;
  return (short)(c.u >> 16);  // exact for small integers
}

// hyp: 1 or 2 — how to pack A and B fragments from row-major matrices
__global__ void probe_16x16x32(const float* A,  // [16][32] row-major
                               const float* B,  // [32][16] row-major
                               float* D,        // [16][16] row-major
                               int hyp) {
  const int lane = threadIdx.x;
  bf16x8_frag a_frag, b_frag;
  for (int i = 0; i < 8; ++i) {
    int k;
    if (hyp == 1) {
      k = 8 * (lane / 16) + i;
    } else {
      k = 4 * (lane / 16) + (i % 4) + 16 * (i / 4);
    }
    const int row = lane % 16;  // A row (m)
    const int col = lane % 16;  // B col (n)
    a_frag[i] = bf_bits(A[row * 32 + k]);
    b_frag[i] = bf_bits(B[k * 16 + col]);
  }
  f32x4 d = {0.f, 0.f, 0.f, 0.f};
  d = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, d, 0, 0, 0);
  for (int i = 0; i < 4; ++i) {
    const int col = lane & 15;
    const int row = (lane >> 4) * 4 + i;
    D[row * 16 + col] = d[i];
  }
}

// Same probe for 32x32x16: A [32][16], B [16][32], D [32][32].
// A: lane l, reg i (8 elems) → row = l%32, k = 8·(l/32)+i (H1)
//                              k = 4·(l/32) + (i%4) + 8·(i/4) (H2)
// C/D (documented): col = lane&31, row = (reg&3) + 8·(reg>>2) + 4·(lane>>5).
typedef __attribute__((ext_vector_type(16))) float f32x16;
__global__ void probe_32x32x16(const float* A, const float* B, float* D,
                               int hyp) {
  const int lane = threadIdx.x;
  bf16x8_frag a_frag, b_frag;
  for (int i = 0; i < 8; ++i) {
    int k;
    if (hyp == 1) {
      k = 8 * (lane / 32) + i;
    } else {
      k = 4 * (lane / 32) + (i % 4) + 8 * (i / 4);
    }
    a_frag[i] = bf_bits(A[(lane % 32) * 16 + k]);
    b_frag[i] = bf_bits(B[k * 32 + (lane % 32)]);
  }
  f32x16 d;
  for (int i = 0; i < 16; ++i) d[i] = 0.f;
  d = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_frag, b_frag, d, 0, 0, 0);
  for (int i = 0; i < 16; ++i) {
    const int col = lane & 31;
    const int row = (i & 3) + 8 * (i >> 2) + 4 * (lane >> 5);
    D[row * 32 + col] = d[i];
  }
}

static int check(const float* D, const float* A, const float* B, int M, int N,
                 int K) {
  int bad = 0;
  for (int m = 0; m < M; ++m)
    for (int n = 0; n < N; ++n) {
      float ref = 0.f;
      for (int k = 0; k < K; ++k) ref += A[m * K + k] * B[k * N + n];
      if (D[m * N + n] != ref && ++bad <= 3)
        printf("    D[%d][%d] = %g, want %g\n", m, n, (double)D[m * N + n],
               (double)ref);
    }
  return bad;
}

int main() {
  srand(7);
  float *A, *B, *D;
  hipMallocManaged(&A, 32 * 32 * sizeof(float));
  hipMallocManaged(&B, 32 * 32 * sizeof(float));
  hipMallocManaged(&D, 32 * 32 * sizeof(float));
  for (int i = 0; i < 32 * 32; ++i) {
    A[i] = (float)(rand() % 9 - 4);  // exact in bf16, asymmetric
    B[i] = (float)(rand() % 9 - 4);
  }
  for (int hyp = 1; hyp <= 2; ++hyp) {
    hipLaunchKernelGGL(probe_16x16x32, dim3(1), dim3(64), 0, 0, A, B, D, hyp);
    hipDeviceSynchronize();
    int bad = check(D, A, B, 16, 16, 32);
    printf("16x16x32_bf16 H%d: %s\n", hyp, bad ? "MISMATCH" : "MATCH");
  }
  for (int hyp = 1; hyp <= 2; ++hyp) {
    hipLaunchKernelGGL(probe_32x32x16, dim3(1), dim3(64), 0, 0, A, B, D, hyp);
    hipDeviceSynchronize();
    int bad = check(D, A, B, 32, 32, 16);
    printf("32x32x16_bf16 H%d: %s\n", hyp, bad ? "MISMATCH" : "MATCH");
  }
  return 0;
}
