// Fused SwiGLU activation: out = silu(gate) * up — MI355X (gfx950).
//
// Pure bandwidth op: 2 reads + 1 write of [rows, inter]. 16 B/lane
// vectorized; flat 1-D grid sized ≫ 256 workgroups so all 8 XCDs fill.

#include "common.h"

__global__ void silu_mul_kernel(
    bf16* __restrict__ out,
    const bf16* __restrict__ gate,
    const bf16* __restrict__ up,
    const long n2) {  // number of bf16x2 elements
  const bf16x2* g2 = reinterpret_cast<const bf16x2*>(gate);
  const bf16x2* u2 = reinterpret_cast<const bf16x2*>(up);
  bf16x2* o2 = reinterpret_cast<bf16x2*>(out);
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n2;
       i += stride) {
    bf16x2 g = g2[i];
    bf16x2 u = u2[i];
    const float glo = bf2f(g.x), ghi = bf2f(g.y);
    // silu(x) = x / (1 + e^-x); fp32 math, fast exp2-based expf
    const float slo = glo / (1.0f + __expf(-glo));
    const float shi = ghi / (1.0f + __expf(-ghi));
    o2[i] = bf16x2{f2bf(slo * bf2f(u.x)), f2bf(shi * bf2f(u.y))};
  }
}

// Row-strided variant: gate and up are the two halves of the fused
// gate_up GEMM output [rows, 2·inter] — reads them in place (no
// .contiguous() slice copies in the decode layer).
__global__ void silu_mul_fused_kernel(
    bf16* __restrict__ out,            // [rows, inter]
    const bf16* __restrict__ gate_up,  // [rows, 2 * inter]
    const long rows,
    const long inter2) {  // inter in bf16x2 units
  const long n2 = rows * inter2;
  const long stride = (long)gridDim.x * blockDim.x;
  bf16x2* o2 = reinterpret_cast<bf16x2*>(out);
  const bf16x2* gu2 = reinterpret_cast<const bf16x2*>(gate_up);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n2;
       i += stride) {
    const long row = i / inter2;
    const long col = i % inter2;
    const bf16x2 g = gu2[row * 2 * inter2 + col];
    const bf16x2 u = gu2[row * 2 * inter2 + inter2 + col];
    const float glo = bf2f(g.x), ghi = bf2f(g.y);
    const float slo = glo / (1.0f + __expf(-glo));
    const float shi = ghi / (1.0f + __expf(-ghi));
    o2[i] = bf16x2{f2bf(slo * bf2f(u.x)), f2bf(shi * bf2f(u.y))};
  }
}

static int silu_grid(long n2) {
  long want = (n2 + 255) / 256;
  return (int)(want > 8192 ? 8192 : (want < 1 ? 1 : want));
}

extern "C" void launch_silu_mul(
    void* out, const void* gate, const void* up, long n, hipStream_t stream) {
  const long n2 = n / 2;
  // ≫256 workgroups to fill 256 CUs / 8 XCDs; cap to keep grid sane.
  hipLaunchKernelGGL(silu_mul_kernel, dim3(silu_grid(n2)), dim3(256), 0,
                     stream, (bf16*)out, (const bf16*)gate, (const bf16*)up,
                     n2);
}

extern "C" void launch_silu_mul_fused(
    void* out, const void* gate_up, long rows, long inter, hipStream_t stream) {
  const long inter2 = inter / 2;
  hipLaunchKernelGGL(silu_mul_fused_kernel, dim3(silu_grid(rows * inter2)),
                     dim3(256), 0, stream, (bf16*)out, (const bf16*)gate_up,
                     rows, inter2);
}
