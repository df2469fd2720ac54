// Fused SwiGLU activation: out = silu(x[:, :I]) * x[:, I:].
// Pure HBM-bound elementwise op — grid-stride, 16-byte bf16 vectors
// (guide Appendix B elementwise pattern).
#include "common.h"

namespace {

__global__ void silu_mul_kernel(__hip_bfloat16* __restrict__ out,   // [T, I]
                                const __hip_bfloat16* __restrict__ x,  // [T, 2I]
                                const int inter, const long total_vec) {
  const long stride = (long)gridDim.x * blockDim.x;
  const int nvec_row = inter / 8;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total_vec;
       i += stride) {
    const long row = i / nvec_row;
    const int col = (int)(i % nvec_row) * 8;
    Bf16x8U g, u, o;
    g.u = *reinterpret_cast<const uint4*>(x + row * 2 * inter + col);
    u.u = *reinterpret_cast<const uint4*>(x + row * 2 * inter + inter + col);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float gv = bf2f(g.e[j]);
      const float silu = gv / (1.f + __expf(-gv));
      o.e[j] = f2bf(silu * bf2f(u.e[j]));
    }
    *reinterpret_cast<uint4*>(out + row * inter + col) = o.u;
  }
}

}  // namespace

extern "C" {

void launch_silu_mul(void* out, const void* x, int tokens, int inter,
                     hipStream_t stream) {
  const long total_vec = (long)tokens * (inter / 8);
  int blocks = (int)((total_vec + 255) / 256);
  if (blocks > 8192) blocks = 8192;   // grid-stride covers the rest
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(silu_mul_kernel, dim3(blocks), dim3(256), 0, stream,
                     (__hip_bfloat16*)out, (const __hip_bfloat16*)x, inter,
                     total_vec);
}

}  // extern "C"
