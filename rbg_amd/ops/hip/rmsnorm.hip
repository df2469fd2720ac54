// RMSNorm kernels (gfx950): y = x * rsqrt(mean(x^2) + eps) * w, with a fused
// residual-add variant.  HBM-bound: bf16 traffic is fully vectorized as
// 16-byte lane accesses (guide Appendix B / Guideline 13); one 256-thread
// block per row, squares accumulated in f32 with a wave+LDS tree.
//
// Capability analog: the RMSNorm the reference delegates to SGLang engines
// (SURVEY §2.3 prefill/decode engine rows).
#include "common.h"

namespace {

template <bool FUSED_ADD>
__global__ void rmsnorm_kernel(__hip_bfloat16* __restrict__ out,      // [T,H]
                               __hip_bfloat16* __restrict__ input,    // [T,H]
                               __hip_bfloat16* __restrict__ residual, // [T,H]
                               const __hip_bfloat16* __restrict__ w,  // [H]
                               const float eps, const int hidden) {
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const int nthreads = blockDim.x;
  __shared__ float red[8];   // up to 8 waves
  __shared__ float s_inv;

  __hip_bfloat16* in_row = input + (size_t)row * hidden;
  __hip_bfloat16* res_row =
      FUSED_ADD ? residual + (size_t)row * hidden : nullptr;
  __hip_bfloat16* out_row = out + (size_t)row * hidden;

  const int nvec = hidden / 8;   // hidden % 8 == 0 enforced host-side
  float ss = 0.f;
  for (int i = tid; i < nvec; i += nthreads) {
    Bf16x8U xv;
    xv.u = *reinterpret_cast<const uint4*>(in_row + i * 8);
    if (FUSED_ADD) {
      Bf16x8U rv;
      rv.u = *reinterpret_cast<const uint4*>(res_row + i * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float s = bf2f(xv.e[j]) + bf2f(rv.e[j]);
        xv.e[j] = f2bf(s);
        ss += s * s;
      }
      // residual accumulates the sum (pre-norm residual stream)
      *reinterpret_cast<uint4*>(res_row + i * 8) = xv.u;
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float s = bf2f(xv.e[j]);
        ss += s * s;
      }
    }
  }
  // block reduce
  const int lane = tid & 63, wave = tid >> 6;
  ss = wave_sum(ss);
  if (lane == 0) red[wave] = ss;
  __syncthreads();
  if (tid == 0) {
    float total = 0.f;
    for (int wv = 0; wv < (nthreads >> 6); ++wv) total += red[wv];
    s_inv = rsqrtf(total / hidden + eps);
  }
  __syncthreads();
  const float inv = s_inv;

  const __hip_bfloat16* src = FUSED_ADD ? res_row : in_row;
  for (int i = tid; i < nvec; i += nthreads) {
    Bf16x8U xv, wv, ov;
    xv.u = *reinterpret_cast<const uint4*>(src + i * 8);
    wv.u = *reinterpret_cast<const uint4*>(w + i * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      ov.e[j] = f2bf(bf2f(xv.e[j]) * inv * bf2f(wv.e[j]));
    *reinterpret_cast<uint4*>(out_row + i * 8) = ov.u;
  }
}

}  // namespace

extern "C" {

void launch_rmsnorm(void* out, void* input, const void* weight, float eps,
                    int tokens, int hidden, hipStream_t stream) {
  dim3 grid(tokens), block(256);
  hipLaunchKernelGGL(rmsnorm_kernel<false>, grid, block, 0, stream,
                     (__hip_bfloat16*)out, (__hip_bfloat16*)input,
                     (__hip_bfloat16*)nullptr, (const __hip_bfloat16*)weight,
                     eps, hidden);
}

// input := rmsnorm(input + residual); residual := input + residual
void launch_fused_add_rmsnorm(void* input, void* residual, const void* weight,
                              float eps, int tokens, int hidden,
                              hipStream_t stream) {
  dim3 grid(tokens), block(256);
  hipLaunchKernelGGL(rmsnorm_kernel<true>, grid, block, 0, stream,
                     (__hip_bfloat16*)input, (__hip_bfloat16*)input,
                     (__hip_bfloat16*)residual, (const __hip_bfloat16*)weight,
                     eps, hidden);
}

}  // extern "C"
