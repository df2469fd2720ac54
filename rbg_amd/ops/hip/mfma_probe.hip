// MFMA fragment-layout probe for v_mfma_f32_16x16x32_bf16 (gfx950).
//
// Computes D = A(16x32) @ B(32x16) with four candidate per-lane k-mappings
// (A contiguous-8 vs split-4+4, crossed with the same for B) so a single GPU
// run identifies the real layout empirically (guide §3: always verify the
// lane->element mapping with asymmetric operands).  The C/D mapping is taken
// as col=lane&15, row=4*(lane>>4)+reg and itself verified by the refcheck.
#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf8_t;
typedef __attribute__((ext_vector_type(4))) float f4_t;

template <int A_SPLIT, int B_SPLIT>
__global__ void mfma_probe_kernel(float* __restrict__ d,        // [16,16]
                                  const __hip_bfloat16* __restrict__ a,  // [16,32]
                                  const __hip_bfloat16* __restrict__ b) { // [32,16]
  const int lane = threadIdx.x & 63;
  const int gl = lane & 15, gs = lane >> 4;
  union { bf8_t v; __hip_bfloat16 e[8]; } af, bf;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int ka, kb;
    if (A_SPLIT) ka = (j < 4) ? (4 * gs + j) : (16 + 4 * gs + (j - 4));
    else         ka = 8 * gs + j;
    if (B_SPLIT) kb = (j < 4) ? (4 * gs + j) : (16 + 4 * gs + (j - 4));
    else         kb = 8 * gs + j;
    af.e[j] = a[gl * 32 + ka];     // A row = gl
    bf.e[j] = b[kb * 16 + gl];     // B col = gl
  }
  f4_t acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af.v, bf.v, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) d[(4 * gs + r) * 16 + gl] = acc[r];
}

// ---- v_mfma_f32_32x32x16_bf16 probe (CDNA4 shape, ROUND2 §2a) ------------
// D(32x32) = A(32x16) @ B(16x32).  Per lane: 8 bf16 of A (row = lane%32),
// 8 bf16 of B (col = lane%32), 16 f32 of D.  Candidate k-mappings as above
// (contiguous-8 vs split-4+4 per half-wave); candidate D mappings:
//   DMAP 0: row = (i%4) + 4*(lane>>5) + 8*(i/4), col = lane%32
//   DMAP 1: row = (i%4) + 8*(lane>>5) + ...  (alternate grouping)
template <int A_SPLIT, int B_SPLIT, int DMAP>
__global__ void mfma_probe32_kernel(
    float* __restrict__ d,                    // [32,32]
    const __hip_bfloat16* __restrict__ a,     // [32,16]
    const __hip_bfloat16* __restrict__ b) {   // [16,32]
  typedef __attribute__((ext_vector_type(16))) float f16_t;
  const int lane = threadIdx.x & 63;
  const int row = lane & 31, half = lane >> 5;
  union { bf8_t v; __hip_bfloat16 e[8]; } af, bf;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int ka, kb;
    if (A_SPLIT) ka = (j < 4) ? (4 * half + j) : (8 + 4 * half + (j - 4));
    else         ka = 8 * half + j;
    if (B_SPLIT) kb = (j < 4) ? (4 * half + j) : (8 + 4 * half + (j - 4));
    else         kb = 8 * half + j;
    af.e[j] = a[row * 16 + ka];
    bf.e[j] = b[kb * 32 + row];
  }
  f16_t acc;
#pragma unroll
  for (int i = 0; i < 16; ++i) acc[i] = 0.f;
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af.v, bf.v, acc, 0, 0, 0);
#pragma unroll
  for (int i = 0; i < 16; ++i) {
    int r;
    if (DMAP == 0) r = (i % 4) + 4 * half + 8 * (i / 4);
    else           r = (i % 4) + 4 * (i / 4) * 2 + 16 * half;  // alt
    d[r * 32 + row] = acc[i];
  }
}

}  // namespace

extern "C" {

void launch_mfma_probe32(void* d, const void* a, const void* b, int a_split,
                         int b_split, int dmap, hipStream_t stream) {
  dim3 grid(1), block(64);
#define CASE32(AS, BS, DM)                                                   \
  hipLaunchKernelGGL((mfma_probe32_kernel<AS, BS, DM>), grid, block, 0,      \
                     stream, (float*)d, (const __hip_bfloat16*)a,            \
                     (const __hip_bfloat16*)b)
  const int key = a_split * 4 + b_split * 2 + dmap;
  switch (key) {
    case 0: CASE32(0, 0, 0); break;
    case 1: CASE32(0, 0, 1); break;
    case 2: CASE32(0, 1, 0); break;
    case 3: CASE32(0, 1, 1); break;
    case 4: CASE32(1, 0, 0); break;
    case 5: CASE32(1, 0, 1); break;
    case 6: CASE32(1, 1, 0); break;
    case 7: CASE32(1, 1, 1); break;
  }
#undef CASE32
}

void launch_mfma_probe(void* d, const void* a, const void* b, int a_split,
                       int b_split, hipStream_t stream) {
  dim3 grid(1), block(64);
#define CASE(AS, BS)                                                       \
  hipLaunchKernelGGL((mfma_probe_kernel<AS, BS>), grid, block, 0, stream,  \
                     (float*)d, (const __hip_bfloat16*)a,                  \
                     (const __hip_bfloat16*)b)
  if (a_split == 0 && b_split == 0) CASE(0, 0);
  else if (a_split == 0 && b_split == 1) CASE(0, 1);
  else if (a_split == 1 && b_split == 0) CASE(1, 0);
  else CASE(1, 1);
#undef CASE
}

}  // extern "C"
