// Fused RoPE + paged-KV write (gfx950).
//
// One kernel applies rotary embedding (Llama rotate-half / NeoX style) to Q
// and K in place AND scatters the rotated K and V rows into the paged KV
// pool — the write is fused into the producing kernel so K/V never make an
// extra HBM round trip (guide: fuse elementwise work into the producer).
// cos/sin come from a host-precomputed table (guide Appendix B: on-device
// sinf/cosf turns this memory-bound op VALU-bound).
//
// Capability analog: RoPE + KV-cache insert of the engines the reference
// orchestrates.
#include "common.h"

namespace {

// grid.x = tokens; block = 256.  Each block rotates all Q heads + K heads of
// one token and writes K/V to cache[slot].
__global__ void rope_store_kv_kernel(
    __hip_bfloat16* __restrict__ q,          // [T, QH*D]
    __hip_bfloat16* __restrict__ k,          // [T, KVH*D]
    const __hip_bfloat16* __restrict__ v,    // [T, KVH*D]
    __hip_bfloat16* __restrict__ key_cache,  // [pages, KVH, page, D]
    __hip_bfloat16* __restrict__ val_cache,  // [pages, KVH, page, D]
    const float* __restrict__ cos_sin,       // [max_pos, D] = [cos(D/2)|sin(D/2)]
    const int* __restrict__ positions,       // [T]
    const int* __restrict__ slot_mapping,    // [T] page*page_size+off; -1 skip
    const int num_q_heads, const int num_kv_heads, const int head_dim,
    const int page_size,
    const int q_stride, const int kv_stride) { // token-row strides (elems):
  // q/k/v may be SLICES of the fused QKV projection (row pitch = the full
  // qkv width) — strided access here removes three .contiguous() copies
  // per layer (profiles: ~73 ms/bench of pure copy kernels)
  const int token = blockIdx.x;
  const int tid = threadIdx.x;
  const int pos = positions[token];
  const int slot = slot_mapping[token];
  const int half = head_dim / 2;
  const float* cs = cos_sin + (size_t)pos * head_dim;

  // ---- rotate Q: num_q_heads * half pairs -------------------------------
  __hip_bfloat16* q_tok = q + (size_t)token * q_stride;
  const int q_pairs = num_q_heads * half;
  for (int i = tid; i < q_pairs; i += blockDim.x) {
    const int h = i / half, d = i % half;
    __hip_bfloat16* base = q_tok + h * head_dim;
    const float x1 = bf2f(base[d]), x2 = bf2f(base[d + half]);
    const float c = cs[d], s = cs[half + d];
    base[d] = f2bf(x1 * c - x2 * s);
    base[d + half] = f2bf(x2 * c + x1 * s);
  }

  // ---- rotate K and scatter K,V into the paged pool ---------------------
  __hip_bfloat16* k_tok = k + (size_t)token * kv_stride;
  const __hip_bfloat16* v_tok = v + (size_t)token * kv_stride;
  const int page = slot >= 0 ? slot / page_size : 0;
  const int poff = slot >= 0 ? slot % page_size : 0;
  const int kv_pairs = num_kv_heads * half;
  for (int i = tid; i < kv_pairs; i += blockDim.x) {
    const int h = i / half, d = i % half;
    __hip_bfloat16* base = k_tok + h * head_dim;
    const float x1 = bf2f(base[d]), x2 = bf2f(base[d + half]);
    const float c = cs[d], s = cs[half + d];
    const __hip_bfloat16 r1 = f2bf(x1 * c - x2 * s);
    const __hip_bfloat16 r2 = f2bf(x2 * c + x1 * s);
    base[d] = r1;
    base[d + half] = r2;
    if (slot >= 0) {
      __hip_bfloat16* kdst = key_cache +
          (((size_t)page * num_kv_heads + h) * page_size + poff) * head_dim;
      kdst[d] = r1;
      kdst[d + half] = r2;
    }
  }
  if (slot >= 0) {
    // V: straight copy, vectorized 16B per lane
    const int nvec = num_kv_heads * head_dim / 8;
    for (int i = tid; i < nvec; i += blockDim.x) {
      const int h = (i * 8) / head_dim;
      const int d = (i * 8) % head_dim;
      __hip_bfloat16* vdst = val_cache +
          (((size_t)page * num_kv_heads + h) * page_size + poff) * head_dim;
      *reinterpret_cast<uint4*>(vdst + d) =
          *reinterpret_cast<const uint4*>(v_tok + h * head_dim + d);
    }
  }
}

}  // namespace

extern "C" {

void launch_rope_store_kv(void* q, void* k, const void* v, void* key_cache,
                          void* val_cache, const void* cos_sin,
                          const void* positions, const void* slot_mapping,
                          int tokens, int num_q_heads, int num_kv_heads,
                          int head_dim, int page_size, int q_stride,
                          int kv_stride, hipStream_t stream) {
  dim3 grid(tokens), block(256);
  hipLaunchKernelGGL(rope_store_kv_kernel, grid, block, 0, stream,
                     (__hip_bfloat16*)q, (__hip_bfloat16*)k,
                     (const __hip_bfloat16*)v, (__hip_bfloat16*)key_cache,
                     (__hip_bfloat16*)val_cache, (const float*)cos_sin,
                     (const int*)positions, (const int*)slot_mapping,
                     num_q_heads, num_kv_heads, head_dim, page_size,
                     q_stride, kv_stride);
}

}  // extern "C"
