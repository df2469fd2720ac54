// Paged KV peer copy over xGMI (gfx950) — the prefill->decode migration
// dataplane.  The decode engine exports its KV pool once via
// hipIpcGetMemHandle; the prefill engine maps it (hipIpcOpenMemHandle with
// lazy peer access, which enables the xGMI route) and PUSHES a sequence's
// pages with this kernel: a fused gather(src pages)->scatter(dst pages)
// whose stores land directly in the peer GPU's HBM over the point-to-point
// xGMI link.  No intermediate gather buffer (the round-1 RCCL path paid a
// full extra HBM pass for index_select+contiguous), no collective library
// in the path, and zero CUs used on the decode GPU — decode keeps stepping
// while the copy streams on a dedicated prefill-side stream.
//
// Layout contract (engine/kv_cache.py): pool = [L, 2, P, kvh, ps, hd] bf16,
// so one (layer, k|v, page) chunk is kvh*ps*hd contiguous elements
// (Llama-3-8B: 32 KB).  Chunk count per migrated page = 2*L.
//
// Grid: one block per (page, chunk) pair; 256 lanes each copy 16 B/iter
// (uint4 — the full-rate global access width).  A 2048-token sequence is
// 128 pages x 64 chunks = 8192 blocks, far above the 256-CU fill line.
//
// Capability analog: the Mooncake transfer engine the reference orchestrates
// (reference keps/74-mooncake-integration/README.md:45-140), re-realized as
// a single-node xGMI push per BASELINE.json's north star
// ("hipMemcpyPeerAsync over xGMI overlapped with decode on HIP streams";
// kernel-scatter form chosen over per-chunk hipMemcpyPeerAsync calls because
// the pool layout makes a page 2*L discontiguous 32 KB chunks — thousands of
// SDMA submissions per sequence would be launch-bound, one kernel is not).
#include "common.h"

namespace {

__global__ void kv_peer_copy_kernel(
    uint4* __restrict__ dst_base,        // peer-mapped pool base
    const uint4* __restrict__ src_base,  // local pool base
    const int* __restrict__ src_pages,   // [n]
    const int* __restrict__ dst_pages,   // [n]
    const int n_pages,
    const int n_chunks,                  // 2 * layers
    const int chunk_vec,                 // kvh*ps*hd*2B / 16B
    const long src_page_stride_vec,      // one page step inside [.., P, ..]
    const long dst_page_stride_vec,
    const long src_chunk_stride_vec,     // one (layer,k|v) step = P*page
    const long dst_chunk_stride_vec) {
  const int page_i = blockIdx.x;
  const int chunk = blockIdx.y;
  if (page_i >= n_pages || chunk >= n_chunks) return;
  const uint4* src = src_base + chunk * src_chunk_stride_vec +
                     (long)src_pages[page_i] * src_page_stride_vec;
  uint4* dst = dst_base + chunk * dst_chunk_stride_vec +
               (long)dst_pages[page_i] * dst_page_stride_vec;
  for (int i = threadIdx.x; i < chunk_vec; i += blockDim.x)
    dst[i] = src[i];
  // system-scope release: the completion event is observed by the HOST
  // and relayed to the PEER process (import_commit), so the remote
  // stores must be visible across agents, not just this one
  __threadfence_system();
}

}  // namespace

extern "C" void launch_kv_peer_copy(
    void* dst_base, const void* src_base, const void* src_pages,
    const void* dst_pages, int n_pages, int n_chunks, int chunk_vec,
    long src_page_stride_vec, long dst_page_stride_vec,
    long src_chunk_stride_vec, long dst_chunk_stride_vec,
    hipStream_t stream) {
  dim3 grid(n_pages, n_chunks);
  const int threads = chunk_vec >= 256 ? 256 : 64;
  hipLaunchKernelGGL(kv_peer_copy_kernel, grid, dim3(threads), 0, stream,
                     (uint4*)dst_base, (const uint4*)src_base,
                     (const int*)src_pages, (const int*)dst_pages, n_pages,
                     n_chunks, chunk_vec, src_page_stride_vec,
                     dst_page_stride_vec, src_chunk_stride_vec,
                     dst_chunk_stride_vec);
  HIP_KERNEL_CHECK();
}
