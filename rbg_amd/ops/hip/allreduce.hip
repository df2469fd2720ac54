// One-shot cross-GPU all-reduce over peer-mapped HBM (gfx950) — the TP
// decode collective, graph-capturable.
//
// Why not RCCL here: decode-shape all-reduces ([batch, hidden] bf16,
// 0.25-8 MB) are LATENCY-bound, and on an 8-GPU MI355X node every pair of
// GPUs has its own xGMI link, so the optimal small-message algorithm is
// one-shot: every rank pushes its block into its own exported buffer,
// peers read ALL ranks' blocks in parallel over their 7 links and reduce
// locally.  A ring all-reduce serializes 2(w-1) link hops; one-shot is a
// single hop with all links active.  And because the whole exchange is
// plain kernels + device-memory flag exchange (no library calls, no
// host-side synchronization), it captures into hipGraphs — which RCCL
// collectives inside torch's ProcessGroup could not be trusted to do on
// the first 8-GPU run.
//
// Synchronization protocol (per block b, rank r, epoch e):
//   1. wait  my.end[b][p]   >= e-1  for all p   (peers done reading my e-1)
//   2. copy my input slice b into my data buffer; __threadfence_system
//   3. push  sig[p].start[b][r] = e  for all p  (remote uncached store)
//   4. wait  my.start[b][p] >= e     for all p  (all slices visible)
//   5. out slice b = sum over p of data[p] slice b   (xGMI reads)
//   6. push  sig[p].end[b][r] = e; bump my per-block epoch counter
// Epochs live in device memory (my.counter[b], single-writer), so graph
// REPLAYS advance them naturally — nothing depends on kernel arguments
// changing between calls.
// Signal buffers are allocated UNCACHED (hipDeviceMallocUncached) so a
// remote store is immediately visible to the local spin loop; data
// buffers are ordinary hipMalloc (the threadfence makes them visible
// before the start flag lands).
//
// Every spin has a cycle budget; on overrun the kernel sets sig.error and
// exits (host raises) — a misconfigured world degrades to an error, not a
// hung GPU.
//
// Capability analog: the TP all-reduce the reference delegates to its
// engines' NCCL (SURVEY §2.2 TP row; §5 "prefer direct reduce-scatter/
// all-gather over per-link-bound ring" note).
#include "common.h"

#define AR_MAX_WORLD 8
#define AR_BLOCKS 64
#define AR_THREADS 256

namespace {

struct __align__(128) Signals {
  unsigned start[AR_BLOCKS][AR_MAX_WORLD];
  unsigned end[AR_BLOCKS][AR_MAX_WORLD];
  unsigned counter[AR_BLOCKS];
  unsigned error;
};

struct RankPtrs {
  Signals* sig[AR_MAX_WORLD];
  __hip_bfloat16* data[AR_MAX_WORLD];
};

DEV_INLINE unsigned ld_flag(volatile unsigned* p) { return *p; }
DEV_INLINE void st_flag(volatile unsigned* p, unsigned v) { *p = v; }

// spin until *p >= want; returns false on timeout
DEV_INLINE bool spin_ge(volatile unsigned* p, unsigned want) {
  for (long i = 0; i < (1L << 28); ++i) {
    if ((int)(ld_flag(p) - want) >= 0) return true;
    __builtin_amdgcn_s_sleep(1);
  }
  return false;
}

template <int WORLD>
__global__ __launch_bounds__(AR_THREADS) void xgmi_allreduce_kernel(
    __hip_bfloat16* __restrict__ out, const __hip_bfloat16* __restrict__ in,
    const RankPtrs ptrs, const int rank, const long n /* elements */) {
  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  Signals* my = ptrs.sig[rank];
  __hip_bfloat16* mydata = ptrs.data[rank];

  // slice for this block, 8-element aligned
  const long per = ((n + AR_BLOCKS - 1) / AR_BLOCKS + 7) & ~7L;
  const long lo = b * per;
  const long hi = min(n, lo + per);

  const unsigned e = my->counter[b] + 1;

  // 1. previous epoch fully consumed by every peer
  if (tid < WORLD && tid != rank) {
    if (!spin_ge(&my->end[b][tid], e - 1)) my->error = 1;
  }
  __syncthreads();

  // 2. publish my slice
  for (long i = lo + tid * 8; i < hi; i += AR_THREADS * 8) {
    *reinterpret_cast<uint4*>(mydata + i) =
        *reinterpret_cast<const uint4*>(in + i);
  }
  __threadfence_system();
  __syncthreads();

  // 3. start flags to every peer (and self)
  if (tid < WORLD) st_flag(&ptrs.sig[tid]->start[b][rank], e);

  // 4. wait for everyone's slice
  if (tid < WORLD && tid != rank) {
    if (!spin_ge(&my->start[b][tid], e)) my->error = 1;
  }
  __syncthreads();

  // 5. reduce: read every rank's slice over its link, sum in f32
  for (long i = lo + tid * 8; i < hi; i += AR_THREADS * 8) {
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
    for (int p = 0; p < WORLD; ++p) {
      Bf16x8U v;
      v.u = *reinterpret_cast<const uint4*>(ptrs.data[p] + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += bf2f(v.e[j]);
    }
    Bf16x8U o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o.e[j] = f2bf(acc[j]);
    *reinterpret_cast<uint4*>(out + i) = o.u;
  }
  __threadfence_system();
  __syncthreads();

  // 6. end flags + epoch bump
  if (tid < WORLD) st_flag(&ptrs.sig[tid]->end[b][rank], e);
  if (tid == 0) my->counter[b] = e;
}

}  // namespace

extern "C" {

void launch_xgmi_allreduce(void* out, const void* in, void** sig_ptrs,
                           void** data_ptrs, int world, int rank, long n,
                           hipStream_t stream) {
  RankPtrs ptrs;
  for (int i = 0; i < AR_MAX_WORLD; ++i) {
    ptrs.sig[i] = (Signals*)(i < world ? sig_ptrs[i] : nullptr);
    ptrs.data[i] = (__hip_bfloat16*)(i < world ? data_ptrs[i] : nullptr);
  }
  dim3 grid(AR_BLOCKS), block(AR_THREADS);
#define AR_LAUNCH(W)                                                         \
  hipLaunchKernelGGL((xgmi_allreduce_kernel<W>), grid, block, 0, stream,     \
                     (__hip_bfloat16*)out, (const __hip_bfloat16*)in, ptrs,  \
                     rank, n)
  switch (world) {
    case 1: AR_LAUNCH(1); break;
    case 2: AR_LAUNCH(2); break;
    case 3: AR_LAUNCH(3); break;
    case 4: AR_LAUNCH(4); break;
    case 5: AR_LAUNCH(5); break;
    case 6: AR_LAUNCH(6); break;
    case 7: AR_LAUNCH(7); break;
    case 8: AR_LAUNCH(8); break;
    default: break;
  }
#undef AR_LAUNCH
  HIP_KERNEL_CHECK();
}

long xgmi_allreduce_signal_bytes() { return (long)sizeof(Signals); }

long xgmi_allreduce_error_offset() { return (long)offsetof(Signals, error); }
long xgmi_allreduce_counter_offset() {
  return (long)offsetof(Signals, counter);
}

}  // extern "C"
