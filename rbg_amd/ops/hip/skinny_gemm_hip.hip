#include "hip/hip_runtime.h"
// Skinny GEMM for decode-shape projections: C[M,N] = A[M,K] x W[N,K]^T,
// bf16 in / f32 accumulate / bf16 out, M <= 256 (a decode batch).
//
// Why hand-written: at M=128 the layer projections are pure weight streams
// (ideal time = W bytes / HBM bandwidth), yet every library backend leaves
// 2-5x on the narrow-N shapes (profiles/gemm_ab_b128.json: hipBLASLt and
// rocBLAS both ~1.7-2.4 TB/s on o/qkv/down vs ~6 TB/s on the wide gate_up;
// composable-kernel far worse).  Library tilings are built for big M; a
// skinny GEMM instead wants every CU streaming DISTINCT weight rows with
// the whole A operand riding in L2.
//
// Shape of the kernel (CDNA4, 64-lane waves, v_mfma_f32_16x16x32_bf16):
//   grid  = (N/32) * SPLITS workgroups, 16 waves each (1024 threads)
//   wave  = (k_slice 0..7) x (n_tile 0..1): 16 waves cover a 32-column
//           C panel, splitting this split's K range 8 ways
//   loop  = per 32-k step: one W fragment (16 rows x 32 k, the MFMA B
//           operand straight from global: lane n = l&15, k = (l>>4)*8 --
//           the contiguous-8 per-lane layout matches row-major [N,K]) and
//           M_TILES A fragments (same layout over A rows, clamped), one
//           MFMA each, 1-ahead prefetched so the weight stream stays
//           ahead of the math
//   end   = the 8 k-slices of each n-tile reduce through LDS (128 KB:
//           16 waves x 8 KB of f32 partials), then one bf16 store --
//           no global atomics, deterministic, single kernel when SPLITS=1
//   SPLITS>1 writes f32 partials to a workspace; reduce_splits() folds
//           them to bf16 (two tiny extra launches only when K is deep)
//
// A-traffic note: computing a 32-column panel reads all of A (M x K);
// with M=128 that is ~1 MB for K=4096 -- resident in each XCD's L2 after
// the first panel, so HBM sees ~W + 8 x A bytes total.
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 gg_bf8;

DEV_INLINE gg_bf8 g_load_bf8(const __hip_bfloat16* p) {
  union { uint4 u; gg_bf8 v; } cvt;
  cvt.u = *reinterpret_cast<const uint4*>(p);
  return cvt.v;
}

template <int M_TILES, int SPLITS>
__global__ __launch_bounds__(1024, 1) void skinny_gemm_kernel(
    const __hip_bfloat16* __restrict__ a,   // [M, K]
    const __hip_bfloat16* __restrict__ w,   // [N, K]
    __hip_bfloat16* __restrict__ c,         // [M, N]   (SPLITS == 1)
    float* __restrict__ ws,                 // [SPLITS, M, N] (SPLITS > 1)
    const int M, const int N, const int K) {
  const int n_groups = N >> 5;
  const int n_group = blockIdx.x % n_groups;
  const int split = blockIdx.x / n_groups;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int gl = lane & 15;        // fragment row lane
  const int gslice = lane >> 4;    // fragment k sub-chunk

  const int n_tile = wave & 1;               // 0..1 within the 32-col panel
  const int k_slice = wave >> 1;             // 0..7
  const int k_per_split = K / SPLITS;
  const int k_per_wave = k_per_split >> 3;   // 8 slices per split
  const int k0 = split * k_per_split + k_slice * k_per_wave;
  const int iters = k_per_wave >> 5;         // 32 k per step

  const int n_base = (n_group << 5) + (n_tile << 4);
  const __hip_bfloat16* wrow = w + (size_t)(n_base + gl) * K + k0 +
                               gslice * 8;
  // A rows clamp to M-1: rows past M compute garbage that the store masks
  const __hip_bfloat16* arow[M_TILES];
#pragma unroll
  for (int mt = 0; mt < M_TILES; ++mt) {
    const int m = min(mt * 16 + gl, M - 1);
    arow[mt] = a + (size_t)m * K + k0 + gslice * 8;
  }

  f32x4 acc[M_TILES];
#pragma unroll
  for (int mt = 0; mt < M_TILES; ++mt) acc[mt] = {0.f, 0.f, 0.f, 0.f};

  // 1-ahead software pipeline: weights + A for step i+1 issue before the
  // MFMAs of step i retire (unconditional loads; the last iteration
  // re-reads its own step, harmless and branch-free)
  gg_bf8 wf = g_load_bf8(wrow);
  gg_bf8 af[M_TILES];
#pragma unroll
  for (int mt = 0; mt < M_TILES; ++mt) af[mt] = g_load_bf8(arow[mt]);

  for (int it = 0; it < iters; ++it) {
    const int nxt = (it + 1 < iters) ? (it + 1) << 5 : it << 5;
    gg_bf8 wn = g_load_bf8(wrow + nxt);
    gg_bf8 an[M_TILES];
#pragma unroll
    for (int mt = 0; mt < M_TILES; ++mt)
      an[mt] = g_load_bf8(arow[mt] + nxt);
#pragma unroll
    for (int mt = 0; mt < M_TILES; ++mt)
      acc[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          af[mt], wf, acc[mt], 0, 0, 0);
    wf = wn;
#pragma unroll
    for (int mt = 0; mt < M_TILES; ++mt) af[mt] = an[mt];
  }

  // ---- cross-slice reduction through LDS --------------------------------
  // wave's partial panel: M_TILES x (16 rows x 16 cols) f32, 4 regs/lane
  __shared__ float red[16][M_TILES * 256];
#pragma unroll
  for (int mt = 0; mt < M_TILES; ++mt) {
    F32x4U u;
    u.v = acc[mt];
#pragma unroll
    for (int r = 0; r < 4; ++r)
      red[wave][mt * 256 + r * 64 + lane] = u.e[r];
  }
  __syncthreads();

  // each thread folds 8 k-slices for its share of the two n-tiles.
  // elem e of tile nt: lane64 = e & 63, reg = (e >> 6) & 3, mt = e >> 8
  //   -> m = mt*16 + 4*(lane64>>4) + reg, n = n_base' + (lane64 & 15)
  const int elems = M_TILES * 256 * 2;              // both n-tiles
  for (int e = tid; e < elems; e += 1024) {
    const int nt = e >= M_TILES * 256;
    const int ee = e - nt * M_TILES * 256;
    float v = 0.f;
#pragma unroll
    for (int ks = 0; ks < 8; ++ks) v += red[ks * 2 + nt][ee];
    const int l64 = ee & 63;
    const int reg = (ee >> 6) & 3;
    const int mt = ee >> 8;
    const int m = mt * 16 + ((l64 >> 4) << 2) + reg;
    if (m < M) {
      const int n = (n_group << 5) + (nt << 4) + (l64 & 15);
      if (SPLITS == 1)
        c[(size_t)m * N + n] = f2bf(v);
      else
        ws[((size_t)split * M + m) * N + n] = v;
    }
  }
}

__global__ void reduce_splits_kernel(const float* __restrict__ ws,
                                     __hip_bfloat16* __restrict__ c,
                                     const int splits, const long mn) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= mn) return;
  float v = 0.f;
  for (int s = 0; s < splits; ++s) v += ws[(size_t)s * mn + i];
  c[i] = f2bf(v);
}

// ---------------------------------------------------------------------------
// C launchers (torch glue lives in bindings.cpp)

extern "C" void launch_skinny_gemm(void* c, void* ws, const void* a,
                                   const void* w, int M, int N, int K,
                                   int splits, hipStream_t stream) {
  const dim3 grid((N >> 5) * splits), block(1024);
  const __hip_bfloat16* ap = (const __hip_bfloat16*)a;
  const __hip_bfloat16* wp = (const __hip_bfloat16*)w;
  __hip_bfloat16* cp = (__hip_bfloat16*)c;
  float* wsp = (float*)ws;
  const int m_tiles = (M + 15) >> 4;

#define LAUNCH_MT(MT)                                                        \
  do {                                                                       \
    if (splits == 1)                                                         \
     hipLaunchKernelGGL(( skinny_gemm_kernel<MT, 1>), dim3(grid), dim3(block), 0, stream,                  \
          ap, wp, cp, nullptr, M, N, K);                                     \
    else if (splits == 2)                                                    \
     hipLaunchKernelGGL(( skinny_gemm_kernel<MT, 2>), dim3(grid), dim3(block), 0, stream,                  \
          ap, wp, cp, wsp, M, N, K);                                         \
    else if (splits == 4)                                                    \
     hipLaunchKernelGGL(( skinny_gemm_kernel<MT, 4>), dim3(grid), dim3(block), 0, stream,                  \
          ap, wp, cp, wsp, M, N, K);                                         \
    else                                                                     \
     hipLaunchKernelGGL(( skinny_gemm_kernel<MT, 8>), dim3(grid), dim3(block), 0, stream,                  \
          ap, wp, cp, wsp, M, N, K);                                         \
  } while (0)

  if (m_tiles <= 1) LAUNCH_MT(1);
  else if (m_tiles <= 2) LAUNCH_MT(2);
  else if (m_tiles <= 4) LAUNCH_MT(4);
  else LAUNCH_MT(8);
#undef LAUNCH_MT
}

extern "C" void launch_reduce_splits(void* c, const void* ws, int splits,
                                     long mn, hipStream_t stream) {
  const int threads = 256;
 hipLaunchKernelGGL(( reduce_splits_kernel), dim3(dim3((mn + threads - 1) / threads)), dim3(dim3(threads)),
                         0, stream, (const float*)ws, (__hip_bfloat16*)c,
                                      splits, mn);
}
