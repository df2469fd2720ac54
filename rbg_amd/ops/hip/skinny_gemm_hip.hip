#include "hip/hip_runtime.h"
// Skinny GEMM for decode-shape projections: C[M,N] = A[M,K] x W[N,K]^T,
// bf16 in / f32 accumulate / bf16 out, M <= 128 (a decode batch).
//
// Why hand-written: at M=128 the layer projections are pure weight streams
// (ideal time = W bytes / HBM bandwidth), yet every library backend leaves
// 2-5x on the narrow-N shapes (profiles/gemm_ab_b128.json: hipBLASLt and
// rocBLAS both ~1.7-2.4 TB/s on o/qkv/down vs ~6 TB/s on the wide gate_up;
// composable-kernel far worse).  Library tilings are built for big M; a
// skinny GEMM wants every CU streaming DISTINCT weight rows with the whole
// A operand riding in L2.
//
// Kernel shape (CDNA4, 64-lane waves, v_mfma_f32_16x16x32_bf16):
//   grid  = (N/32) * SPLITS workgroups, 8 waves each
//   wave  = (k_slice 0..3) x (n_tile 0..1): the WG owns a 32-column C
//           panel and walks its split's K range 128 k per step, the 4
//           k-slices covering 32 k each.  8 waves + 64 KB LDS keep TWO
//           workgroups resident per CU, so one WG's MFMAs and global
//           loads run while the other sits in its staging barrier — with
//           a single 16-wave WG (v2 of this kernel) every wave stalled
//           on the same barrier and the step time collapsed to the raw
//           HBM latency (measured 5x slower than the library)
//   stage = W[32 x 256] and A[M x 256] land in LDS through COALESCED
//           16-lane x 16 B row pieces (direct per-lane fragment gathers
//           from global were 3-5x slower: 16 discontiguous 16 B requests
//           per instruction — measured, first cut of this kernel), with
//           the next step's pieces prefetched into registers under the
//           current step's MFMAs (attn_prefill.hip staging discipline)
//   math  = per step each wave reads its fragment window from LDS
//           (1 W frag + M_TILES A frags, contiguous-8 per-lane layout)
//           and issues M_TILES MFMAs
//   end   = the 8 k-slices of each n-tile fold through the SAME LDS
//           buffer (re-used as f32 scratch), one bf16 store; SPLITS>1
//           writes f32 partials + reduce_splits() folds them
//
// A-traffic note: a 32-column panel reads all of A (M x K, ~1 MB at
// M=128/K=4096) — L2-resident after the first panel per XCD, so HBM sees
// ~W + 8 x A bytes; LDS sees ~2x A (two waves share a k-window), well
// under the 128 B/clk LDS budget needed to keep W streaming at HBM rate.
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 gg_bf8;

DEV_INLINE gg_bf8 lds_frag(const __hip_bfloat16* p) {
  union { uint4 u; gg_bf8 v; } cvt;
  cvt.u = *reinterpret_cast<const uint4*>(p);
  return cvt.v;
}

#define SG_KSTEP 128              // k consumed per WG step
#define SG_LDS_PITCH (SG_KSTEP + 8)   // +8 bf16: bank-conflict pad

template <int M_TILES, int SPLITS>
__global__ __launch_bounds__(512, 2) void skinny_gemm_kernel(
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
  const int k_slice = wave >> 1;             // 0..3, 32 k each per step
  const int k_wg = K / SPLITS;               // this WG's k span
  const int k0 = split * k_wg;
  const int nsteps = k_wg / SG_KSTEP;

  // LDS: staged W/A panels for the current step, re-used afterwards as the
  // f32 cross-slice reduction scratch (64 KB at M_TILES=8 -> 2 WGs/CU)
  constexpr int A_ROWS = M_TILES * 16;
  __shared__ union {
    __hip_bfloat16 stage[(32 + A_ROWS) * SG_LDS_PITCH];
    float red[8][M_TILES * 256];
  } lds;
  __hip_bfloat16* w_lds = lds.stage;                        // [32][pitch]
  __hip_bfloat16* a_lds = lds.stage + 32 * SG_LDS_PITCH;    // [A_ROWS][pitch]

  // staging pieces: 16 B per (row, chunk); a 128-k row is 16 chunks, so
  // W = 32 x 16 = 512 pieces (one per thread, 16 lanes x 16 B = 256 B
  // contiguous per row) and A = A_ROWS x 16 pieces (up to 4 per thread)
  const int wrow_st = tid >> 4, wchk = (tid & 15) * 8;      // W piece
  constexpr int A_PIECES = (A_ROWS * 16 + 511) / 512;
  uint4 wreg, areg[A_PIECES];
  auto issue_loads = [&](int step) {
    const int kb = k0 + step * SG_KSTEP;
    {
      const int n = (n_group << 5) + wrow_st;
      wreg = *reinterpret_cast<const uint4*>(w + (size_t)n * K + kb + wchk);
    }
#pragma unroll
    for (int s = 0; s < A_PIECES; ++s) {
      const int i = tid + s * 512;
      const int row = min(i >> 4, M - 1);       // clamp: garbage rows masked
      const int chk = (i & 15) * 8;
      areg[s] = *reinterpret_cast<const uint4*>(a + (size_t)row * K + kb + chk);
    }
  };
  auto write_tile = [&]() {
    *reinterpret_cast<uint4*>(&w_lds[wrow_st * SG_LDS_PITCH + wchk]) = wreg;
#pragma unroll
    for (int s = 0; s < A_PIECES; ++s) {
      const int i = tid + s * 512;
      if (i < A_ROWS * 16)
        *reinterpret_cast<uint4*>(
            &a_lds[(i >> 4) * SG_LDS_PITCH + (i & 15) * 8]) = areg[s];
    }
  };

  f32x4 acc[M_TILES];
#pragma unroll
  for (int mt = 0; mt < M_TILES; ++mt) acc[mt] = {0.f, 0.f, 0.f, 0.f};

  issue_loads(0);
  write_tile();

  for (int step = 0; step < nsteps; ++step) {
    __syncthreads();                     // tile `step` visible to all waves
    if (step + 1 < nsteps)
      issue_loads(step + 1);             // in flight under the MFMAs

    const int kf = k_slice * 32 + gslice * 8;     // fragment k offset
    const gg_bf8 wf = lds_frag(
        &w_lds[((n_tile << 4) + gl) * SG_LDS_PITCH + kf]);
#pragma unroll
    for (int mt = 0; mt < M_TILES; ++mt) {
      const gg_bf8 af = lds_frag(
          &a_lds[(mt * 16 + gl) * SG_LDS_PITCH + kf]);
      acc[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, wf, acc[mt],
                                                        0, 0, 0);
    }
    __syncthreads();                     // all reads done before overwrite
    if (step + 1 < nsteps)
      write_tile();
  }

  // ---- cross-slice reduction through LDS (buffer re-use) ----------------
  __syncthreads();
#pragma unroll
  for (int mt = 0; mt < M_TILES; ++mt) {
    F32x4U u;
    u.v = acc[mt];
#pragma unroll
    for (int r = 0; r < 4; ++r)
      lds.red[wave][mt * 256 + r * 64 + lane] = u.e[r];
  }
  __syncthreads();

  // fold 4 k-slices; elem e of n-tile nt: lane64 = e & 63,
  // reg = (e >> 6) & 3, mt = e >> 8 -> m = mt*16 + 4*(lane64>>4) + reg,
  // n = panel + nt*16 + (lane64 & 15)
  const int elems = M_TILES * 256 * 2;
  for (int e = tid; e < elems; e += 512) {
    const int nt = e >= M_TILES * 256;
    const int ee = e - nt * M_TILES * 256;
    float v = 0.f;
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) v += lds.red[ks * 2 + nt][ee];
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
  const dim3 grid((N >> 5) * splits), block(512);
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
