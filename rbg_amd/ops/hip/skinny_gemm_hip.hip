#include "hip/hip_runtime.h"
// Skinny GEMM for decode-shape projections: C[M,N] = A[M,K] x W[N,K]^T,
// bf16 in / f32 accumulate / bf16 out, M <= 256 (a decode batch).
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
//   grid  = (N/32) * SPLITS workgroups of 4 waves: (k_slice 0..1) x
//           (n_tile 0..1) over a 32-column C panel, 64 k per step
//   stage = W[32 x 64] + A[M x 64] through LDS via coalesced 16-lane row
//           pieces (direct per-lane fragment gathers measured 3-5x slow)
//   pipe  = DOUBLE-buffered LDS + TWO register load-sets issuing 2-3
//           steps ahead, ONE barrier per step.  Earlier cuts of this
//           kernel staged 1 step ahead behind 2 barriers: every step then
//           serialized on full HBM latency (rocprof: SQ_WAIT_ANY = 90%
//           of cycles, 42 us for a 33 MB weight stream).  Small WGs
//           (23 KB stage + 32 KB reduce scratch) keep ~3 WGs resident
//           per CU so stalled WGs are covered by runnable ones.
//   math  = per step each wave reads 1 W + M_TILES A fragments from LDS
//           (contiguous-8 per-lane MFMA layout) and issues M_TILES MFMAs
//   end   = the 2 k-slices of each n-tile fold through the LDS reduce
//           scratch, one bf16 store; SPLITS>1 writes f32 partials +
//           reduce_splits() folds them
//
// A-traffic note: a 32-column panel reads all of A (M x K, ~1 MB at
// M=128/K=4096) — L2-resident after the first panel per XCD, so HBM sees
// ~W + 8 x A bytes.
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 gg_bf8;

DEV_INLINE gg_bf8 lds_frag(const __hip_bfloat16* p) {
  union { uint4 u; gg_bf8 v; } cvt;
  cvt.u = *reinterpret_cast<const uint4*>(p);
  return cvt.v;
}

#define SG_KSTEP 64                   // k consumed per WG step
#define SG_LDS_PITCH (SG_KSTEP + 8)   // +8 bf16: bank-conflict pad
#define SG_NT 256                     // threads (4 waves)

template <int M_TILES, int SPLITS>
__global__ __launch_bounds__(SG_NT, 3) void skinny_gemm_kernel(
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
  const int k_slice = wave >> 1;             // 0..1, 32 k each per step
  const int k_wg = K / SPLITS;               // this WG's k span
  const int k0 = split * k_wg;
  const int nsteps = k_wg / SG_KSTEP;

  constexpr int A_ROWS = M_TILES * 16;
  constexpr int BUF = (32 + A_ROWS) * SG_LDS_PITCH;
  __shared__ union {
    __hip_bfloat16 stage[2][BUF];
    float red[4][M_TILES * 256];
  } lds;

  // staging pieces, 16 B per (row, chunk): a 64-k row is 8 chunks, so
  // W = 32 x 8 = 256 pieces (one per thread) and A = A_ROWS x 8 pieces
  // (up to 4 per thread); 8 lanes x 16 B = 128 B contiguous per row.
  // The two in-flight load sets are NAMED variables and the loop is
  // unrolled by 2: a runtime-indexed set[2] array was demoted to scratch
  // memory and every prefetch round-tripped through HBM (seen in the ISA
  // as scratch_load/store_dwordx4 in the hot loop — 5x slowdown)
  const int wrow_st = tid >> 3, wchk = (tid & 7) * 8;
  constexpr int A_PIECES = (A_ROWS * 8 + SG_NT - 1) / SG_NT;
  // the two in-flight load sets are SEPARATE named arrays: a struct array
  // passed through lambda references was demoted to scratch memory and
  // every prefetch round-tripped through HBM (ISA: scratch_*_dwordx4 in
  // the hot loop)
  // fully scalar load sets (a0..a3 per set): any array or struct here,
  // even with unrolled constant indices, was demoted to scratch memory by
  // the compiler and every prefetch round-tripped through HBM
  uint4 wreg0, wreg1;
  uint4 a00 = {}, a01 = {}, a02 = {}, a03 = {};
  uint4 a04 = {}, a05 = {}, a06 = {}, a07 = {};
  uint4 a10 = {}, a11 = {}, a12 = {}, a13 = {};
  uint4 a14 = {}, a15 = {}, a16 = {}, a17 = {};
  const __hip_bfloat16* wbase =
      w + (size_t)((n_group << 5) + wrow_st) * K + wchk;
  const __hip_bfloat16* ab0 = nullptr;
  const __hip_bfloat16* ab1 = nullptr;
  const __hip_bfloat16* ab2 = nullptr;
  const __hip_bfloat16* ab3 = nullptr;
  const __hip_bfloat16* ab4 = nullptr;
  const __hip_bfloat16* ab5 = nullptr;
  const __hip_bfloat16* ab6 = nullptr;
  const __hip_bfloat16* ab7 = nullptr;
#define SG_ABASE(s) (a + (size_t)min((tid + (s) * SG_NT) >> 3, M - 1) * K + \
                     ((tid + (s) * SG_NT) & 7) * 8)
  ab0 = SG_ABASE(0);
  if (A_PIECES > 1) ab1 = SG_ABASE(1);
  if (A_PIECES > 2) ab2 = SG_ABASE(2);
  if (A_PIECES > 3) ab3 = SG_ABASE(3);
  if (A_PIECES > 4) ab4 = SG_ABASE(4);
  if (A_PIECES > 5) ab5 = SG_ABASE(5);
  if (A_PIECES > 6) ab6 = SG_ABASE(6);
  if (A_PIECES > 7) ab7 = SG_ABASE(7);
  static_assert(A_PIECES <= 8, "A_PIECES grew past the scalar sets");

#define SG_LD(p, kb) (*reinterpret_cast<const uint4*>((p) + (kb)))
#define SG_ISSUE(step, wreg, a0, a1, a2, a3, a4, a5, a6, a7)                \
  do {                                                                      \
    const int kb = k0 + min(step, nsteps - 1) * SG_KSTEP;                   \
    wreg = SG_LD(wbase, kb);                                                \
    a0 = SG_LD(ab0, kb);                                                    \
    if (A_PIECES > 1) a1 = SG_LD(ab1, kb);                                  \
    if (A_PIECES > 2) a2 = SG_LD(ab2, kb);                                  \
    if (A_PIECES > 3) a3 = SG_LD(ab3, kb);                                  \
    if (A_PIECES > 4) a4 = SG_LD(ab4, kb);                                  \
    if (A_PIECES > 5) a5 = SG_LD(ab5, kb);                                  \
    if (A_PIECES > 6) a6 = SG_LD(ab6, kb);                                  \
    if (A_PIECES > 7) a7 = SG_LD(ab7, kb);                                  \
  } while (0)

  // LDS store address for A piece s (rows past A_ROWS never occur: the
  // last partial piece only exists when A_PIECES*SG_NT > A_ROWS*8, and
  // then piece s covers i = tid + s*SG_NT < A_ROWS*8 by construction of
  // A_PIECES for the supported M_TILES set)
#define SG_AST(st, s) reinterpret_cast<uint4*>(                             \
    &st[(32 + ((tid + (s) * SG_NT) >> 3)) * SG_LDS_PITCH +                  \
        ((tid + (s) * SG_NT) & 7) * 8])
#define SG_WRITE(st, wreg, a0, a1, a2, a3, a4, a5, a6, a7)                  \
  do {                                                                      \
    *reinterpret_cast<uint4*>(&st[wrow_st * SG_LDS_PITCH + wchk]) = wreg;   \
    if (tid < A_ROWS * 8) *SG_AST(st, 0) = a0;                              \
    if (A_PIECES > 1) *SG_AST(st, 1) = a1;                                  \
    if (A_PIECES > 2) *SG_AST(st, 2) = a2;                                  \
    if (A_PIECES > 3) *SG_AST(st, 3) = a3;                                  \
    if (A_PIECES > 4) *SG_AST(st, 4) = a4;                                  \
    if (A_PIECES > 5) *SG_AST(st, 5) = a5;                                  \
    if (A_PIECES > 6) *SG_AST(st, 6) = a6;                                  \
    if (A_PIECES > 7) *SG_AST(st, 7) = a7;                                  \
  } while (0)

  f32x4 acc[M_TILES];
#pragma unroll
  for (int mt = 0; mt < M_TILES; ++mt) acc[mt] = {0.f, 0.f, 0.f, 0.f};

  const int kf = k_slice * 32 + gslice * 8;
  auto compute = [&](const __hip_bfloat16* st) {
    const gg_bf8 wf = lds_frag(
        &st[((n_tile << 4) + gl) * SG_LDS_PITCH + kf]);
#pragma unroll
    for (int mt = 0; mt < M_TILES; ++mt) {
      const gg_bf8 af = lds_frag(
          &st[(32 + mt * 16 + gl) * SG_LDS_PITCH + kf]);
      acc[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, wf, acc[mt],
                                                        0, 0, 0);
    }
  };

  __hip_bfloat16* buf0 = lds.stage[0];
  __hip_bfloat16* buf1 = lds.stage[1];

  // pipeline prologue: tiles 0,1 in flight; tile 0 lands in buffer 0
  SG_ISSUE(0, wreg0, a00, a01, a02, a03, a04, a05, a06, a07);
  SG_ISSUE(1, wreg1, a10, a11, a12, a13, a14, a15, a16, a17);
  SG_WRITE(buf0, wreg0, a00, a01, a02, a03, a04, a05, a06, a07);
  SG_ISSUE(2, wreg0, a00, a01, a02, a03, a04, a05, a06, a07);

  // nsteps is even (K % (SPLITS * 2*SG_KSTEP) == 0, enforced host-side);
  // tile t+1 was issued 2 steps ago so its vmcnt wait is covered, and a
  // buffer's previous tile was fully read before the barrier, so each
  // step needs exactly one barrier
  for (int step = 0; step < nsteps; step += 2) {
    __syncthreads();
    compute(buf0);                                 // tile step
    SG_WRITE(buf1, wreg1, a10, a11, a12, a13, a14, a15, a16, a17);
    SG_ISSUE(step + 3, wreg1, a10, a11, a12, a13, a14, a15, a16, a17);
    __syncthreads();
    compute(buf1);                                 // tile step+1
    SG_WRITE(buf0, wreg0, a00, a01, a02, a03, a04, a05, a06, a07);
    SG_ISSUE(step + 4, wreg0, a00, a01, a02, a03, a04, a05, a06, a07);
  }
#undef SG_ISSUE
#undef SG_WRITE
#undef SG_ABASE
#undef SG_LD
#undef SG_AST

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

  // fold 2 k-slices; elem e of n-tile nt: lane64 = e & 63,
  // reg = (e >> 6) & 3, mt = e >> 8 -> m = mt*16 + 4*(lane64>>4) + reg,
  // n = panel + nt*16 + (lane64 & 15)
  const int elems = M_TILES * 256 * 2;
  for (int e = tid; e < elems; e += SG_NT) {
    const int nt = e >= M_TILES * 256;
    const int ee = e - nt * M_TILES * 256;
    const float v = lds.red[nt][ee] + lds.red[2 + nt][ee];
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
  const dim3 grid((N >> 5) * splits), block(SG_NT);
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
  else if (m_tiles <= 8) LAUNCH_MT(8);
  else LAUNCH_MT(16);
#undef LAUNCH_MT
}

extern "C" void launch_reduce_splits(void* c, const void* ws, int splits,
                                     long mn, hipStream_t stream) {
  const int threads = 256;
 hipLaunchKernelGGL(( reduce_splits_kernel), dim3(dim3((mn + threads - 1) / threads)), dim3(dim3(threads)),
                         0, stream, (const float*)ws, (__hip_bfloat16*)c,
                                      splits, mn);
}
