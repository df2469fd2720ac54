// Varlen causal flash-attention prefill (gfx950), MFMA-tiled.
//
// Structure (guide §5/Appendix B "fused attention prefill"): one workgroup =
// 64 query rows x one q-head; 4 waves, 16 q-rows each; K/V tiles of 64 keys
// staged in LDS (K row-major, V transposed for the PV B-operand), rows
// padded by 8 bf16 (16 B) so the 16-lane ds_read_b128 groups are
// bank-conflict-free; QK^T and PV on v_mfma_f32_16x16x32_bf16; online
// softmax with the running (m, l) per q-row kept in registers, row
// reductions via 16-lane xor shuffles (no serial-lane softmax —
// guide common-mistake #6).
//
// v1 is the correctness-first instance of the guide's ladder; the staged
// upgrades (XOR-swizzle + tr_read V, 8-wave 32x32 structure, async-STAGE)
// are applied in later rounds against rocprof evidence.
//
// Capability analog: prefill attention of the engines the reference
// orchestrates (SURVEY §2.3 prefill engine row).
#include "common.h"

namespace {

constexpr int HEAD_DIM = 128;
constexpr int KTILE = 64;      // keys per LDS tile
constexpr int PAD = 8;         // bf16 row padding (16 B) — bank-conflict fix
constexpr float NEG_INF = -1e30f;

typedef __attribute__((ext_vector_type(8))) __bf16 mfma_bf8;
typedef __attribute__((ext_vector_type(4))) float mfma_f4;

DEV_INLINE mfma_bf8 load_bf8(const __hip_bfloat16* p) {
  union { uint4 u; mfma_bf8 v; } cvt;
  cvt.u = *reinterpret_cast<const uint4*>(p);
  return cvt.v;
}

DEV_INLINE mfma_bf8 lds_bf8(const __hip_bfloat16* p) {
  union { uint4 u; mfma_bf8 v; } cvt;
  cvt.u = *reinterpret_cast<const uint4*>(p);
  return cvt.v;
}

// V image element offset for the ds_read_b64_tr_b16 PV path.  Per 32-key
// step (ks) and 16-dim column tile (dt), a 16-lane group's [4 key][16 dim]
// block lives at elems {q*64 + j*16 + d16} with the tr-read redistributing
// it so lane l ends with keys q*8+half*4+j at its dim column (l&15) —
// exactly the MFMA B-fragment (guide §2 tr-read layout; conflict-free
// subtiling).  Writes land as contiguous 8-element (16 B) runs per
// (key, 8-dim chunk), so staging stays fully vectorized.
DEV_INLINE int v_img_off(int key, int dim) {
  const int ks = key >> 5, kk = key & 31, q = kk >> 3, jj = kk & 7;
  return ks * 4096 + (dim >> 4) * 512 + (jj >> 2) * 256 + q * 64 +
         (jj & 3) * 16 + (dim & 15);
}

// grid: 1D, nblocks * num_q_heads.  SWZ=1 remaps block ids so the QPG
// q-heads sharing one (q-block, kv-head) K/V tile land on the SAME XCD
// (dispatcher places block b on XCD b%8 — guide T1) temporally adjacent:
// the sharers then hit that XCD's L2 instead of re-pulling HBM (the GQA
// staging re-read is the dominant prefill cost, profiles/).
// DB=1: T14 register staging (issue tile t+1's global loads BEFORE tile
// t's compute, write them to LDS after the barrier) — HBM latency hides
// under the MFMAs at the cost of ~32 staging VGPRs (guide G15/T14).
// NW: waves per block (4 -> 64 q rows, 8 -> 128 q rows; the bigger tile
// halves K/V staging traffic per output row).
template <int SWZ, int DB, int NW>
__global__ __launch_bounds__(NW * 64) void prefill_attn_kernel(
    __hip_bfloat16* __restrict__ out,        // [T, QH, D]
    const __hip_bfloat16* __restrict__ q,    // [T, QH, D]
    const __hip_bfloat16* __restrict__ k,    // [T, KVH, D]
    const __hip_bfloat16* __restrict__ v,    // [T, KVH, D]
    const int* __restrict__ block_info,      // [nblocks, 4] =
    //   (q_start_row, q_block, kv_start_row, q_offset) — q_offset is the
    //   position of q row 0 within the sequence (prefix caching / chunked
    //   prefill attend over [past; new] K/V)
    const int* __restrict__ seq_lens,        // [nblocks] KV length of the seq
    const float scale, const int num_q_heads, const int num_kv_heads,
    const int q_stride, const int kv_stride,
    const int nblocks) {
  const int qpg_n = num_q_heads / num_kv_heads;
  int blk, qh;
  if (SWZ && qpg_n > 1) {
    // group g = (block, kvh); its qpg_n sharers get ids c*8*qpg + q*8 + g%8
    const int lin = blockIdx.x;
    const int G = nblocks * num_kv_heads;
    const int chunk = 8 * qpg_n;
    const int full = (G / 8) * chunk;
    int g, qpg_idx;
    if (lin < full) {
      const int c = lin / chunk, r = lin % chunk;
      qpg_idx = r >> 3;
      g = c * 8 + (r & 7);
    } else {
      const int idx = lin - full;
      g = (G / 8) * 8 + idx / qpg_n;
      qpg_idx = idx % qpg_n;
    }
    blk = g / num_kv_heads;
    qh = (g % num_kv_heads) * qpg_n + qpg_idx;
  } else {
    blk = blockIdx.x / num_q_heads;
    qh = blockIdx.x % num_q_heads;
  }
  const int kvh = qh / qpg_n;
  const int q_start = block_info[blk * 4];
  const int qblock = block_info[blk * 4 + 1];
  const int kv_start = block_info[blk * 4 + 2];
  const int q_offset = block_info[blk * 4 + 3];
  const int seq_len = seq_lens[blk];     // total KV tokens

  constexpr int QTILE = NW * 16;     // q rows per block (16 per wave)
  constexpr int NTHREADS = NW * 64;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int gl = lane & 15;          // fragment row/col lane
  const int gslice = lane >> 4;      // fragment k-slice (x8 elements)

  __shared__ __hip_bfloat16 k_lds[KTILE][HEAD_DIM + PAD];
  __shared__ __hip_bfloat16 v_img[KTILE * HEAD_DIM];   // tr-read image
  __shared__ __hip_bfloat16 p_lds[NW][16][KTILE + PAD];

  // ---- Q fragments: wave's 16 rows, 4 k-steps of 32 dims ---------------
  const int num_q_rows = seq_len - q_offset;    // rows in the q tensor
  const int q_row_in_chunk = qblock * QTILE + wave * 16 + gl;
  const int q_row = q_start + min(q_row_in_chunk, num_q_rows - 1);
  mfma_bf8 q_frag[4];
#pragma unroll
  for (int ks = 0; ks < 4; ++ks)
    q_frag[ks] = load_bf8(q + (size_t)q_row * q_stride +
                          (size_t)qh * HEAD_DIM + ks * 32 + gslice * 8);

  // per-lane softmax state: 4 q-rows (rows 4*gslice + r of the wave tile)
  float m_run[4], l_run[4];
  mfma_f4 acc_o[8];    // O[16 x 128]: 8 col-tiles of 16 dims
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = NEG_INF; l_run[r] = 0.f; }
#pragma unroll
  for (int dt = 0; dt < 8; ++dt) acc_o[dt] = {0.f, 0.f, 0.f, 0.f};

  // causal: this block needs keys up to its last valid q POSITION
  const int block_q_max = min(q_offset + qblock * QTILE + QTILE - 1,
                              seq_len - 1);
  const int wave_q_max = min(q_offset + qblock * QTILE + wave * 16 + 15,
                             seq_len - 1);
  const int nktiles = block_q_max / KTILE + 1;

  // staging helpers: each thread owns 4 (key, chunk) pieces of a tile.
  // UNCONDITIONAL clamped loads: a per-piece `if (krow < seq_len)` guard
  // makes hipcc branch around each load and drain vmcnt(0) per element
  // (guide §5 trap 4c); out-of-range rows are masked in softmax anyway.
  constexpr int PIECES = KTILE * (HEAD_DIM / 8) / NTHREADS;  // per thread
  static_assert(PIECES <= 4, "scalar staging sets cover 4 pieces");
  // SCALAR staging registers: `uint4 kreg[PIECES]` arrays were demoted to
  // scratch memory by the compiler (ISA: scratch_store/load_dwordx4 in
  // the k-tile loop), silently tripling the staged tile's HBM traffic —
  // the same demotion found while building skinny_gemm.hip
  uint4 kr0 = {}, kr1 = {}, kr2 = {}, kr3 = {};
  uint4 vr0 = {}, vr1 = {}, vr2 = {}, vr3 = {};
#define PF_ROW(s, kt2)                                                      \
  ((size_t)(kv_start + min((kt2) * KTILE + ((tid + (s) * NTHREADS) >> 4),  \
                           seq_len - 1)) * kv_stride +                      \
   (size_t)kvh * HEAD_DIM + ((tid + (s) * NTHREADS) & 15) * 8)
#define PF_LD(s, kt2, kr, vr)                                               \
  do {                                                                      \
    const size_t row = PF_ROW(s, kt2);                                      \
    kr = *reinterpret_cast<const uint4*>(k + row);                          \
    vr = *reinterpret_cast<const uint4*>(v + row);                          \
  } while (0)
  auto issue_loads = [&](int kt2) {
    PF_LD(0, kt2, kr0, vr0);
    if (PIECES > 1) PF_LD(1, kt2, kr1, vr1);
    if (PIECES > 2) PF_LD(2, kt2, kr2, vr2);
    if (PIECES > 3) PF_LD(3, kt2, kr3, vr3);
  };
#define PF_ST(s, kr, vr)                                                    \
  do {                                                                      \
    const int key = (tid + (s) * NTHREADS) >> 4;                            \
    const int chunk = ((tid + (s) * NTHREADS) & 15) * 8;                    \
    *reinterpret_cast<uint4*>(&k_lds[key][chunk]) = kr;                     \
    *reinterpret_cast<uint4*>(&v_img[v_img_off(key, chunk)]) = vr;          \
  } while (0)
  auto write_tile = [&]() {
    PF_ST(0, kr0, vr0);
    if (PIECES > 1) PF_ST(1, kr1, vr1);
    if (PIECES > 2) PF_ST(2, kr2, vr2);
    if (PIECES > 3) PF_ST(3, kr3, vr3);
  };

  if (DB) {   // prologue: tile 0 resident before the loop
    issue_loads(0);
    write_tile();
  }

  for (int kt = 0; kt < nktiles; ++kt) {
    const int k_base = kt * KTILE;
    if (DB) {
      __syncthreads();                 // tile kt visible to every wave
      if (kt + 1 < nktiles)
        issue_loads(kt + 1);           // in flight under the MFMAs below
    } else {
      __syncthreads();
      issue_loads(kt);
      write_tile();
      __syncthreads();
    }

    const bool compute_this = k_base <= wave_q_max;
    if (compute_this) {

    // ---- S = Q K^T : 4 col-tiles x 4 k-steps of MFMA --------------------
    mfma_f4 s[4];
#pragma unroll
    for (int ct = 0; ct < 4; ++ct) {
      s[ct] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        mfma_bf8 kf = lds_bf8(&k_lds[ct * 16 + gl][ks * 32 + gslice * 8]);
        s[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(q_frag[ks], kf, s[ct],
                                                        0, 0, 0);
      }
    }

    // ---- online softmax (rows 4*gslice+r, key col = ct*16+gl) -----------
    float p[4][4];
    float m_new[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int q_pos = q_offset + qblock * QTILE + wave * 16 + 4 * gslice + r;
      float row_max = NEG_INF;
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        const int k_pos = k_base + ct * 16 + gl;
        float sv = s[ct][r] * scale;
        if (k_pos > q_pos || k_pos >= seq_len || q_pos >= seq_len)
          sv = NEG_INF;
        p[ct][r] = sv;
        row_max = fmaxf(row_max, sv);
      }
      row_max = group16_max(row_max);
      m_new[r] = fmaxf(m_run[r], row_max);
      const float alpha = __expf(m_run[r] - m_new[r]);
      float psum = 0.f;
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        p[ct][r] = (p[ct][r] <= NEG_INF) ? 0.f : __expf(p[ct][r] - m_new[r]);
        psum += p[ct][r];
      }
      l_run[r] = l_run[r] * alpha + psum;
      m_run[r] = m_new[r];
#pragma unroll
      for (int dt = 0; dt < 8; ++dt) acc_o[dt][r] *= alpha;
    }

    // ---- P -> LDS (transpose to the A-fragment layout) ------------------
#pragma unroll
    for (int r = 0; r < 4; ++r)
#pragma unroll
      for (int ct = 0; ct < 4; ++ct)
        p_lds[wave][4 * gslice + r][ct * 16 + gl] = f2bf(p[ct][r]);
    // same-wave LDS read-after-write across lanes: drain the DS queue
    // explicitly (no cross-wave sharing, so no barrier needed)
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    // ---- O += P V : 8 col-tiles x 2 key-steps; V arrives by batched
    // hardware-transpose reads (ds_read_b64_tr_b16) ------------------------
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      mfma_bf8 pa = lds_bf8(&p_lds[wave][gl][ks * 32 + gslice * 8]);
      const unsigned vaddr = (unsigned)(unsigned long long)(
          &v_img[ks * 4096 + gslice * 64 + gl * 4]);
      unsigned long long vlo[8], vhi[8];
      asm volatile(
          "ds_read_b64_tr_b16 %0, %16\n\t"
          "ds_read_b64_tr_b16 %1, %16 offset:512\n\t"
          "ds_read_b64_tr_b16 %2, %16 offset:1024\n\t"
          "ds_read_b64_tr_b16 %3, %16 offset:1536\n\t"
          "ds_read_b64_tr_b16 %4, %16 offset:2048\n\t"
          "ds_read_b64_tr_b16 %5, %16 offset:2560\n\t"
          "ds_read_b64_tr_b16 %6, %16 offset:3072\n\t"
          "ds_read_b64_tr_b16 %7, %16 offset:3584\n\t"
          "ds_read_b64_tr_b16 %8, %16 offset:4096\n\t"
          "ds_read_b64_tr_b16 %9, %16 offset:4608\n\t"
          "ds_read_b64_tr_b16 %10, %16 offset:5120\n\t"
          "ds_read_b64_tr_b16 %11, %16 offset:5632\n\t"
          "ds_read_b64_tr_b16 %12, %16 offset:6144\n\t"
          "ds_read_b64_tr_b16 %13, %16 offset:6656\n\t"
          "ds_read_b64_tr_b16 %14, %16 offset:7168\n\t"
          "ds_read_b64_tr_b16 %15, %16 offset:7680\n\t"
          "s_waitcnt lgkmcnt(0)"
          : "=&v"(vlo[0]), "=&v"(vhi[0]), "=&v"(vlo[1]), "=&v"(vhi[1]),
            "=&v"(vlo[2]), "=&v"(vhi[2]), "=&v"(vlo[3]), "=&v"(vhi[3]),
            "=&v"(vlo[4]), "=&v"(vhi[4]), "=&v"(vlo[5]), "=&v"(vhi[5]),
            "=&v"(vlo[6]), "=&v"(vhi[6]), "=&v"(vlo[7]), "=&v"(vhi[7])
          : "v"(vaddr)
          : "memory");
      __builtin_amdgcn_sched_barrier(0);   // rule 18: keep MFMAs below
#pragma unroll
      for (int dt = 0; dt < 8; ++dt) {
        union { struct { unsigned long long lo, hi; } u; mfma_bf8 vf2; } vf;
        vf.u.lo = vlo[dt];
        vf.u.hi = vhi[dt];
        acc_o[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, vf.vf2,
                                                            acc_o[dt], 0, 0, 0);
      }
    }
    }  // compute_this

    if (DB && kt + 1 < nktiles) {
      __syncthreads();                 // every wave done READING tile kt
      write_tile();                    // land tile kt+1 (write-after-barrier)
    }
  }

  // ---- epilogue: normalize and store ------------------------------------
#pragma unroll
  for (int r = 0; r < 4; ++r) l_run[r] = group16_sum(l_run[r]);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int q_row_out = qblock * QTILE + wave * 16 + 4 * gslice + r;
    if (q_row_out >= num_q_rows) continue;
    const float inv_l = l_run[r] > 0.f ? 1.f / l_run[r] : 0.f;
    __hip_bfloat16* orow =
        out + ((size_t)(q_start + q_row_out) * num_q_heads + qh) * HEAD_DIM;
#pragma unroll
    for (int dt = 0; dt < 8; ++dt)
      orow[dt * 16 + gl] = f2bf(acc_o[dt][r] * inv_l);
  }
}

// ---------------------------------------------------------------------------
// 32x32x16 MFMA prefill (swz bit 3, ROUND2 §2a): 4 waves x 32 q-rows =
// the same 128-row block as the 8-wave 16x16 kernel, but each MFMA does
// 2x the flops, halving the MFMA instruction count and the per-flop
// fragment-read traffic.  Operand layouts verified empirically
// (tools/mfma_probe32.py: contiguous-8 k per half-wave for A and B;
// D row = (i%4) + 4*(l>>5) + 8*(i/4), col = l%32).
//
// The PV B-fragment (col = l%32, k = 8*(l>>5)+j) needs its own tr-read
// image: 4x16 row-major blocks indexed q = 2*(k-half) + (col>=16) — which
// equals the hardware's (l>>4) group index — so ds_read_b64_tr_b16 with
// per-lane addr base + (l>>4)*64 + (l&15)*4 lands each lane's column
// exactly (guide LDS §tr-read).  Staging writes stay 16-byte vectors.
DEV_INLINE int v_img32_off(int key, int dim) {
  const int ks = key >> 4, kk = key & 15;
  const int h = kk >> 3, j7 = kk & 7;
  return ks * 2048 + (dim >> 5) * 512 + ((j7 >= 4) ? 256 : 0) +
         (2 * h + ((dim >> 4) & 1)) * 64 + (j7 & 3) * 16 + (dim & 15);
}

typedef __attribute__((ext_vector_type(16))) float mfma_f16;

template <int SWZ, int DB>
__global__ __launch_bounds__(256, 2) void prefill_attn32_kernel(
    __hip_bfloat16* __restrict__ out,        // [T, QH, D]
    const __hip_bfloat16* __restrict__ q,    // [T, QH, D]
    const __hip_bfloat16* __restrict__ k,    // [T, KVH, D]
    const __hip_bfloat16* __restrict__ v,    // [T, KVH, D]
    const int* __restrict__ block_info,      // [nblocks, 4]
    const int* __restrict__ seq_lens,        // [nblocks]
    const float scale, const int num_q_heads, const int num_kv_heads,
    const int q_stride, const int kv_stride, const int nblocks) {
  const int qpg_n = num_q_heads / num_kv_heads;
  int blk, qh;
  if (SWZ && qpg_n > 1) {
    const int lin = blockIdx.x;
    const int G = nblocks * num_kv_heads;
    const int chunk = 8 * qpg_n;
    const int full = (G / 8) * chunk;
    int g, qpg_idx;
    if (lin < full) {
      const int c = lin / chunk, r = lin % chunk;
      qpg_idx = r >> 3;
      g = c * 8 + (r & 7);
    } else {
      const int idx = lin - full;
      g = (G / 8) * 8 + idx / qpg_n;
      qpg_idx = idx % qpg_n;
    }
    blk = g / num_kv_heads;
    qh = (g % num_kv_heads) * qpg_n + qpg_idx;
  } else {
    blk = blockIdx.x / num_q_heads;
    qh = blockIdx.x % num_q_heads;
  }
  const int kvh = qh / qpg_n;
  const int q_start = block_info[blk * 4];
  const int qblock = block_info[blk * 4 + 1];
  const int kv_start = block_info[blk * 4 + 2];
  const int q_offset = block_info[blk * 4 + 3];
  const int seq_len = seq_lens[blk];

  constexpr int QTILE = 128;        // 4 waves x 32 rows
  constexpr int NTHREADS = 256;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int col = lane & 31;        // fragment column (key / out dim)
  const int half = lane >> 5;       // k-half of the fragment

  __shared__ __hip_bfloat16 k_lds[KTILE][HEAD_DIM + PAD];
  __shared__ __hip_bfloat16 v_img[KTILE * HEAD_DIM];   // v_img32 layout
  __shared__ __hip_bfloat16 p_lds[4][32][KTILE + 4];

  // ---- Q fragments: wave's 32 rows, 8 k-steps of 16 dims ---------------
  const int num_q_rows = seq_len - q_offset;
  const int q_row_in_chunk = qblock * QTILE + wave * 32 + col;
  const int q_row = q_start + min(q_row_in_chunk, num_q_rows - 1);
  mfma_bf8 q_frag[8];
#pragma unroll
  for (int ks = 0; ks < 8; ++ks)
    q_frag[ks] = load_bf8(q + (size_t)q_row * q_stride +
                          (size_t)qh * HEAD_DIM + ks * 16 + half * 8);

  // softmax state: the lane's 16 rows r(i) = (i%4) + 4*half + 8*(i/4)
  float m_run[16], l_run[16];
  mfma_f16 acc_o[4];                // O[32 x 128]: 4 col-tiles of 32 dims
#pragma unroll
  for (int i = 0; i < 16; ++i) { m_run[i] = NEG_INF; l_run[i] = 0.f; }
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int i = 0; i < 16; ++i) acc_o[dt][i] = 0.f;

  const int block_q_max = min(q_offset + qblock * QTILE + QTILE - 1,
                              seq_len - 1);
  const int wave_q_max = min(q_offset + qblock * QTILE + wave * 32 + 31,
                             seq_len - 1);
  const int nktiles = block_q_max / KTILE + 1;

  constexpr int PIECES = KTILE * (HEAD_DIM / 8) / NTHREADS;   // = 4
  uint4 kr0 = {}, kr1 = {}, kr2 = {}, kr3 = {};
  uint4 vr0 = {}, vr1 = {}, vr2 = {}, vr3 = {};
  auto issue_loads32 = [&](int kt2) {
    PF_LD(0, kt2, kr0, vr0);
    PF_LD(1, kt2, kr1, vr1);
    PF_LD(2, kt2, kr2, vr2);
    PF_LD(3, kt2, kr3, vr3);
  };
#define PF_ST32(s, kr, vr)                                                  \
  do {                                                                      \
    const int key = (tid + (s) * NTHREADS) >> 4;                            \
    const int chunk = ((tid + (s) * NTHREADS) & 15) * 8;                    \
    *reinterpret_cast<uint4*>(&k_lds[key][chunk]) = kr;                     \
    *reinterpret_cast<uint4*>(&v_img[v_img32_off(key, chunk)]) = vr;        \
  } while (0)
  auto write_tile32 = [&]() {
    PF_ST32(0, kr0, vr0);
    PF_ST32(1, kr1, vr1);
    PF_ST32(2, kr2, vr2);
    PF_ST32(3, kr3, vr3);
  };

  if (DB) {
    issue_loads32(0);
    write_tile32();
  }

  for (int kt = 0; kt < nktiles; ++kt) {
    const int k_base = kt * KTILE;
    if (DB) {
      __syncthreads();
      if (kt + 1 < nktiles)
        issue_loads32(kt + 1);
    } else {
      __syncthreads();
      issue_loads32(kt);
      write_tile32();
      __syncthreads();
    }

    const bool compute_this = k_base <= wave_q_max;
    if (compute_this) {

    // ---- S = Q K^T : 2 col-tiles (32 keys) x 8 k-steps ------------------
    mfma_f16 s[2];
#pragma unroll
    for (int ct = 0; ct < 2; ++ct) {
#pragma unroll
      for (int i = 0; i < 16; ++i) s[ct][i] = 0.f;
#pragma unroll
      for (int ks = 0; ks < 8; ++ks) {
        mfma_bf8 kf = lds_bf8(
            &k_lds[ct * 32 + col][ks * 16 + half * 8]);
        s[ct] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            q_frag[ks], kf, s[ct], 0, 0, 0);
      }
    }

    // ---- online softmax: 16 rows per lane, 32-lane row groups -----------
#pragma unroll
    for (int i = 0; i < 16; ++i) {
      const int r = (i % 4) + 4 * half + 8 * (i / 4);
      const int q_pos = q_offset + qblock * QTILE + wave * 32 + r;
      float p0 = s[0][i] * scale, p1 = s[1][i] * scale;
      const int k0 = k_base + col, k1 = k_base + 32 + col;
      if (k0 > q_pos || k0 >= seq_len || q_pos >= seq_len) p0 = NEG_INF;
      if (k1 > q_pos || k1 >= seq_len || q_pos >= seq_len) p1 = NEG_INF;
      float row_max = half_wave_max(fmaxf(p0, p1));
      const float m_new = fmaxf(m_run[i], row_max);
      const float alpha = __expf(m_run[i] - m_new);
      p0 = (p0 <= NEG_INF) ? 0.f : __expf(p0 - m_new);
      p1 = (p1 <= NEG_INF) ? 0.f : __expf(p1 - m_new);
      l_run[i] = l_run[i] * alpha + p0 + p1;     // summed in epilogue
      m_run[i] = m_new;
#pragma unroll
      for (int dt = 0; dt < 4; ++dt) acc_o[dt][i] *= alpha;
      p_lds[wave][r][col] = f2bf(p0);
      p_lds[wave][r][32 + col] = f2bf(p1);
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    // ---- O += P V : 4 col-tiles x 4 key-steps ---------------------------
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      mfma_bf8 pa = lds_bf8(&p_lds[wave][col][ks * 16 + half * 8]);
      const unsigned vbase = (unsigned)(unsigned long long)(
          &v_img[ks * 2048 + (lane >> 4) * 64 + (lane & 15) * 4]);
#pragma unroll
      for (int dt = 0; dt < 4; ++dt) {
        unsigned long long vlo, vhi;
        asm volatile(
            "ds_read_b64_tr_b16 %0, %2 offset:%3\n\t"
            "ds_read_b64_tr_b16 %1, %2 offset:%4\n\t"
            "s_waitcnt lgkmcnt(0)"
            : "=&v"(vlo), "=&v"(vhi)
            : "v"(vbase), "i"(dt * 1024), "i"(dt * 1024 + 512)
            : "memory");
        union { struct { unsigned long long lo, hi; } u; mfma_bf8 vf; } vv;
        vv.u.lo = vlo;
        vv.u.hi = vhi;
        acc_o[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            pa, vv.vf, acc_o[dt], 0, 0, 0);
      }
    }
    }  // compute_this

    if (DB && kt + 1 < nktiles) {
      __syncthreads();
      write_tile32();
    }
  }

  // ---- epilogue ---------------------------------------------------------
#pragma unroll
  for (int i = 0; i < 16; ++i) l_run[i] = half_wave_sum(l_run[i]);
#pragma unroll
  for (int i = 0; i < 16; ++i) {
    const int r = (i % 4) + 4 * half + 8 * (i / 4);
    const int q_row_out = qblock * QTILE + wave * 32 + r;
    if (q_row_out >= num_q_rows) continue;
    const float inv_l = l_run[i] > 0.f ? 1.f / l_run[i] : 0.f;
    __hip_bfloat16* orow =
        out + ((size_t)(q_start + q_row_out) * num_q_heads + qh) * HEAD_DIM;
#pragma unroll
    for (int dt = 0; dt < 4; ++dt)
      orow[dt * 32 + col] = f2bf(acc_o[dt][i] * inv_l);
  }
}

}  // namespace

extern "C" {

void launch_prefill_attention(void* out, const void* q, const void* k,
                              const void* v, const void* block_info,
                              const void* seq_lens, float scale, int nblocks,
                              int num_q_heads, int num_kv_heads, int swz,
                              int q_stride, int kv_stride,
                              hipStream_t stream) {
  dim3 grid(nblocks * num_q_heads);
  // swz bit 0: XCD-affine GQA swizzle; bit 1: T14 double-buffer staging;
  // bit 2: 8-wave blocks (host must build block_info with qtile=128);
  // bit 3: 32x32x16 MFMA kernel (4 waves, qtile=128)
  if (swz & 8) {
#define PF32_LAUNCH(S, D)                                                   \
    hipLaunchKernelGGL((prefill_attn32_kernel<S, D>), grid, dim3(256), 0,   \
                       stream,                                              \
                       (__hip_bfloat16*)out, (const __hip_bfloat16*)q,      \
                       (const __hip_bfloat16*)k, (const __hip_bfloat16*)v,  \
                       (const int*)block_info, (const int*)seq_lens, scale, \
                       num_q_heads, num_kv_heads, q_stride, kv_stride,      \
                       nblocks)
    switch (swz & 3) {
      case 0: PF32_LAUNCH(0, 0); break;
      case 1: PF32_LAUNCH(1, 0); break;
      case 2: PF32_LAUNCH(0, 1); break;
      case 3: PF32_LAUNCH(1, 1); break;
    }
#undef PF32_LAUNCH
    return;
  }
#define PF_LAUNCH(S, D, NW)                                                 \
  hipLaunchKernelGGL((prefill_attn_kernel<S, D, NW>), grid, dim3(NW * 64),  \
                     0, stream,                                             \
                     (__hip_bfloat16*)out, (const __hip_bfloat16*)q,        \
                     (const __hip_bfloat16*)k, (const __hip_bfloat16*)v,    \
                     (const int*)block_info, (const int*)seq_lens, scale,   \
                     num_q_heads, num_kv_heads, q_stride, kv_stride, nblocks)
  switch (swz & 7) {
    case 0: PF_LAUNCH(0, 0, 4); break;
    case 1: PF_LAUNCH(1, 0, 4); break;
    case 2: PF_LAUNCH(0, 1, 4); break;
    case 3: PF_LAUNCH(1, 1, 4); break;
    case 4: PF_LAUNCH(0, 0, 8); break;
    case 5: PF_LAUNCH(1, 0, 8); break;
    case 6: PF_LAUNCH(0, 1, 8); break;
    case 7: PF_LAUNCH(1, 1, 8); break;
  }
#undef PF_LAUNCH
}

}  // extern "C"
