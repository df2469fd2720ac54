// Paged flash-decode attention (gfx950) — single query token per sequence.
//
// HBM-bound: the cost is streaming the sequence's paged K/V once.  Layout
// and access pattern are designed for that: K/V rows are head_dim=128
// contiguous bf16 (256 B), read as 16-byte lane vectors by 16-lane groups
// (4 keys per wave-instruction), GQA handled by computing all q-heads of a
// kv-head in one pass so K/V bytes are read exactly once.  Split-KV
// partitions ("flash-decoding") keep >=256 workgroups resident at small
// batch; partials merge in a combine kernel via the (m, l) log-sum-exp
// algebra.
//
// Capability analog: the decode-side paged attention of the engines the
// reference orchestrates (SURVEY §2.3 decode engine row).
#include "common.h"

namespace {

constexpr int HEAD_DIM = 128;
constexpr float NEG_INF = -1e30f;

// QPG = q heads per kv head (GQA group). One block = (seq, kv_head, split).
// 4 waves; each wave covers 4 keys per iteration (16-lane groups, 16 B/lane),
// 1-ahead K/V prefetch.
// VARIANT 0: f32 FMA scores (more VGPRs, 3-4 waves/SIMD);
// VARIANT 1: packed-bf16 v_dot2 scores (96 VGPRs, 5 waves/SIMD for QPG<=4).
// Runtime-selected (RBG_DECODE_VARIANT) for within-probe A/B.
template <int QPG, int VARIANT>
__global__ __launch_bounds__(256,
    (VARIANT == 3) ? 4 : ((VARIANT >= 1 && QPG <= 4) ? 5 : 2))
// V3 doubles the in-flight K/V state: 4 waves/SIMD (128-VGPR budget) so
// the extra pairs stay in registers instead of spilling
void decode_attn_kernel(
    float* __restrict__ partial_o,        // [splits, seqs, QH, D]
    float* __restrict__ partial_ml,       // [splits, seqs, QH, 2]
    __hip_bfloat16* __restrict__ out,     // [seqs, QH, D] (splits==1 path)
    const __hip_bfloat16* __restrict__ q, // [seqs, QH, D]
    const __hip_bfloat16* __restrict__ key_cache,  // [pages, KVH, page, D]
    const __hip_bfloat16* __restrict__ val_cache,
    const int* __restrict__ block_tables, // [seqs, max_pages]
    const int* __restrict__ context_lens, // [seqs]
    const float scale, const int num_kv_heads, const int page_size,
    const int max_pages, const int num_splits, const int q_stride) {
  const int kvh = blockIdx.x;
  const int seq = blockIdx.y;
  const int split = blockIdx.z;
  const int num_q_heads = num_kv_heads * QPG;
  const int ctx = context_lens[seq];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int group = lane >> 4;        // which of the wave's 4 keys
  const int gl = lane & 15;           // lane within the 16-lane group
  const int dbase = gl * 8;           // this lane's 8 dims

  // split bounds (rounded to 16-key chunks)
  const int chunk = 16;
  const int nchunks = (ctx + chunk - 1) / chunk;
  const int per_split = (nchunks + num_splits - 1) / num_splits;
  const int key_begin = split * per_split * chunk;
  const int key_end = min(ctx, (split + 1) * per_split * chunk);

  // q fragment: VARIANT 1 keeps it packed bf16 (v_dot2 pairs, half the
  // registers); VARIANT 0 pre-converts to f32 with the scale folded in
  typedef __attribute__((ext_vector_type(2))) __bf16 bfpair;
  bfpair qp[VARIANT >= 1 ? QPG : 1][4];
  float qf[VARIANT == 0 ? QPG : 1][8];
  {
    const __hip_bfloat16* qrow =
        q + (size_t)seq * q_stride + (size_t)kvh * QPG * HEAD_DIM;
#pragma unroll
    for (int h = 0; h < QPG; ++h) {
      Bf16x8U qv;
      qv.u = *reinterpret_cast<const uint4*>(qrow + h * HEAD_DIM + dbase);
      if (VARIANT >= 1) {
#pragma unroll
        for (int j = 0; j < 4; ++j)
          qp[h][j] = reinterpret_cast<const bfpair*>(&qv)[j];
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) qf[h][j] = bf2f(qv.e[j]) * scale;
      }
    }
  }

  float m[QPG], l[QPG], acc[QPG][8];
#pragma unroll
  for (int h = 0; h < QPG; ++h) {
    m[h] = NEG_INF;
    l[h] = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[h][j] = 0.f;
  }

  const int* btab = block_tables + (size_t)seq * max_pages;
  // page_size is a power of two (engine config); shift/mask beats idiv
  const int ps_shift = 31 - __clz(page_size);
  const int ps_mask = page_size - 1;

  auto row_offset = [&](int base) -> size_t {
    const int key = base + group;
    const int kslot = key < key_end ? key : key_begin;  // clamp: safe addr
    const int page = btab[kslot >> ps_shift];
    return (((size_t)page * num_kv_heads + kvh) * page_size +
            (kslot & ps_mask)) * HEAD_DIM;
  };

  // One 4-key quad of online-softmax work (scores via 16-lane-group dot
  // reduction, tile max/psum across the wave's quad, V accumulate).
  auto process4 = [&](const Bf16x8U& kv, const Bf16x8U& vv, bool valid) {
    const bfpair* kp = reinterpret_cast<const bfpair*>(&kv);
    float kf[VARIANT == 0 ? 8 : 1];
    if (VARIANT == 0) {
#pragma unroll
      for (int j = 0; j < 8; ++j) kf[j] = bf2f(kv.e[j]);
    }
    float p[QPG];
    float tile_max[QPG];
#pragma unroll
    for (int h = 0; h < QPG; ++h) {
      float s = 0.f;
      if (VARIANT >= 1) {
#pragma unroll
        for (int j = 0; j < 4; ++j)
          s = __builtin_amdgcn_fdot2_f32_bf16(qp[h][j], kp[j], s, false);
        s *= scale;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) s += qf[h][j] * kf[j];
      }
      s = group16_sum(s);             // full dot across the 16-lane group
      if (!valid) s = NEG_INF;
      float tm = s;
      tm = fmaxf(tm, __shfl_xor(tm, 16, 64));
      tm = fmaxf(tm, __shfl_xor(tm, 32, 64));
      tile_max[h] = tm;
      p[h] = s;
    }
#pragma unroll
    for (int h = 0; h < QPG; ++h) {
      const float m_new = fmaxf(m[h], tile_max[h]);
      const float alpha = __expf(m[h] - m_new);
      const float pv = valid ? __expf(p[h] - m_new) : 0.f;
      float psum = pv;
      psum += __shfl_xor(psum, 16, 64);
      psum += __shfl_xor(psum, 32, 64);
      l[h] = l[h] * alpha + psum;
      m[h] = m_new;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        acc[h][j] = acc[h][j] * alpha + pv * bf2f(vv.e[j]);
    }
  };

  auto load_pair = [&](int b, Bf16x8U& kd, Bf16x8U& vd) {
    const size_t off = row_offset(b);
    kd.u = *reinterpret_cast<const uint4*>(key_cache + off + dbase);
    vd.u = *reinterpret_cast<const uint4*>(val_cache + off + dbase);
  };

  // 4 keys per wave-iteration with 1-ahead data prefetch.  (Wider 8-key
  // and 2-ahead-ring variants measured null-to-negative —
  // profiles/decode_breakdown.md.)  VARIANT 2 additionally computes the
  // page-table address one MORE iteration ahead so the btab read never
  // serializes in front of the K/V loads.
  if (VARIANT == 2) {
    int base = key_begin + wave * 4;
    Bf16x8U k_pref, v_pref;
    size_t off_next = 0;
    if (base < key_end) {
      const size_t off0 = row_offset(base);
      k_pref.u = *reinterpret_cast<const uint4*>(key_cache + off0 + dbase);
      v_pref.u = *reinterpret_cast<const uint4*>(val_cache + off0 + dbase);
      off_next = row_offset(base + 16);
    }
    for (; base < key_end; base += 16) {
      const bool valid = base + group < key_end;
      Bf16x8U kv = k_pref, vv = v_pref;
      k_pref.u = *reinterpret_cast<const uint4*>(key_cache + off_next + dbase);
      v_pref.u = *reinterpret_cast<const uint4*>(val_cache + off_next + dbase);
      off_next = row_offset(base + 32);
      process4(kv, vv, valid);
    }
  } else if (VARIANT == 3) {
    // 2-ahead prefetch with SCALAR named register pairs, unrolled by 2.
    // An earlier 2-ahead used a ring array indexed by (i & 1) — which the
    // compiler demoted to scratch memory (docs/cdna_lessons.md §1), so it
    // measured -22%; with named pairs both iterations' loads stay in
    // flight and the per-iteration vmcnt slack doubles.
    int base = key_begin + wave * 4;
    Bf16x8U k0, v0, k1, v1;
    load_pair(base, k0, v0);          // row_offset clamps out-of-range
    load_pair(base + 16, k1, v1);
    for (; base < key_end; base += 32) {
      const bool valid0 = base + group < key_end;
      Bf16x8U a = k0, b = v0;
      load_pair(base + 32, k0, v0);
      process4(a, b, valid0);
      const bool valid1 = base + 16 + group < key_end;
      Bf16x8U c = k1, d = v1;
      load_pair(base + 48, k1, v1);
      process4(c, d, valid1);
    }
  } else {
    int base = key_begin + wave * 4;
    Bf16x8U k_pref, v_pref;
    if (base < key_end)
      load_pair(base, k_pref, v_pref);
    for (; base < key_end; base += 16) {
      const bool valid = base + group < key_end;
      Bf16x8U kv = k_pref, vv = v_pref;
      // unconditional clamped prefetch (row_offset clamps): a branch
      // around the loads forces a vmcnt drain per iteration (guide trap 4c)
      load_pair(base + 16, k_pref, v_pref);
      process4(kv, vv, valid);
    }
  }

  // fold the 4 key-groups of the wave (same dims, disjoint keys, same m/l)
#pragma unroll
  for (int h = 0; h < QPG; ++h)
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      acc[h][j] += __shfl_xor(acc[h][j], 16, 64);
      acc[h][j] += __shfl_xor(acc[h][j], 32, 64);
    }

  // cross-wave combine through LDS
  __shared__ float lds_acc[4][QPG][HEAD_DIM];
  __shared__ float lds_ml[4][QPG][2];
  if (lane < 16) {
#pragma unroll
    for (int h = 0; h < QPG; ++h) {
#pragma unroll
      for (int j = 0; j < 8; ++j) lds_acc[wave][h][dbase + j] = acc[h][j];
      if (gl == 0) {
        lds_ml[wave][h][0] = m[h];
        lds_ml[wave][h][1] = l[h];
      }
    }
  }
  __syncthreads();

  // threads cover (head, dim): QPG*128 outputs
  for (int i = tid; i < QPG * HEAD_DIM; i += blockDim.x) {
    const int h = i / HEAD_DIM, d = i % HEAD_DIM;
    float m_star = NEG_INF;
#pragma unroll
    for (int w = 0; w < 4; ++w) m_star = fmaxf(m_star, lds_ml[w][h][0]);
    float o = 0.f, lsum = 0.f;
#pragma unroll
    for (int w = 0; w < 4; ++w) {
      const float f = __expf(lds_ml[w][h][0] - m_star);
      o += lds_acc[w][h][d] * f;
      lsum += lds_ml[w][h][1] * f;
    }
    const int qh = kvh * QPG + h;
    if (num_splits == 1) {
      out[((size_t)seq * num_q_heads + qh) * HEAD_DIM + d] =
          f2bf(lsum > 0.f ? o / lsum : 0.f);
    } else {
      partial_o[(((size_t)split * gridDim.y + seq) * num_q_heads + qh) *
                    HEAD_DIM + d] = o;
      if (d == 0) {
        float* ml = partial_ml +
            (((size_t)split * gridDim.y + seq) * num_q_heads + qh) * 2;
        ml[0] = m_star;
        ml[1] = lsum;
      }
    }
  }
}


// V image offset for the decode PV tr-read path (32-key pair image):
// identical sub-tiling to attn_prefill.hip's v_img_off at ks == 0
DEV_INLINE int v_img_off(int key, int dim) {
  const int q = key >> 3, jj = key & 7;
  return (dim >> 4) * 512 + (jj >> 2) * 256 + q * 64 + (jj & 3) * 16 +
         (dim & 15);
}


// ---------------------------------------------------------------------------
// MFMA-tiled decode (VARIANT 4 dispatch): one wave per sequence, 32-key
// page-pair tiles on v_mfma_f32_16x16x32_bf16 — the prefill kernel's
// engine applied to decode.  The dot2 kernel (V1) is latency-bound on its
// serial cross-lane shuffle chain (a group16 reduction round every 4
// keys, SQ_WAIT_ANY 68%); here one MFMA chain scores 32 keys and the
// reduction round runs once per PAIR of pages (8x fewer), with PV
// through the same 32-k fragment + ds_read_b64_tr_b16 V image as
// attn_prefill.hip.
//
// Layout notes:
//  * page_size must be 16: a KV page slice [16 tokens][128 d] is 4 KB
//    contiguous; two pages form one 32-key MFMA step.
//  * q A-fragment rows = the kv-head's QPG query heads (rows >= QPG
//    replicate the last head; their outputs are never written).
//  * S is C-layout: key col = lane&15, q row = 4*(lane>>4)+reg; the two
//    pages' columns combine lane-locally before ONE group16 max/sum.
//  * 2-wave workgroups, ~18 KB LDS per wave -> 4 WGs (8 waves) per CU;
//    MFMA depth inside the wave covers the low waves/SIMD (the same
//    budget shape as the 8-wave prefill kernel).
// DIET (variant 5): the round-2 register diet targeting 4 waves/SIMD
// (docs/ROUND2_DESIGNS.md §1a+1b).  Two changes vs variant 4:
//  (a) q fragments are RELOADED from global per page pair instead of held
//      in 16 persistent VGPRs — the rows are the same cache lines every
//      pair, so after the first touch they come from L1.  Issued BEFORE
//      the V stage so the compiler's dependency waitcnt for the first S
//      MFMA is vmcnt(8) (V still in flight), not a pipeline drain.
//  (b) V stages through 4 of its own registers + 4 of K's (ka0-3 are dead
//      between DM_WRITE_K and DM_ISSUE_K(pg+2)) — 16 fewer VGPRs.  The
//      round-67 FULL merge (one 8-set) hit in-loop spills and lost 27%;
//      this partial merge keeps issue order WAR-safe without aliasing
//      pressure.  Verified ScratchSize==0 at compile time.
template <int QPG, int DIET>
__global__ __launch_bounds__(128, DIET ? 4 : 3) void decode_attn_mfma_kernel(
    float* __restrict__ partial_o,        // [splits, seqs, QH, D]
    float* __restrict__ partial_ml,       // [splits, seqs, QH, 2]
    __hip_bfloat16* __restrict__ out,     // [seqs, QH, D] (splits==1 path)
    const __hip_bfloat16* __restrict__ q, // [seqs, QH, D]
    const __hip_bfloat16* __restrict__ key_cache,  // [pages, KVH, 16, D]
    const __hip_bfloat16* __restrict__ val_cache,
    const int* __restrict__ block_tables, // [seqs, max_pages]
    const int* __restrict__ context_lens, // [seqs]
    const float scale, const int num_kv_heads, const int max_pages,
    const int num_splits, const int num_seqs, const int q_stride) {
  typedef __attribute__((ext_vector_type(8))) __bf16 dm_bf8;
  const int kvh = blockIdx.x;
  const int wave = threadIdx.x >> 6;
  const int seq = blockIdx.y * 2 + wave;
  const int split = blockIdx.z;
  const int lane = threadIdx.x & 63;
  const int gl = lane & 15;
  const int gslice = lane >> 4;
  const int num_q_heads = num_kv_heads * QPG;

  // per-wave LDS (no cross-wave sharing, no barriers).  K and V SHARE one
  // buffer: the pair's K panel serves the S MFMAs, then the V image
  // overwrites it for PV (same-wave DS ops retire in order, so the
  // write-after-read needs no fence) — halving LDS doubles resident
  // waves.  V is fully rewritten every pair (tail pages clamp to real
  // pages and are masked through P=0), so no zero-init is needed.
  constexpr int KP = 132;                        // k panel pitch (bf16)
  __shared__ __hip_bfloat16 kv_all[2][32 * KP];  // K panel / V tr image
  __shared__ __hip_bfloat16 p_lds_all[2][16 * 40];
  __hip_bfloat16* k_lds = kv_all[wave];
  __hip_bfloat16* v_img = kv_all[wave];
  __hip_bfloat16* p_lds = p_lds_all[wave];
  if (seq >= num_seqs) return;
  const int ctx = context_lens[seq];

  const int chunk = 16;
  const int nchunks = (ctx + chunk - 1) / chunk;
  const int per_split = (nchunks + num_splits - 1) / num_splits;
  const int key_begin = split * per_split * chunk;
  const int key_end = min(ctx, (split + 1) * per_split * chunk);
  const int pg_begin = key_begin >> 4;
  const int pg_end = (key_end + 15) >> 4;

  // q fragments: row = head (clamped), 4 k-steps of 32 dims.
  // DIET: only the row pointer persists; fragments reload per pair (L1).
  const __hip_bfloat16* qrow =
      q + (size_t)seq * q_stride +
      (size_t)(kvh * QPG + min(gl, QPG - 1)) * HEAD_DIM;
  dm_bf8 qa[4];
  if constexpr (!DIET) {
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      union { uint4 u; dm_bf8 v; } cvt;
      cvt.u = *reinterpret_cast<const uint4*>(qrow + ks * 32 + gslice * 8);
      qa[ks] = cvt.v;
    }
  }

  const int* btab = block_tables + (size_t)seq * max_pages;
  // staging: lane covers K/V key row = lane>>1 (32 rows), 64-elem chunk
  // = (lane&1)*64; 2 lanes x 128 B = one contiguous page row
  const int srow = lane >> 1, schunk = (lane & 1) * 64;
  const int pg_last = pg_end - 1;
  auto page_base = [&](int pg, int local) -> size_t {
    // pair page `local` (0/1): clamp to the last real page
    const int p = btab[min(pg + local, pg_last)];
    return (((size_t)p * num_kv_heads + kvh) * 16 + (srow & 15)) * HEAD_DIM +
           schunk;
  };
  const int myloc = srow >> 4;     // which page of the pair this lane stages

  // SCALAR staging registers (docs/cdna_lessons.md §1: local arrays are
  // demoted to scratch memory, round-tripping every staged page via HBM).
  // Each lane owns HALF a page row = 64 elements = 8 x 16 B pieces.
  uint4 ka0, ka1, ka2, ka3, ka4, ka5, ka6, ka7;
  uint4 va0, va1, va2, va3, va4, va5, va6, va7;
#define DM_LOAD8(base, r0, r1, r2, r3, r4, r5, r6, r7, off)                 \
  do {                                                                      \
    r0 = *reinterpret_cast<const uint4*>(base + off);                       \
    r1 = *reinterpret_cast<const uint4*>(base + off + 8);                   \
    r2 = *reinterpret_cast<const uint4*>(base + off + 16);                  \
    r3 = *reinterpret_cast<const uint4*>(base + off + 24);                  \
    r4 = *reinterpret_cast<const uint4*>(base + off + 32);                  \
    r5 = *reinterpret_cast<const uint4*>(base + off + 40);                  \
    r6 = *reinterpret_cast<const uint4*>(base + off + 48);                  \
    r7 = *reinterpret_cast<const uint4*>(base + off + 56);                  \
  } while (0)
#define DM_ISSUE_K(pg) DM_LOAD8(key_cache, ka0, ka1, ka2, ka3, ka4, ka5,    \
                                ka6, ka7, page_base(pg, myloc))
#define DM_ISSUE_V(pg) DM_LOAD8(val_cache, va0, va1, va2, va3, va4, va5,    \
                                va6, va7, page_base(pg, myloc))
#define DM_WRITE_K()                                                        \
  do {                                                                      \
    *reinterpret_cast<uint4*>(&k_lds[srow * KP + schunk]) = ka0;            \
    *reinterpret_cast<uint4*>(&k_lds[srow * KP + schunk + 8]) = ka1;        \
    *reinterpret_cast<uint4*>(&k_lds[srow * KP + schunk + 16]) = ka2;       \
    *reinterpret_cast<uint4*>(&k_lds[srow * KP + schunk + 24]) = ka3;       \
    *reinterpret_cast<uint4*>(&k_lds[srow * KP + schunk + 32]) = ka4;       \
    *reinterpret_cast<uint4*>(&k_lds[srow * KP + schunk + 40]) = ka5;       \
    *reinterpret_cast<uint4*>(&k_lds[srow * KP + schunk + 48]) = ka6;       \
    *reinterpret_cast<uint4*>(&k_lds[srow * KP + schunk + 56]) = ka7;       \
  } while (0)
#define DM_WRITE_V()                                                        \
  do {                                                                      \
    *reinterpret_cast<uint4*>(&v_img[v_img_off(srow, schunk)]) = va0;       \
    *reinterpret_cast<uint4*>(&v_img[v_img_off(srow, schunk + 8)]) = va1;   \
    *reinterpret_cast<uint4*>(&v_img[v_img_off(srow, schunk + 16)]) = va2;  \
    *reinterpret_cast<uint4*>(&v_img[v_img_off(srow, schunk + 24)]) = va3;  \
    *reinterpret_cast<uint4*>(&v_img[v_img_off(srow, schunk + 32)]) = va4;  \
    *reinterpret_cast<uint4*>(&v_img[v_img_off(srow, schunk + 40)]) = va5;  \
    *reinterpret_cast<uint4*>(&v_img[v_img_off(srow, schunk + 48)]) = va6;  \
    *reinterpret_cast<uint4*>(&v_img[v_img_off(srow, schunk + 56)]) = va7;  \
  } while (0)
  // DIET: V stages through vb0-7, declared in the loop with liveness
  // DISJOINT from the q fragments (qd dies at the last S MFMA; vb issues
  // right after), so the allocator colors them into the same registers —
  // V latency hides under the softmax shuffle chain instead of under S.
#define DM_ISSUE_V_D(pg)                                                    \
  do {                                                                      \
    const size_t vb_ = page_base(pg, myloc);                                \
    vb0 = *reinterpret_cast<const uint4*>(val_cache + vb_);                 \
    vb1 = *reinterpret_cast<const uint4*>(val_cache + vb_ + 8);             \
    vb2 = *reinterpret_cast<const uint4*>(val_cache + vb_ + 16);            \
    vb3 = *reinterpret_cast<const uint4*>(val_cache + vb_ + 24);            \
    vb4 = *reinterpret_cast<const uint4*>(val_cache + vb_ + 32);            \
    vb5 = *reinterpret_cast<const uint4*>(val_cache + vb_ + 40);            \
    vb6 = *reinterpret_cast<const uint4*>(val_cache + vb_ + 48);            \
    vb7 = *reinterpret_cast<const uint4*>(val_cache + vb_ + 56);            \
  } while (0)
#define DM_WRITE_V_D()                                                      \
  do {                                                                      \
    *reinterpret_cast<uint4*>(&v_img[v_img_off(srow, schunk)]) = vb0;       \
    *reinterpret_cast<uint4*>(&v_img[v_img_off(srow, schunk + 8)]) = vb1;   \
    *reinterpret_cast<uint4*>(&v_img[v_img_off(srow, schunk + 16)]) = vb2;  \
    *reinterpret_cast<uint4*>(&v_img[v_img_off(srow, schunk + 24)]) = vb3;  \
    *reinterpret_cast<uint4*>(&v_img[v_img_off(srow, schunk + 32)]) = vb4;  \
    *reinterpret_cast<uint4*>(&v_img[v_img_off(srow, schunk + 40)]) = vb5;  \
    *reinterpret_cast<uint4*>(&v_img[v_img_off(srow, schunk + 48)]) = vb6;  \
    *reinterpret_cast<uint4*>(&v_img[v_img_off(srow, schunk + 56)]) = vb7;  \
  } while (0)

  float m[4], l[4];
  f32x4 acc_o[8];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m[r] = NEG_INF; l[r] = 0.f; }
#pragma unroll
  for (int dt = 0; dt < 8; ++dt) acc_o[dt] = {0.f, 0.f, 0.f, 0.f};

  DM_ISSUE_K(pg_begin);

  for (int pg = pg_begin; pg < pg_end; pg += 2) {
    // DIET: reload q fragments BEFORE the V issue, so the dependency wait
    // in front of the first S MFMA is vmcnt(8) — V keeps streaming
    dm_bf8 qd[4];
    if constexpr (DIET) {
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        union { uint4 u; dm_bf8 v; } cvt;
        cvt.u = *reinterpret_cast<const uint4*>(qrow + ks * 32 + gslice * 8);
        qd[ks] = cvt.v;
      }
    }
    DM_WRITE_K();
    uint4 vb0, vb1, vb2, vb3, vb4, vb5, vb6, vb7;
    if constexpr (!DIET) DM_ISSUE_V(pg);  // V hides under S + softmax
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    // S = q K^T over the pair's 32 keys (B-frag: key row = gl / gl+16)
    f32x4 sA = {0.f, 0.f, 0.f, 0.f}, sB = sA;
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      union { uint4 u; dm_bf8 v; } kfA, kfB;
      kfA.u = *reinterpret_cast<const uint4*>(
          &k_lds[gl * KP + ks * 32 + gslice * 8]);
      kfB.u = *reinterpret_cast<const uint4*>(
          &k_lds[(16 + gl) * KP + ks * 32 + gslice * 8]);
      const dm_bf8 qk = DIET ? qd[ks] : qa[ks];
      sA = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qk, kfA.v, sA, 0, 0, 0);
      sB = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qk, kfB.v, sB, 0, 0, 0);
    }
    // DIET: q fragments are dead now — V stages into their registers and
    // its latency hides under the softmax shuffle chain below
    if constexpr (DIET) DM_ISSUE_V_D(pg);

    // online softmax: lane holds rows 4*gslice+r for key cols
    // (pg*16+gl, pg*16+16+gl); ONE group16 round covers both pages
    const int keyA = pg * 16 + gl;
    const int keyB = keyA + 16;
    const bool vA = keyA >= key_begin && keyA < key_end;
    const bool vB = keyB >= key_begin && keyB < key_end;
    float pA[4], pB[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float a = vA ? sA[r] * scale : NEG_INF;
      const float b = vB ? sB[r] * scale : NEG_INF;
      const float tm = group16_max(fmaxf(a, b));
      const float m_new = fmaxf(m[r], tm);
      const float alpha = __expf(m[r] - m_new);
      pA[r] = vA ? __expf(a - m_new) : 0.f;
      pB[r] = vB ? __expf(b - m_new) : 0.f;
      l[r] = l[r] * alpha + group16_sum(pA[r] + pB[r]);
      m[r] = m_new;
      acc_o[0][r] *= alpha; acc_o[1][r] *= alpha;
      acc_o[2][r] *= alpha; acc_o[3][r] *= alpha;
      acc_o[4][r] *= alpha; acc_o[5][r] *= alpha;
      acc_o[6][r] *= alpha; acc_o[7][r] *= alpha;
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      p_lds[(4 * gslice + r) * 40 + gl] = f2bf(pA[r]);
      p_lds[(4 * gslice + r) * 40 + 16 + gl] = f2bf(pB[r]);
    }
    // V overwrites the K panel: same-wave DS ordering protects the S
    // fragment reads issued above
    if constexpr (DIET) DM_WRITE_V_D();
    else DM_WRITE_V();
    DM_ISSUE_K(pg + 2);      // next pair's K hides under PV
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    // O += P V via the 32-k fragment + tr-read V image (attn_prefill PV)
    if constexpr (!DIET) {
      union { uint4 u; dm_bf8 v; } paf;
      const int pk = gslice * 8;
      paf.u.x = *reinterpret_cast<const uint*>(&p_lds[gl * 40 + pk]);
      paf.u.y = *reinterpret_cast<const uint*>(&p_lds[gl * 40 + pk + 2]);
      paf.u.z = *reinterpret_cast<const uint*>(&p_lds[gl * 40 + pk + 4]);
      paf.u.w = *reinterpret_cast<const uint*>(&p_lds[gl * 40 + pk + 6]);
      const unsigned vaddr = (unsigned)(unsigned long long)(
          &v_img[gslice * 64 + gl * 4]);
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
      __builtin_amdgcn_sched_barrier(0);
#pragma unroll
      for (int dt = 0; dt < 8; ++dt) {
        union { struct { unsigned long long lo, hi; } u; dm_bf8 vf2; } vf;
        vf.u.lo = vlo[dt];
        vf.u.hi = vhi[dt];
        acc_o[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            paf.v, vf.vf2, acc_o[dt], 0, 0, 0);
      }
    } else {
      // DIET PV: two 4-dt halves so only 16 tr-read VGPRs are live at a
      // time (the full 32-reg batch overlapped the 32 in-flight K staging
      // registers and was the register-pressure peak)
      union { uint4 u; dm_bf8 v; } paf;
      const int pk = gslice * 8;
      paf.u.x = *reinterpret_cast<const uint*>(&p_lds[gl * 40 + pk]);
      paf.u.y = *reinterpret_cast<const uint*>(&p_lds[gl * 40 + pk + 2]);
      paf.u.z = *reinterpret_cast<const uint*>(&p_lds[gl * 40 + pk + 4]);
      paf.u.w = *reinterpret_cast<const uint*>(&p_lds[gl * 40 + pk + 6]);
      const unsigned vaddr = (unsigned)(unsigned long long)(
          &v_img[gslice * 64 + gl * 4]);
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        unsigned long long vlo[4], vhi[4];
        if (half == 0) {
          asm volatile(
              "ds_read_b64_tr_b16 %0, %8\n\t"
              "ds_read_b64_tr_b16 %1, %8 offset:512\n\t"
              "ds_read_b64_tr_b16 %2, %8 offset:1024\n\t"
              "ds_read_b64_tr_b16 %3, %8 offset:1536\n\t"
              "ds_read_b64_tr_b16 %4, %8 offset:2048\n\t"
              "ds_read_b64_tr_b16 %5, %8 offset:2560\n\t"
              "ds_read_b64_tr_b16 %6, %8 offset:3072\n\t"
              "ds_read_b64_tr_b16 %7, %8 offset:3584\n\t"
              "s_waitcnt lgkmcnt(0)"
              : "=&v"(vlo[0]), "=&v"(vhi[0]), "=&v"(vlo[1]), "=&v"(vhi[1]),
                "=&v"(vlo[2]), "=&v"(vhi[2]), "=&v"(vlo[3]), "=&v"(vhi[3])
              : "v"(vaddr)
              : "memory");
        } else {
          asm volatile(
              "ds_read_b64_tr_b16 %0, %8 offset:4096\n\t"
              "ds_read_b64_tr_b16 %1, %8 offset:4608\n\t"
              "ds_read_b64_tr_b16 %2, %8 offset:5120\n\t"
              "ds_read_b64_tr_b16 %3, %8 offset:5632\n\t"
              "ds_read_b64_tr_b16 %4, %8 offset:6144\n\t"
              "ds_read_b64_tr_b16 %5, %8 offset:6656\n\t"
              "ds_read_b64_tr_b16 %6, %8 offset:7168\n\t"
              "ds_read_b64_tr_b16 %7, %8 offset:7680\n\t"
              "s_waitcnt lgkmcnt(0)"
              : "=&v"(vlo[0]), "=&v"(vhi[0]), "=&v"(vlo[1]), "=&v"(vhi[1]),
                "=&v"(vlo[2]), "=&v"(vhi[2]), "=&v"(vlo[3]), "=&v"(vhi[3])
              : "v"(vaddr)
              : "memory");
        }
        __builtin_amdgcn_sched_barrier(0);
#pragma unroll
        for (int dt = 0; dt < 4; ++dt) {
          union { struct { unsigned long long lo, hi; } u; dm_bf8 vf2; } vf;
          vf.u.lo = vlo[dt];
          vf.u.hi = vhi[dt];
          acc_o[half * 4 + dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              paf.v, vf.vf2, acc_o[half * 4 + dt], 0, 0, 0);
        }
      }
    }
  }

  // epilogue: lane holds O rows 4*gslice+r at dim cols dt*16+gl
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = 4 * gslice + r;
    if (row >= QPG) continue;
    const int qh = kvh * QPG + row;
    if (num_splits == 1) {
      const float inv_l = l[r] > 0.f ? 1.f / l[r] : 0.f;
#pragma unroll
      for (int dt = 0; dt < 8; ++dt)
        out[((size_t)seq * num_q_heads + qh) * HEAD_DIM + dt * 16 + gl] =
            f2bf(acc_o[dt][r] * inv_l);
    } else {
#pragma unroll
      for (int dt = 0; dt < 8; ++dt)
        partial_o[(((size_t)split * num_seqs + seq) * num_q_heads + qh) *
                      HEAD_DIM + dt * 16 + gl] = acc_o[dt][r];
      if (gl == 0) {
        float* ml = partial_ml +
            (((size_t)split * num_seqs + seq) * num_q_heads + qh) * 2;
        ml[0] = m[r];
        ml[1] = l[r];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Producer/consumer MFMA decode (variant 6, ROUND2 §1c): TWO waves per
// sequence — wave 1 (producer) owns all global->LDS staging, wave 0
// (consumer) owns all math (S, softmax, PV) and never issues a global
// load, so the math pipeline never parks on vmcnt.  Unlike a double
// buffer, K and V keep ONE LDS panel each with a fine-grained flag
// interlock (monotonic pair counters in LDS):
//   producer pair i: wait cons_s >= i-1  -> write K_i, flag prod_k = i,
//                    issue K_{i+1};
//                    wait cons_pv >= i-1 -> write V_i, flag prod_v = i,
//                    issue V_{i+1}
//   consumer pair i: wait prod_k >= i -> S + softmax, flag cons_s = i;
//                    wait prod_v >= i -> PV, flag cons_pv = i
// LDS stays ~17 KB/WG (like variant 4's shared panel) and the producer
// runs a full pair ahead.  Cross-wave LDS ordering: panel ds_writes are
// drained (lgkmcnt(0)) before each flag write.
template <int QPG>
__global__ __launch_bounds__(128, 4) void decode_attn_pc_kernel(
    float* __restrict__ partial_o,        // [splits, seqs, QH, D]
    float* __restrict__ partial_ml,       // [splits, seqs, QH, 2]
    __hip_bfloat16* __restrict__ out,     // [seqs, QH, D] (splits==1 path)
    const __hip_bfloat16* __restrict__ q, // [seqs, QH, D]
    const __hip_bfloat16* __restrict__ key_cache,  // [pages, KVH, 16, D]
    const __hip_bfloat16* __restrict__ val_cache,
    const int* __restrict__ block_tables, // [seqs, max_pages]
    const int* __restrict__ context_lens, // [seqs]
    const float scale, const int num_kv_heads, const int max_pages,
    const int num_splits, const int num_seqs, const int q_stride) {
  typedef __attribute__((ext_vector_type(8))) __bf16 dm_bf8;
  const int kvh = blockIdx.x;
  const int seq = blockIdx.y;
  const int split = blockIdx.z;
  const int wave = threadIdx.x >> 6;     // 0 = consumer, 1 = producer
  const int lane = threadIdx.x & 63;
  const int gl = lane & 15;
  const int gslice = lane >> 4;
  const int num_q_heads = num_kv_heads * QPG;

  constexpr int KP = 132;
  __shared__ __hip_bfloat16 k_lds[32 * KP];
  __shared__ __hip_bfloat16 v_img[32 * 128];
  __shared__ __hip_bfloat16 p_lds[16 * 40];
  __shared__ int flags[4];   // prod_k, prod_v, cons_s, cons_pv (pair idx)
  if (threadIdx.x == 0) {
    flags[0] = -1; flags[1] = -1; flags[2] = -1; flags[3] = -1;
  }
  __syncthreads();
  volatile int* f_prod_k = &flags[0];
  volatile int* f_prod_v = &flags[1];
  volatile int* f_cons_s = &flags[2];
  volatile int* f_cons_pv = &flags[3];

  const int ctx = context_lens[seq];
  const int chunk = 16;
  const int nchunks = (ctx + chunk - 1) / chunk;
  const int per_split = (nchunks + num_splits - 1) / num_splits;
  const int key_begin = split * per_split * chunk;
  const int key_end = min(ctx, (split + 1) * per_split * chunk);
  const int pg_begin = key_begin >> 4;
  const int pg_end = (key_end + 15) >> 4;
  const int npairs = max(0, (pg_end - pg_begin + 1) >> 1);

  const int* btab = block_tables + (size_t)seq * max_pages;
  // ctx >= 1 so pg_end >= 1; for empty splits the clamp keeps block-table
  // reads at the last REAL page (same convention as variant 4)
  const int pg_last = pg_end - 1;

#define PC_SPIN(fl, want)                                                    \
  while (*(fl) < (want)) __builtin_amdgcn_s_sleep(2)

  if (wave == 1) {
    // ---- producer -------------------------------------------------------
    const int srow = lane >> 1, schunk = (lane & 1) * 64;
    const int myloc = srow >> 4;
    auto page_base = [&](int pg, int local) -> size_t {
      const int p = btab[min(pg + local, pg_last)];
      return (((size_t)p * num_kv_heads + kvh) * 16 + (srow & 15)) *
                 HEAD_DIM + schunk;
    };
    uint4 ka0, ka1, ka2, ka3, ka4, ka5, ka6, ka7;
    uint4 va0, va1, va2, va3, va4, va5, va6, va7;
    if (npairs > 0) {
      DM_LOAD8(key_cache, ka0, ka1, ka2, ka3, ka4, ka5, ka6, ka7,
               page_base(pg_begin, myloc));
      DM_LOAD8(val_cache, va0, va1, va2, va3, va4, va5, va6, va7,
               page_base(pg_begin, myloc));
    }
    for (int i = 0; i < npairs; ++i) {
      const int pg_next = pg_begin + 2 * (i + 1);
      PC_SPIN(f_cons_s, i - 1);          // K panel free
      DM_WRITE_K();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (lane == 0) *f_prod_k = i;
      if (i + 1 < npairs)
        DM_LOAD8(key_cache, ka0, ka1, ka2, ka3, ka4, ka5, ka6, ka7,
                 page_base(pg_next, myloc));
      PC_SPIN(f_cons_pv, i - 1);         // V image free
      DM_WRITE_V();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (lane == 0) *f_prod_v = i;
      if (i + 1 < npairs)
        DM_LOAD8(val_cache, va0, va1, va2, va3, va4, va5, va6, va7,
                 page_base(pg_next, myloc));
    }
    return;
  }

  // ---- consumer ---------------------------------------------------------
  dm_bf8 qa[4];
  {
    const __hip_bfloat16* qrow =
        q + (size_t)seq * q_stride +
        (size_t)(kvh * QPG + min(gl, QPG - 1)) * HEAD_DIM;
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      union { uint4 u; dm_bf8 v; } cvt;
      cvt.u = *reinterpret_cast<const uint4*>(qrow + ks * 32 + gslice * 8);
      qa[ks] = cvt.v;
    }
  }
  float m[4], l[4];
  f32x4 acc_o[8];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m[r] = NEG_INF; l[r] = 0.f; }
#pragma unroll
  for (int dt = 0; dt < 8; ++dt) acc_o[dt] = {0.f, 0.f, 0.f, 0.f};

  for (int i = 0; i < npairs; ++i) {
    const int pg = pg_begin + 2 * i;
    PC_SPIN(f_prod_k, i);
    f32x4 sA = {0.f, 0.f, 0.f, 0.f}, sB = sA;
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      union { uint4 u; dm_bf8 v; } kfA, kfB;
      kfA.u = *reinterpret_cast<const uint4*>(
          &k_lds[gl * KP + ks * 32 + gslice * 8]);
      kfB.u = *reinterpret_cast<const uint4*>(
          &k_lds[(16 + gl) * KP + ks * 32 + gslice * 8]);
      sA = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qa[ks], kfA.v, sA, 0, 0, 0);
      sB = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qa[ks], kfB.v, sB, 0, 0, 0);
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");  // frag reads done
    if (lane == 0) *f_cons_s = i;        // K panel may be overwritten

    const int keyA = pg * 16 + gl;
    const int keyB = keyA + 16;
    const bool vA = keyA >= key_begin && keyA < key_end;
    const bool vB = keyB >= key_begin && keyB < key_end;
    float pA[4], pB[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float a = vA ? sA[r] * scale : NEG_INF;
      const float b = vB ? sB[r] * scale : NEG_INF;
      const float tm = group16_max(fmaxf(a, b));
      const float m_new = fmaxf(m[r], tm);
      const float alpha = __expf(m[r] - m_new);
      pA[r] = vA ? __expf(a - m_new) : 0.f;
      pB[r] = vB ? __expf(b - m_new) : 0.f;
      l[r] = l[r] * alpha + group16_sum(pA[r] + pB[r]);
      m[r] = m_new;
      acc_o[0][r] *= alpha; acc_o[1][r] *= alpha;
      acc_o[2][r] *= alpha; acc_o[3][r] *= alpha;
      acc_o[4][r] *= alpha; acc_o[5][r] *= alpha;
      acc_o[6][r] *= alpha; acc_o[7][r] *= alpha;
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      p_lds[(4 * gslice + r) * 40 + gl] = f2bf(pA[r]);
      p_lds[(4 * gslice + r) * 40 + 16 + gl] = f2bf(pB[r]);
    }

    PC_SPIN(f_prod_v, i);
    {
      union { uint4 u; dm_bf8 v; } paf;
      const int pk = gslice * 8;
      paf.u.x = *reinterpret_cast<const uint*>(&p_lds[gl * 40 + pk]);
      paf.u.y = *reinterpret_cast<const uint*>(&p_lds[gl * 40 + pk + 2]);
      paf.u.z = *reinterpret_cast<const uint*>(&p_lds[gl * 40 + pk + 4]);
      paf.u.w = *reinterpret_cast<const uint*>(&p_lds[gl * 40 + pk + 6]);
      const unsigned vaddr = (unsigned)(unsigned long long)(
          &v_img[gslice * 64 + gl * 4]);
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        unsigned long long vlo[4], vhi[4];
        if (half == 0) {
          asm volatile(
              "ds_read_b64_tr_b16 %0, %8\n\t"
              "ds_read_b64_tr_b16 %1, %8 offset:512\n\t"
              "ds_read_b64_tr_b16 %2, %8 offset:1024\n\t"
              "ds_read_b64_tr_b16 %3, %8 offset:1536\n\t"
              "ds_read_b64_tr_b16 %4, %8 offset:2048\n\t"
              "ds_read_b64_tr_b16 %5, %8 offset:2560\n\t"
              "ds_read_b64_tr_b16 %6, %8 offset:3072\n\t"
              "ds_read_b64_tr_b16 %7, %8 offset:3584\n\t"
              "s_waitcnt lgkmcnt(0)"
              : "=&v"(vlo[0]), "=&v"(vhi[0]), "=&v"(vlo[1]), "=&v"(vhi[1]),
                "=&v"(vlo[2]), "=&v"(vhi[2]), "=&v"(vlo[3]), "=&v"(vhi[3])
              : "v"(vaddr)
              : "memory");
        } else {
          asm volatile(
              "ds_read_b64_tr_b16 %0, %8 offset:4096\n\t"
              "ds_read_b64_tr_b16 %1, %8 offset:4608\n\t"
              "ds_read_b64_tr_b16 %2, %8 offset:5120\n\t"
              "ds_read_b64_tr_b16 %3, %8 offset:5632\n\t"
              "ds_read_b64_tr_b16 %4, %8 offset:6144\n\t"
              "ds_read_b64_tr_b16 %5, %8 offset:6656\n\t"
              "ds_read_b64_tr_b16 %6, %8 offset:7168\n\t"
              "ds_read_b64_tr_b16 %7, %8 offset:7680\n\t"
              "s_waitcnt lgkmcnt(0)"
              : "=&v"(vlo[0]), "=&v"(vhi[0]), "=&v"(vlo[1]), "=&v"(vhi[1]),
                "=&v"(vlo[2]), "=&v"(vhi[2]), "=&v"(vlo[3]), "=&v"(vhi[3])
              : "v"(vaddr)
              : "memory");
        }
        __builtin_amdgcn_sched_barrier(0);
#pragma unroll
        for (int dt = 0; dt < 4; ++dt) {
          union { struct { unsigned long long lo, hi; } u; dm_bf8 vf2; } vf;
          vf.u.lo = vlo[dt];
          vf.u.hi = vhi[dt];
          acc_o[half * 4 + dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              paf.v, vf.vf2, acc_o[half * 4 + dt], 0, 0, 0);
        }
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    if (lane == 0) *f_cons_pv = i;       // V image may be overwritten
  }
#undef PC_SPIN

  // epilogue identical to variant 4
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = 4 * gslice + r;
    if (row >= QPG) continue;
    const int qh = kvh * QPG + row;
    if (num_splits == 1) {
      const float inv_l = l[r] > 0.f ? 1.f / l[r] : 0.f;
#pragma unroll
      for (int dt = 0; dt < 8; ++dt)
        out[((size_t)seq * num_q_heads + qh) * HEAD_DIM + dt * 16 + gl] =
            f2bf(acc_o[dt][r] * inv_l);
    } else {
#pragma unroll
      for (int dt = 0; dt < 8; ++dt)
        partial_o[(((size_t)split * num_seqs + seq) * num_q_heads + qh) *
                      HEAD_DIM + dt * 16 + gl] = acc_o[dt][r];
      if (gl == 0) {
        float* ml = partial_ml +
            (((size_t)split * num_seqs + seq) * num_q_heads + qh) * 2;
        ml[0] = m[r];
        ml[1] = l[r];
      }
    }
  }
}

// Merge split partials: one block per (seq, q_head).
__global__ void decode_combine_kernel(
    __hip_bfloat16* __restrict__ out,       // [seqs, QH, D]
    const float* __restrict__ partial_o,    // [splits, seqs, QH, D]
    const float* __restrict__ partial_ml,   // [splits, seqs, QH, 2]
    const int num_splits, const int num_q_heads) {
  const int seq = blockIdx.x, qh = blockIdx.y, d = threadIdx.x;
  const int seqs = gridDim.x;
  float m_star = NEG_INF;
  for (int s = 0; s < num_splits; ++s)
    m_star = fmaxf(m_star,
                   partial_ml[(((size_t)s * seqs + seq) * num_q_heads + qh) * 2]);
  float o = 0.f, lsum = 0.f;
  for (int s = 0; s < num_splits; ++s) {
    const float* ml =
        partial_ml + (((size_t)s * seqs + seq) * num_q_heads + qh) * 2;
    const float f = __expf(ml[0] - m_star);
    o += partial_o[(((size_t)s * seqs + seq) * num_q_heads + qh) * HEAD_DIM +
                   d] * f;
    lsum += ml[1] * f;
  }
  out[((size_t)seq * num_q_heads + qh) * HEAD_DIM + d] =
      f2bf(lsum > 0.f ? o / lsum : 0.f);
}

}  // namespace

extern "C" {

void launch_decode_attention(void* out, void* partial_o, void* partial_ml,
                             const void* q, const void* key_cache,
                             const void* val_cache, const void* block_tables,
                             const void* context_lens, float scale,
                             int num_seqs, int num_q_heads, int num_kv_heads,
                             int page_size, int max_pages, int num_splits,
                             int variant, int q_stride, hipStream_t stream) {
  const int qpg = num_q_heads / num_kv_heads;
  dim3 grid(num_kv_heads, num_seqs, num_splits), block(256);
#define LAUNCH_QPG(QPG, VAR)                                                  \
  hipLaunchKernelGGL((decode_attn_kernel<QPG, VAR>), grid, block, 0, stream,  \
                     (float*)partial_o, (float*)partial_ml,                   \
                     (__hip_bfloat16*)out, (const __hip_bfloat16*)q,          \
                     (const __hip_bfloat16*)key_cache,                        \
                     (const __hip_bfloat16*)val_cache,                        \
                     (const int*)block_tables, (const int*)context_lens,      \
                     scale, num_kv_heads, page_size, max_pages, num_splits,  \
                     q_stride)
  if (variant == 6 && page_size == 16) {
    dim3 pgrid(num_kv_heads, num_seqs, num_splits), pblock(128);
#define LAUNCH_PC(QPG)                                                       \
    hipLaunchKernelGGL((decode_attn_pc_kernel<QPG>), pgrid, pblock, 0,       \
                       stream, (float*)partial_o, (float*)partial_ml,        \
                       (__hip_bfloat16*)out, (const __hip_bfloat16*)q,       \
                       (const __hip_bfloat16*)key_cache,                     \
                       (const __hip_bfloat16*)val_cache,                     \
                       (const int*)block_tables, (const int*)context_lens,   \
                       scale, num_kv_heads, max_pages, num_splits,           \
                       num_seqs, q_stride)
    if (qpg == 1) LAUNCH_PC(1);
    else if (qpg == 2) LAUNCH_PC(2);
    else if (qpg == 4) LAUNCH_PC(4);
    else if (qpg == 8) LAUNCH_PC(8);
    else return;
#undef LAUNCH_PC
    if (num_splits > 1) {
      dim3 cgrid(num_seqs, num_q_heads), cblock(HEAD_DIM);
      hipLaunchKernelGGL(decode_combine_kernel, cgrid, cblock, 0, stream,
                         (__hip_bfloat16*)out, (const float*)partial_o,
                         (const float*)partial_ml, num_splits, num_q_heads);
    }
    return;
  }
  if ((variant == 4 || variant == 5) && page_size == 16) {
    dim3 mgrid(num_kv_heads, (num_seqs + 1) / 2, num_splits), mblock(128);
#define LAUNCH_MFMA(QPG, DIET)                                               \
    hipLaunchKernelGGL((decode_attn_mfma_kernel<QPG, DIET>), mgrid, mblock,  \
                       0,                                                    \
                       stream, (float*)partial_o, (float*)partial_ml,        \
                       (__hip_bfloat16*)out, (const __hip_bfloat16*)q,       \
                       (const __hip_bfloat16*)key_cache,                     \
                       (const __hip_bfloat16*)val_cache,                     \
                       (const int*)block_tables, (const int*)context_lens,   \
                       scale, num_kv_heads, max_pages, num_splits,         \
                       num_seqs, q_stride)
    if (variant == 5) {
      if (qpg == 1) LAUNCH_MFMA(1, 1);
      else if (qpg == 2) LAUNCH_MFMA(2, 1);
      else if (qpg == 4) LAUNCH_MFMA(4, 1);
      else if (qpg == 8) LAUNCH_MFMA(8, 1);
      else return;
    } else {
      if (qpg == 1) LAUNCH_MFMA(1, 0);
      else if (qpg == 2) LAUNCH_MFMA(2, 0);
      else if (qpg == 4) LAUNCH_MFMA(4, 0);
      else if (qpg == 8) LAUNCH_MFMA(8, 0);
      else return;
    }
#undef LAUNCH_MFMA
    if (num_splits > 1) {
      dim3 cgrid(num_seqs, num_q_heads), cblock(HEAD_DIM);
      hipLaunchKernelGGL(decode_combine_kernel, cgrid, cblock, 0, stream,
                         (__hip_bfloat16*)out, (const float*)partial_o,
                         (const float*)partial_ml, num_splits, num_q_heads);
    }
    return;
  }
  // variant 4 with a non-16 page size falls back to the dot2 kernel
  const int v = variant < 0 ? 1 : (variant > 3 ? 1 : variant);
  switch (qpg * 8 + v) {
    case 8: LAUNCH_QPG(1, 0); break;
    case 9: LAUNCH_QPG(1, 1); break;
    case 10: LAUNCH_QPG(1, 2); break;
    case 11: LAUNCH_QPG(1, 3); break;
    case 16: LAUNCH_QPG(2, 0); break;
    case 17: LAUNCH_QPG(2, 1); break;
    case 18: LAUNCH_QPG(2, 2); break;
    case 19: LAUNCH_QPG(2, 3); break;
    case 32: LAUNCH_QPG(4, 0); break;
    case 33: LAUNCH_QPG(4, 1); break;
    case 34: LAUNCH_QPG(4, 2); break;
    case 35: LAUNCH_QPG(4, 3); break;
    case 64: LAUNCH_QPG(8, 0); break;
    case 65: LAUNCH_QPG(8, 1); break;
    case 66: LAUNCH_QPG(8, 2); break;
    case 67: LAUNCH_QPG(8, 3); break;
    default: return;   // validated host-side
  }
#undef LAUNCH_QPG
  if (num_splits > 1) {
    dim3 cgrid(num_seqs, num_q_heads), cblock(HEAD_DIM);
    hipLaunchKernelGGL(decode_combine_kernel, cgrid, cblock, 0, stream,
                       (__hip_bfloat16*)out, (const float*)partial_o,
                       (const float*)partial_ml, num_splits, num_q_heads);
  }
}

}  // extern "C"
