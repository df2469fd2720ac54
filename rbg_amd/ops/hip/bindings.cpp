// PyTorch bindings for the rbg_amd CDNA4 HIP kernels.
// Thin shape/dtype checks here; all math lives in the .hip translation units.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <hip/hip_runtime.h>

#include <cstring>
#include <vector>

extern "C" {
void launch_rmsnorm(void*, void*, const void*, float, int, int, hipStream_t);
void launch_fused_add_rmsnorm(void*, void*, const void*, float, int, int,
                              hipStream_t);
void launch_silu_mul(void*, const void*, int, int, hipStream_t);
void launch_rope_store_kv(void*, void*, const void*, void*, void*,
                          const void*, const void*, const void*, int, int,
                          int, int, int, int, int, hipStream_t);
void launch_decode_attention(void*, void*, void*, const void*, const void*,
                             const void*, const void*, const void*, float,
                             int, int, int, int, int, int, int, int,
                             hipStream_t);
void launch_prefill_attention(void*, const void*, const void*, const void*,
                              const void*, const void*, float, int, int, int,
                              int, int, int, hipStream_t);
void launch_mfma_probe(void*, const void*, const void*, int, int, hipStream_t);
void launch_mfma_probe32(void*, const void*, const void*, int, int, int,
                         hipStream_t);
void launch_skinny_gemm(void*, void*, const void*, const void*, int, int,
                        int, int, hipStream_t);
void launch_reduce_splits(void*, const void*, int, long, hipStream_t);
void launch_kv_peer_copy(void*, const void*, const void*, const void*, int,
                         int, int, long, long, long, long, hipStream_t);
void launch_xgmi_allreduce(void*, const void*, void**, void**, int, int,
                           long, hipStream_t);
long xgmi_allreduce_signal_bytes();
long xgmi_allreduce_error_offset();
long xgmi_allreduce_counter_offset();
}

namespace {

hipStream_t current_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

void check_bf16_contig(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

torch::Tensor rmsnorm(torch::Tensor x, torch::Tensor weight, double eps) {
  check_bf16_contig(x, "x");
  check_bf16_contig(weight, "weight");
  const int hidden = x.size(-1);
  TORCH_CHECK(hidden % 8 == 0, "hidden must be a multiple of 8");
  const int tokens = x.numel() / hidden;
  auto out = torch::empty_like(x);
  launch_rmsnorm(out.data_ptr(), x.data_ptr(), weight.data_ptr(), (float)eps,
                 tokens, hidden, current_stream());
  return out;
}

void fused_add_rmsnorm(torch::Tensor x, torch::Tensor residual,
                       torch::Tensor weight, double eps) {
  check_bf16_contig(x, "x");
  check_bf16_contig(residual, "residual");
  const int hidden = x.size(-1);
  TORCH_CHECK(hidden % 8 == 0, "hidden must be a multiple of 8");
  const int tokens = x.numel() / hidden;
  launch_fused_add_rmsnorm(x.data_ptr(), residual.data_ptr(),
                           weight.data_ptr(), (float)eps, tokens, hidden,
                           current_stream());
}

torch::Tensor silu_mul(torch::Tensor x) {
  check_bf16_contig(x, "x");
  const int inter2 = x.size(-1);
  TORCH_CHECK(inter2 % 16 == 0, "2*inter must be a multiple of 16");
  const int inter = inter2 / 2;
  const int tokens = x.numel() / inter2;
  auto out = torch::empty({x.size(0), inter}, x.options());
  launch_silu_mul(out.data_ptr(), x.data_ptr(), tokens, inter,
                  current_stream());
  return out;
}

static void check_bf16_rowslice(const torch::Tensor& t, const char* name) {
  // bf16 on GPU, [T, W] or [T, H, D], innermost dims contiguous; the
  // TOKEN stride may be wider (a column slice of the fused QKV buffer)
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  if (t.dim() == 2) {
    TORCH_CHECK(t.stride(1) == 1, name, " rows must be contiguous");
  } else {
    TORCH_CHECK(t.dim() == 3 && t.stride(2) == 1 &&
                t.stride(1) == t.size(2), name,
                " must be [T, H, D] with contiguous (H, D) rows");
  }
}

void rope_store_kv(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                   torch::Tensor key_cache, torch::Tensor value_cache,
                   torch::Tensor cos_sin, torch::Tensor positions,
                   torch::Tensor slot_mapping) {
  check_bf16_rowslice(q, "q");
  check_bf16_rowslice(k, "k");
  check_bf16_rowslice(v, "v");
  TORCH_CHECK(k.stride(0) == v.stride(0), "k/v stride mismatch");
  TORCH_CHECK(cos_sin.scalar_type() == torch::kFloat32, "cos_sin fp32");
  TORCH_CHECK(positions.scalar_type() == torch::kInt32, "positions int32");
  TORCH_CHECK(slot_mapping.scalar_type() == torch::kInt32, "slots int32");
  const int tokens = q.size(0);
  const int head_dim = key_cache.size(3);
  const int num_kv_heads = key_cache.size(1);
  const int page_size = key_cache.size(2);
  const int num_q_heads = (int)(q.numel() / tokens / head_dim);
  launch_rope_store_kv(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                       key_cache.data_ptr(), value_cache.data_ptr(),
                       cos_sin.data_ptr(), positions.data_ptr(),
                       slot_mapping.data_ptr(), tokens, num_q_heads,
                       num_kv_heads, head_dim, page_size,
                       (int)q.stride(0), (int)k.stride(0),
                       current_stream());
}

torch::Tensor decode_attention(torch::Tensor q, torch::Tensor key_cache,
                               torch::Tensor value_cache,
                               torch::Tensor block_tables,
                               torch::Tensor context_lens, double scale,
                               int64_t num_splits, int64_t wide) {
  // q may be a slice of the fused QKV projection: 3-D bf16 with
  // contiguous (head, dim) rows and an arbitrary token stride
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16 &&
              q.dim() == 3 && q.stride(2) == 1 &&
              q.stride(1) == q.size(2), "q must be [T, QH, D] row-sliced");
  const int num_seqs = q.size(0);
  const int num_q_heads = q.size(1);
  const int head_dim = q.size(2);
  TORCH_CHECK(head_dim == 128, "head_dim must be 128");
  const int num_kv_heads = key_cache.size(1);
  const int page_size = key_cache.size(2);
  const int max_pages = block_tables.size(1);
  const int qpg = num_q_heads / num_kv_heads;
  TORCH_CHECK(qpg == 1 || qpg == 2 || qpg == 4 || qpg == 8,
              "GQA group must be 1/2/4/8");
  TORCH_CHECK(block_tables.scalar_type() == torch::kInt32, "tables int32");
  TORCH_CHECK(context_lens.scalar_type() == torch::kInt32, "lens int32");
  auto out = torch::empty({num_seqs, num_q_heads, head_dim}, q.options());
  torch::Tensor partial_o, partial_ml;
  void *po = nullptr, *pml = nullptr;
  if (num_splits > 1) {
    auto opts = q.options().dtype(torch::kFloat32);
    partial_o = torch::empty({num_splits, num_seqs, num_q_heads, head_dim}, opts);
    partial_ml = torch::empty({num_splits, num_seqs, num_q_heads, 2}, opts);
    po = partial_o.data_ptr();
    pml = partial_ml.data_ptr();
  }
  launch_decode_attention(out.data_ptr(), po, pml, q.data_ptr(),
                          key_cache.data_ptr(), value_cache.data_ptr(),
                          block_tables.data_ptr(), context_lens.data_ptr(),
                          (float)scale, num_seqs, num_q_heads, num_kv_heads,
                          page_size, max_pages, (int)num_splits, (int)wide,
                          (int)q.stride(0), current_stream());
  return out;
}

torch::Tensor prefill_attention(torch::Tensor q, torch::Tensor k,
                                torch::Tensor v, torch::Tensor block_info,
                                torch::Tensor seq_lens, double scale,
                                int64_t swz) {
  auto rowsliced3 = [](const torch::Tensor& t, const char* name) {
    TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kBFloat16 &&
                t.dim() == 3 && t.stride(2) == 1 &&
                t.stride(1) == t.size(2), name,
                " must be [T, H, D] row-sliced");
  };
  rowsliced3(q, "q");
  rowsliced3(k, "k");
  rowsliced3(v, "v");
  TORCH_CHECK(k.stride(0) == v.stride(0), "k/v stride mismatch");
  TORCH_CHECK(q.size(2) == 128, "head_dim must be 128");
  TORCH_CHECK(block_info.scalar_type() == torch::kInt32, "block_info int32");
  TORCH_CHECK(seq_lens.scalar_type() == torch::kInt32, "seq_lens int32");
  const int nblocks = block_info.size(0);
  const int num_q_heads = q.size(1);
  const int num_kv_heads = k.size(1);
  auto out = torch::empty({q.size(0), (int64_t)num_q_heads, (int64_t)128},
                          q.options());
  launch_prefill_attention(out.data_ptr(), q.data_ptr(), k.data_ptr(),
                           v.data_ptr(), block_info.data_ptr(),
                           seq_lens.data_ptr(), (float)scale, nblocks,
                           num_q_heads, num_kv_heads, (int)swz,
                           (int)q.stride(0), (int)k.stride(0),
                           current_stream());
  return out;
}

torch::Tensor mfma_probe(torch::Tensor a, torch::Tensor b, int64_t a_split,
                         int64_t b_split) {
  auto d = torch::zeros({16, 16}, a.options().dtype(torch::kFloat32));
  launch_mfma_probe(d.data_ptr(), a.data_ptr(), b.data_ptr(), (int)a_split,
                    (int)b_split, current_stream());
  return d;
}

torch::Tensor mfma_probe32(torch::Tensor a, torch::Tensor b, int64_t a_split,
                           int64_t b_split, int64_t dmap) {
  auto d = torch::zeros({32, 32}, a.options().dtype(torch::kFloat32));
  launch_mfma_probe32(d.data_ptr(), a.data_ptr(), b.data_ptr(), (int)a_split,
                      (int)b_split, (int)dmap, current_stream());
  return d;
}

}  // namespace

static int pick_gemm_splits(int N, int K) {
  // fill the chip with 16-wave workgroups (>= ~256 WGs); each split's K
  // range divides into 8 x 32-k wave slices (K % (splits*256) == 0)
  const int n_groups = N >> 5;
  for (int s : {8, 4, 2})
    if (n_groups * s <= 512 && n_groups * s >= 192 && K % (s * 256) == 0)
      return s;
  for (int s : {4, 2})
    if (n_groups * s >= 128 && K % (s * 256) == 0) return s;
  return 1;
}

bool skinny_gemm_supported(int64_t M, int64_t N, int64_t K) {
  // M <= 256: the cross-slice LDS reduce holds 4 waves x M_TILES x 1 KB
  // (64 KB at M_TILES=16); larger batches fall back to hipBLASLt
  return M >= 1 && M <= 256 && (N % 32) == 0 && (K % 256) == 0;
}

torch::Tensor skinny_gemm(torch::Tensor a, torch::Tensor w,
                          int64_t force_splits = 0) {
  check_bf16_contig(a, "a");
  check_bf16_contig(w, "w");
  const int64_t M = a.size(0), K = a.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K, "K mismatch");
  TORCH_CHECK(skinny_gemm_supported(M, N, K), "unsupported skinny shape");
  auto c = torch::empty({M, N}, a.options());
  int splits = force_splits > 0 ? (int)force_splits
                                : pick_gemm_splits((int)N, (int)K);
  if (K % (splits * 256) != 0) splits = pick_gemm_splits((int)N, (int)K);
  torch::Tensor ws;
  void* wsp = nullptr;
  if (splits > 1) {
    ws = torch::empty({splits, M, N}, a.options().dtype(torch::kFloat32));
    wsp = ws.data_ptr();
  }
  launch_skinny_gemm(c.data_ptr(), wsp, a.data_ptr(), w.data_ptr(), (int)M,
                     (int)N, (int)K, splits, current_stream());
  if (splits > 1)
    launch_reduce_splits(c.data_ptr(), wsp, splits, (long)(M * N),
                         current_stream());
  return c;
}

// ---- KV-pool IPC export + xGMI peer push (parallel/kv_peer.py) -----------
// The decode engine's KV pool is allocated with hipMalloc directly (not the
// caching allocator) so data_ptr() IS the allocation base hipIpcGetMemHandle
// requires; the prefill process maps it with lazy peer access (the xGMI
// route) and pushes pages with the kv_peer_copy kernel.

torch::Tensor ipc_alloc_bf16(std::vector<int64_t> sizes) {
  int64_t numel = 1;
  for (auto s : sizes) numel *= s;
  void* ptr = nullptr;
  hipError_t err = hipMalloc(&ptr, numel * 2);
  TORCH_CHECK(err == hipSuccess, "hipMalloc(", numel * 2,
              " B) failed: ", hipGetErrorString(err));
  int dev = 0;
  (void)hipGetDevice(&dev);
  auto opts = torch::TensorOptions()
                  .dtype(torch::kBFloat16)
                  .device(torch::kCUDA, dev);
  return torch::from_blob(
      ptr, sizes, [](void* p) { (void)hipFree(p); }, opts);
}

py::bytes kv_ipc_export(torch::Tensor pool) {
  TORCH_CHECK(pool.is_cuda(), "pool must be on GPU");
  hipIpcMemHandle_t handle;
  hipError_t err = hipIpcGetMemHandle(&handle, pool.data_ptr());
  TORCH_CHECK(err == hipSuccess,
              "hipIpcGetMemHandle failed (pool must be an ipc_alloc_bf16 "
              "base allocation): ", hipGetErrorString(err));
  return py::bytes(reinterpret_cast<const char*>(&handle), sizeof(handle));
}

int64_t kv_ipc_open(std::string raw) {
  // std::string parameter: converted from py::bytes during argument
  // processing (GIL held) so the gil_scoped_release guard is safe
  TORCH_CHECK(raw.size() == sizeof(hipIpcMemHandle_t), "bad handle size");
  hipIpcMemHandle_t handle;
  memcpy(&handle, raw.data(), sizeof(handle));
  void* ptr = nullptr;
  hipError_t err =
      hipIpcOpenMemHandle(&ptr, handle, hipIpcMemLazyEnablePeerAccess);
  TORCH_CHECK(err == hipSuccess,
              "hipIpcOpenMemHandle failed: ", hipGetErrorString(err));
  return reinterpret_cast<int64_t>(ptr);
}

void kv_ipc_close(int64_t ptr) {
  hipError_t err = hipIpcCloseMemHandle(reinterpret_cast<void*>(ptr));
  TORCH_CHECK(err == hipSuccess,
              "hipIpcCloseMemHandle failed: ", hipGetErrorString(err));
}

void kv_peer_copy(int64_t dst_base, torch::Tensor src_kv,
                  torch::Tensor src_pages, torch::Tensor dst_pages,
                  int64_t dst_num_pages) {
  // src_kv: [L, 2, Ps, kvh, ps, hd] bf16 contiguous; dst pool has the SAME
  // per-page layout but its own page count (dst_num_pages)
  check_bf16_contig(src_kv, "src_kv");
  TORCH_CHECK(src_kv.dim() == 6 && src_kv.size(1) == 2, "kv must be 6-D");
  TORCH_CHECK(src_pages.is_cuda() && dst_pages.is_cuda() &&
                  src_pages.scalar_type() == torch::kInt32 &&
                  dst_pages.scalar_type() == torch::kInt32,
              "page lists must be int32 on GPU");
  TORCH_CHECK(src_pages.numel() == dst_pages.numel(), "page count mismatch");
  const int n_pages = src_pages.numel();
  if (n_pages == 0) return;
  const int layers = src_kv.size(0);
  const long Ps = src_kv.size(2);
  const long chunk_elems = src_kv.size(3) * src_kv.size(4) * src_kv.size(5);
  TORCH_CHECK(chunk_elems % 8 == 0, "chunk must be 16B-aligned");
  const long chunk_vec = chunk_elems / 8;  // uint4 = 8 bf16
  launch_kv_peer_copy(reinterpret_cast<void*>(dst_base), src_kv.data_ptr(),
                      src_pages.data_ptr(), dst_pages.data_ptr(), n_pages,
                      2 * layers, (int)chunk_vec, chunk_vec,
                      chunk_vec, Ps * chunk_vec, dst_num_pages * chunk_vec,
                      current_stream());
}

// ---- one-shot xGMI all-reduce (parallel/xgmi_allreduce.py) ---------------
// Data buffer: ordinary hipMalloc (IPC-exportable base).  Signal buffer:
// UNCACHED device memory so remote flag stores are immediately visible to
// the local spin loops.

int64_t ar_alloc_signals() {
  void* ptr = nullptr;
  hipError_t err = hipExtMallocWithFlags(
      &ptr, (size_t)xgmi_allreduce_signal_bytes(), hipDeviceMallocUncached);
  TORCH_CHECK(err == hipSuccess,
              "uncached signal alloc failed: ", hipGetErrorString(err));
  (void)hipMemset(ptr, 0, (size_t)xgmi_allreduce_signal_bytes());
  (void)hipDeviceSynchronize();
  return reinterpret_cast<int64_t>(ptr);
}

py::bytes ar_export_ptr(int64_t ptr) {
  hipIpcMemHandle_t handle;
  hipError_t err =
      hipIpcGetMemHandle(&handle, reinterpret_cast<void*>(ptr));
  TORCH_CHECK(err == hipSuccess,
              "hipIpcGetMemHandle failed: ", hipGetErrorString(err));
  return py::bytes(reinterpret_cast<const char*>(&handle), sizeof(handle));
}

unsigned ar_error_flag(int64_t sig_ptr) {
  unsigned err_word = 0;
  // offsetof, NOT sizeof-4: the 128-byte struct alignment pads the tail,
  // so sizeof-4 read the padding and timeouts were invisible (r2 GPU run)
  char* base = reinterpret_cast<char*>(sig_ptr);
  (void)hipMemcpy(&err_word, base + xgmi_allreduce_error_offset(),
                  sizeof(unsigned), hipMemcpyDeviceToHost);
  return err_word;
}

torch::Tensor ar_dump_signals(int64_t sig_ptr) {
  // debug: the whole Signals block as int32 on CPU
  long n = xgmi_allreduce_signal_bytes() / 4;
  auto t = torch::empty({n}, torch::TensorOptions().dtype(torch::kInt32));
  (void)hipMemcpy(t.data_ptr(), reinterpret_cast<void*>(sig_ptr), n * 4,
                  hipMemcpyDeviceToHost);
  return t;
}

torch::Tensor xgmi_allreduce(torch::Tensor inp,
                             std::vector<int64_t> sig_ptrs,
                             std::vector<int64_t> data_ptrs, int64_t rank) {
  check_bf16_contig(inp, "inp");
  const int world = (int)sig_ptrs.size();
  TORCH_CHECK(world >= 1 && world <= 8 &&
                  data_ptrs.size() == (size_t)world && rank >= 0 &&
                  rank < world,
              "bad world/rank");
  TORCH_CHECK(inp.numel() % 8 == 0, "numel must be a multiple of 8");
  auto out = torch::empty_like(inp);
  void* sp[8];
  void* dp[8];
  for (int i = 0; i < world; ++i) {
    sp[i] = reinterpret_cast<void*>(sig_ptrs[i]);
    dp[i] = reinterpret_cast<void*>(data_ptrs[i]);
  }
  launch_xgmi_allreduce(out.data_ptr(), inp.data_ptr(), sp, dp, world,
                        (int)rank, (long)inp.numel(), current_stream());
  return out;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm", &rmsnorm, "RMSNorm (bf16)");
  m.def("fused_add_rmsnorm", &fused_add_rmsnorm, "x,residual += ; rmsnorm");
  m.def("silu_mul", &silu_mul, "silu(gate)*up");
  m.def("rope_store_kv", &rope_store_kv, "RoPE + paged KV write");
  m.def("decode_attention", &decode_attention, "paged flash-decode");
  m.def("prefill_attention", &prefill_attention, "varlen causal flash prefill");
  m.def("mfma_probe", &mfma_probe, "MFMA layout probe");
  m.def("mfma_probe32", &mfma_probe32, "32x32x16 MFMA layout probe");
  m.def("skinny_gemm", &skinny_gemm, "decode-shape GEMM (M<=128)",
        py::arg("a"), py::arg("w"), py::arg("force_splits") = 0);
  // gil_scoped_release on the IPC/copy entry points: mapping or launching
  // against a dying peer can block inside the HIP runtime, and holding the
  // GIL there freezes heartbeats until the controller kills the process
  m.def("ipc_alloc_bf16", &ipc_alloc_bf16,
        "hipMalloc-backed bf16 tensor (IPC-exportable base allocation)",
        py::call_guard<py::gil_scoped_release>());
  m.def("kv_ipc_export", &kv_ipc_export, "hipIpcGetMemHandle of a KV pool");
  m.def("kv_ipc_open", &kv_ipc_open,
        "map a peer KV pool (lazy xGMI peer access)",
        py::call_guard<py::gil_scoped_release>());
  m.def("kv_ipc_close", &kv_ipc_close, "unmap a peer KV pool",
        py::call_guard<py::gil_scoped_release>());
  m.def("kv_peer_copy", &kv_peer_copy,
        "gather local pages -> scatter into peer pool over xGMI",
        py::call_guard<py::gil_scoped_release>());
  m.def("ar_alloc_signals", &ar_alloc_signals,
        "uncached signal buffer for the xGMI all-reduce");
  m.def("ar_export_ptr", &ar_export_ptr, "hipIpc handle of a raw pointer");
  m.def("ar_error_flag", &ar_error_flag, "read the all-reduce error word");
  m.def("ar_dump_signals", &ar_dump_signals, "debug: dump the signal block");
  m.def("ar_counter_offset", &xgmi_allreduce_counter_offset);
  m.def("xgmi_allreduce", &xgmi_allreduce,
        "one-shot all-reduce over peer-mapped HBM (graph-capturable)");
}
