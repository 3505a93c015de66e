// ops.hip — torch extension binding for the gfx950 kernels (`_hip_ops`).
// Pure HIP (no hipify, no CUDA shims): includes the ROCm-native ATen HIP
// headers directly and drives hipStream_t launchers from the .hip kernels.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

#include <hip/hip_runtime.h>

#define CHECK_HIP(expr)                                                    \
  do {                                                                     \
    hipError_t err__ = (expr);                                             \
    TORCH_CHECK(err__ == hipSuccess, "HIP error: ", hipGetErrorString(err__)); \
  } while (0)

#define CHECK_GPU(t) TORCH_CHECK((t).is_cuda() && (t).is_contiguous(), #t " must be contiguous on GPU")

extern "C" {
hipError_t lds_hash_prompts(const int32_t*, const int64_t*, int, int, int,
                            uint64_t, uint64_t*, int32_t*, hipStream_t);
hipError_t lds_table_update(uint64_t*, unsigned long long*, uint32_t,
                            const uint64_t*, int64_t, int, int, hipStream_t);
hipError_t lds_match_longest(const uint64_t*, const unsigned long long*,
                             uint32_t, const uint64_t*, const int32_t*, int,
                             int, int, int32_t*, hipStream_t);
hipError_t lds_rmsnorm(const void*, void*, const void*, void*, int64_t, int,
                       float, hipStream_t);
hipError_t lds_rope(void*, void*, const float*, const int32_t*, int, int, int,
                    int, hipStream_t);
hipError_t lds_silu_mul(const void*, void*, int64_t, int, hipStream_t);
hipError_t lds_reshape_and_cache(const void*, const void*, void*, void*,
                                 const int64_t*, int, int, int, int, int,
                                 hipStream_t);
hipError_t lds_gather_blocks(const void*, void*, const int32_t*, int, int,
                             int64_t, int64_t, int, hipStream_t);
hipError_t lds_paged_attention(const void*, const void*, const void*,
                               const int32_t*, const int32_t*, void*, int, int,
                               int, int, int, int, int, int, float,
                               hipStream_t);
hipError_t lds_paged_attention_split(const void*, const void*, const void*,
                                     const int32_t*, const int32_t*, void*,
                                     float*, float*, int, int, int, int, int,
                                     int, int, int, int, int, float,
                                     hipStream_t);
hipError_t lds_flash_prefill(const void*, const void*, const void*,
                             const int32_t*, const int32_t*, const int32_t*,
                             void*, int, int, int, int, int, int, int, float,
                             hipStream_t);
hipError_t lds_flash_prefill_glds(const void*, const void*, const void*,
                                  const int32_t*, const int32_t*,
                                  const int32_t*, void*, int, int, int, int,
                                  int, int, float, hipStream_t);
}

namespace {

int kv_fp8_flag(const torch::Tensor& cache) {
  if (cache.scalar_type() == at::kFloat8_e4m3fn) return 1;
  TORCH_CHECK(cache.scalar_type() == at::kBFloat16,
              "KV cache must be bf16 or fp8_e4m3fn");
  return 0;
}

hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

// ---- prefix-cache kernels ----

std::vector<torch::Tensor> hash_prompts(torch::Tensor tokens,
                                        torch::Tensor offsets,
                                        int64_t block_tokens,
                                        int64_t max_blocks, int64_t seed0) {
  CHECK_GPU(tokens);
  CHECK_GPU(offsets);
  TORCH_CHECK(tokens.dtype() == torch::kInt32 && offsets.dtype() == torch::kInt64);
  int n_req = (int)offsets.size(0) - 1;
  auto hashes = torch::zeros({n_req, max_blocks},
                             tokens.options().dtype(torch::kUInt64));
  auto counts = torch::zeros({n_req}, tokens.options().dtype(torch::kInt32));
  CHECK_HIP(lds_hash_prompts(
      tokens.data_ptr<int32_t>(), offsets.data_ptr<int64_t>(), n_req,
      (int)block_tokens, (int)max_blocks, (uint64_t)seed0,
      (uint64_t*)hashes.data_ptr(), counts.data_ptr<int32_t>(), cur_stream()));
  return {hashes, counts};
}

void table_update(torch::Tensor keys, torch::Tensor masks,
                  torch::Tensor hashes, int64_t endpoint, bool is_remove) {
  CHECK_GPU(keys); CHECK_GPU(masks); CHECK_GPU(hashes);
  TORCH_CHECK((keys.size(0) & (keys.size(0) - 1)) == 0, "table cap must be 2^k");
  CHECK_HIP(lds_table_update(
      (uint64_t*)keys.data_ptr(), (unsigned long long*)masks.data_ptr(),
      (uint32_t)(keys.size(0) - 1), (const uint64_t*)hashes.data_ptr(),
      hashes.numel(), (int)endpoint, is_remove ? 1 : 0, cur_stream()));
}

torch::Tensor match_longest(torch::Tensor keys, torch::Tensor masks,
                            torch::Tensor hashes, torch::Tensor counts,
                            int64_t n_endpoints) {
  CHECK_GPU(keys); CHECK_GPU(masks); CHECK_GPU(hashes); CHECK_GPU(counts);
  int n_req = (int)hashes.size(0);
  auto out = torch::zeros({n_req, n_endpoints},
                          hashes.options().dtype(torch::kInt32));
  CHECK_HIP(lds_match_longest(
      (const uint64_t*)keys.data_ptr(), (const unsigned long long*)masks.data_ptr(),
      (uint32_t)(keys.size(0) - 1), (const uint64_t*)hashes.data_ptr(),
      counts.data_ptr<int32_t>(), n_req, (int)hashes.size(1), (int)n_endpoints,
      out.data_ptr<int32_t>(), cur_stream()));
  return out;
}

// ---- engine kernels ----

torch::Tensor rmsnorm(torch::Tensor x, torch::Tensor w, double eps,
                      c10::optional<torch::Tensor> residual) {
  CHECK_GPU(x); CHECK_GPU(w);
  TORCH_CHECK(x.dtype() == torch::kBFloat16, "rmsnorm expects bf16");
  auto y = torch::empty_like(x);
  int H = (int)x.size(-1);
  int64_t rows = x.numel() / H;
  void* res_ptr = nullptr;
  if (residual.has_value()) {
    CHECK_GPU(*residual);
    res_ptr = residual->data_ptr();
  }
  CHECK_HIP(lds_rmsnorm(x.data_ptr(), res_ptr, w.data_ptr(), y.data_ptr(),
                        rows, H, (float)eps, cur_stream()));
  return y;
}

void rope(torch::Tensor q, torch::Tensor k, torch::Tensor cos_sin,
          torch::Tensor positions) {
  CHECK_GPU(q); CHECK_GPU(k); CHECK_GPU(cos_sin); CHECK_GPU(positions);
  int T = (int)q.size(0);
  int qh = (int)q.size(1), kvh = (int)k.size(1), d = (int)q.size(2);
  CHECK_HIP(lds_rope(q.data_ptr(), k.data_ptr(), cos_sin.data_ptr<float>(),
                     positions.data_ptr<int32_t>(), T, qh, kvh, d,
                     cur_stream()));
}

torch::Tensor silu_mul(torch::Tensor gate_up) {
  CHECK_GPU(gate_up);
  int64_t I2 = gate_up.size(-1);
  int64_t T = gate_up.numel() / I2;
  auto sizes = gate_up.sizes().vec();
  sizes.back() = I2 / 2;
  auto y = torch::empty(sizes, gate_up.options());
  CHECK_HIP(lds_silu_mul(gate_up.data_ptr(), y.data_ptr(), T, (int)(I2 / 2),
                         cur_stream()));
  return y;
}

void reshape_and_cache(torch::Tensor k_new, torch::Tensor v_new,
                       torch::Tensor k_cache, torch::Tensor v_cache,
                       torch::Tensor slots) {
  CHECK_GPU(k_new); CHECK_GPU(v_new); CHECK_GPU(k_cache); CHECK_GPU(v_cache);
  CHECK_GPU(slots);
  int T = (int)k_new.size(0), kvh = (int)k_new.size(1), d = (int)k_new.size(2);
  int bs = (int)k_cache.size(2);
  CHECK_HIP(lds_reshape_and_cache(k_new.data_ptr(), v_new.data_ptr(),
                                  k_cache.data_ptr(), v_cache.data_ptr(),
                                  slots.data_ptr<int64_t>(), T, kvh, bs, d,
                                  kv_fp8_flag(k_cache), cur_stream()));
}

void move_blocks(torch::Tensor pool, torch::Tensor staging,
                 torch::Tensor block_ids, bool is_scatter) {
  CHECK_GPU(pool); CHECK_GPU(staging); CHECK_GPU(block_ids);
  // pool: [L, 2, NB, KVH, BS, D]; staging: [n, L, 2, KVH, BS, D]
  int L = (int)pool.size(0);
  int64_t nb = pool.size(2);
  int64_t block_bytes = pool.size(3) * pool.size(4) * pool.size(5) *
                        pool.element_size();
  CHECK_HIP(lds_gather_blocks(pool.data_ptr(), staging.data_ptr(),
                              block_ids.data_ptr<int32_t>(),
                              (int)block_ids.size(0), L, nb, block_bytes,
                              is_scatter ? 1 : 0, cur_stream()));
}

// ---- HIP IPC surface for the direct peer-pull transfer path -------------
// The KV pool is allocated with its own hipMalloc (NOT the caching
// allocator, whose suballocation offsets would corrupt IPC handles); the
// decode rank opens the prefill rank's handle once and pulls blocks with
// copy_blocks_peer over the mapped pointer (xGMI one-sided read).

torch::Tensor ipc_alloc_tensor(std::vector<int64_t> shape,
                               torch::Tensor like) {
  int64_t numel = 1;
  for (auto s : shape) numel *= s;
  int64_t nbytes = numel * like.element_size();
  void* ptr = nullptr;
  CHECK_HIP(hipMalloc(&ptr, nbytes));
  CHECK_HIP(hipMemset(ptr, 0, nbytes));
  auto deleter = [](void* p) { (void)hipFree(p); };
  return torch::from_blob(ptr, shape, deleter,
                          like.options());
}

py::bytes ipc_handle(torch::Tensor t) {
  hipIpcMemHandle_t h;
  CHECK_HIP(hipIpcGetMemHandle(&h, t.data_ptr()));
  return py::bytes(reinterpret_cast<const char*>(&h), sizeof(h));
}

uintptr_t ipc_open(py::bytes handle) {
  std::string s = handle;
  TORCH_CHECK(s.size() == sizeof(hipIpcMemHandle_t), "bad IPC handle size");
  hipIpcMemHandle_t h;
  memcpy(&h, s.data(), sizeof(h));
  void* ptr = nullptr;
  hipError_t err =
      hipIpcOpenMemHandle(&ptr, h, hipIpcMemLazyEnablePeerAccess);
  TORCH_CHECK(err == hipSuccess, "hipIpcOpenMemHandle: ",
              hipGetErrorString(err));
  return reinterpret_cast<uintptr_t>(ptr);
}

void ipc_close(uintptr_t ptr) {
  CHECK_HIP(hipIpcCloseMemHandle(reinterpret_cast<void*>(ptr)));
}

extern "C" hipError_t lds_copy_blocks_peer(const void*, void*, const int32_t*,
                                           const int32_t*, int, int, int64_t,
                                           int64_t, int64_t, hipStream_t);

void copy_blocks_peer(uintptr_t src_pool_ptr, torch::Tensor dst_pool,
                      torch::Tensor src_ids, torch::Tensor dst_ids,
                      int64_t src_nb) {
  CHECK_GPU(dst_pool); CHECK_GPU(src_ids); CHECK_GPU(dst_ids);
  int L = (int)dst_pool.size(0);
  int64_t dst_nb = dst_pool.size(2);
  int64_t block_bytes = dst_pool.size(3) * dst_pool.size(4) *
                        dst_pool.size(5) * dst_pool.element_size();
  CHECK_HIP(lds_copy_blocks_peer(
      reinterpret_cast<const void*>(src_pool_ptr), dst_pool.data_ptr(),
      src_ids.data_ptr<int32_t>(), dst_ids.data_ptr<int32_t>(),
      (int)src_ids.size(0), L, src_nb, dst_nb, block_bytes, cur_stream()));
}

torch::Tensor paged_attention(torch::Tensor q, torch::Tensor k_cache,
                              torch::Tensor v_cache,
                              torch::Tensor block_tables,
                              torch::Tensor seq_lens, double scale,
                              int64_t chunk) {
  CHECK_GPU(q); CHECK_GPU(k_cache); CHECK_GPU(v_cache);
  CHECK_GPU(block_tables); CHECK_GPU(seq_lens);
  int B = (int)q.size(0), qh = (int)q.size(1), d = (int)q.size(2);
  int kvh = (int)k_cache.size(1), bs = (int)k_cache.size(2);
  auto out = torch::empty_like(q);
  CHECK_HIP(lds_paged_attention(
      q.data_ptr(), k_cache.data_ptr(), v_cache.data_ptr(),
      block_tables.data_ptr<int32_t>(), seq_lens.data_ptr<int32_t>(),
      out.data_ptr(), B, qh, kvh, bs, d, (int)block_tables.size(1),
      kv_fp8_flag(k_cache), (int)chunk, (float)scale, cur_stream()));
  return out;
}

torch::Tensor paged_attention_split(torch::Tensor q, torch::Tensor k_cache,
                                    torch::Tensor v_cache,
                                    torch::Tensor block_tables,
                                    torch::Tensor seq_lens, int64_t n_parts,
                                    int64_t part_tokens, double scale,
                                    int64_t chunk) {
  CHECK_GPU(q); CHECK_GPU(k_cache); CHECK_GPU(v_cache);
  CHECK_GPU(block_tables); CHECK_GPU(seq_lens);
  int B = (int)q.size(0), qh = (int)q.size(1), d = (int)q.size(2);
  int kvh = (int)k_cache.size(1), bs = (int)k_cache.size(2);
  int qpg = qh / kvh;
  auto out = torch::empty_like(q);
  auto f32 = q.options().dtype(torch::kFloat32);
  auto part_o = torch::empty({B, kvh, n_parts, qpg, d}, f32);
  auto part_ml = torch::empty({B, kvh, n_parts, qpg, 2}, f32);
  CHECK_HIP(lds_paged_attention_split(
      q.data_ptr(), k_cache.data_ptr(), v_cache.data_ptr(),
      block_tables.data_ptr<int32_t>(), seq_lens.data_ptr<int32_t>(),
      out.data_ptr(), part_o.data_ptr<float>(), part_ml.data_ptr<float>(), B,
      qh, kvh, bs, d, (int)block_tables.size(1), (int)n_parts,
      (int)part_tokens, kv_fp8_flag(k_cache), (int)chunk, (float)scale, cur_stream()));
  return out;
}

torch::Tensor flash_prefill(torch::Tensor q, torch::Tensor k_cache,
                            torch::Tensor v_cache, torch::Tensor block_tables,
                            torch::Tensor seq_meta, torch::Tensor tiles,
                            double scale) {
  CHECK_GPU(q); CHECK_GPU(k_cache); CHECK_GPU(v_cache);
  CHECK_GPU(block_tables); CHECK_GPU(seq_meta); CHECK_GPU(tiles);
  TORCH_CHECK(q.dtype() == torch::kBFloat16, "flash_prefill expects bf16");
  int qh = (int)q.size(1), d = (int)q.size(2);
  int kvh = (int)k_cache.size(1), bs = (int)k_cache.size(2);
  auto out = torch::empty_like(q);
  // bf16 KV rides the glds staging pipeline; fp8 needs convert-on-stage.
  // LLMD_NO_GLDS=1 forces the plain-VGPR kernel (bisect knob for the
  // cross-stream wedge investigation, profiles/r02_notes.md).
  static const bool no_glds = []() {
    const char* e = getenv("LLMD_NO_GLDS");
    return e && e[0] == '1';
  }();
  if (!no_glds && !kv_fp8_flag(k_cache) && block_tables.size(1) <= 1024) {
    CHECK_HIP(lds_flash_prefill_glds(
        q.data_ptr(), k_cache.data_ptr(), v_cache.data_ptr(),
        block_tables.data_ptr<int32_t>(), seq_meta.data_ptr<int32_t>(),
        tiles.data_ptr<int32_t>(), out.data_ptr(), (int)tiles.size(0), qh,
        kvh, bs, d, (int)block_tables.size(1), (float)scale, cur_stream()));
    return out;
  }
  CHECK_HIP(lds_flash_prefill(
      q.data_ptr(), k_cache.data_ptr(), v_cache.data_ptr(),
      block_tables.data_ptr<int32_t>(), seq_meta.data_ptr<int32_t>(),
      tiles.data_ptr<int32_t>(), out.data_ptr(), (int)tiles.size(0), qh, kvh,
      bs, d, (int)block_tables.size(1), kv_fp8_flag(k_cache), (float)scale,
      cur_stream()));
  return out;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "MI355X gfx950 kernels: prefix-cache hashing/match, paged KV, decode attention";
  m.def("hash_prompts", &hash_prompts, "batched chained block hashing");
  m.def("table_update", &table_update, "device prefix table insert/remove");
  m.def("match_longest", &match_longest, "device longest-prefix match");
  m.def("rmsnorm", &rmsnorm, py::arg("x"), py::arg("w"), py::arg("eps"),
        py::arg("residual") = py::none());
  m.def("rope", &rope, "in-place NeoX RoPE");
  m.def("silu_mul", &silu_mul, "fused SiLU*up on packed gate_up");
  m.def("reshape_and_cache", &reshape_and_cache, "scatter K/V into paged pool");
  m.def("move_blocks", &move_blocks, "gather/scatter KV blocks for xGMI transfer");
  m.def("ipc_alloc_tensor", &ipc_alloc_tensor,
        "hipMalloc-backed tensor (IPC-shareable, outside the caching allocator)");
  m.def("ipc_handle", &ipc_handle, "hipIpcGetMemHandle of a tensor");
  m.def("ipc_open", &ipc_open, "map a peer pool (hipIpcOpenMemHandle)");
  m.def("ipc_close", &ipc_close, "unmap a peer pool");
  m.def("copy_blocks_peer", &copy_blocks_peer,
        "one-sided xGMI pull of KV blocks from a mapped peer pool");
  m.def("paged_attention", &paged_attention,
        "GQA decode attention over paged KV", py::arg("q"),
        py::arg("k_cache"), py::arg("v_cache"), py::arg("block_tables"),
        py::arg("seq_lens"), py::arg("scale"), py::arg("chunk") = 256);
  m.def("paged_attention_split", &paged_attention_split,
        "flash-decoding GQA attention with sequence partitioning",
        py::arg("q"), py::arg("k_cache"), py::arg("v_cache"),
        py::arg("block_tables"), py::arg("seq_lens"), py::arg("n_parts"),
        py::arg("part_tokens"), py::arg("scale"), py::arg("chunk") = 256);
  m.def("flash_prefill", &flash_prefill,
        "fused MFMA causal varlen prefill attention over paged KV");
}
