// Python bindings for the MI355X-native runtime: the C++ block manager and
// the CDNA4 kernel launchers, tensor-checked at this layer.
#include <pybind11/pybind11.h>
#include <torch/extension.h>

#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

namespace py = pybind11;

void register_block_manager(py::module_& m);

extern "C" {
void launch_rmsnorm(void*, const void*, const void*, float, int, int, hipStream_t);
void launch_fused_add_rmsnorm(void*, void*, const void*, const void*, float, int, int,
                              hipStream_t);
void launch_swiglu(void*, const void*, int64_t, int, hipStream_t);
void launch_rope_cache(void*, void*, const void*, void*, void*, const int64_t*,
                       const int64_t*, const float*, int, int, int, int, int64_t,
                       int64_t, int64_t, hipStream_t);
void launch_decode_attn(void*, const void*, const void*, const void*, const int*,
                        const int*, float, int, int, int, int, int, int, int64_t,
                        hipStream_t);
void launch_prefill_attn(void*, const void*, const void*, const void*, const int*,
                         const int*, const int*, const int*, const int*, const int*,
                         float, int, int, int, int, int, int, int64_t, hipStream_t);
void launch_sample_fullvocab(int64_t*, const void*, bool, const float*, const int64_t*,
                             const float*, const float*, const uint8_t*, const int32_t*,
                             int, int, hipStream_t);
void launch_sample(int64_t*, const float*, const float*, const int64_t*, const float*,
                   const float*, const uint8_t*, int, int, hipStream_t);
void launch_prefill_attn_mfma(void*, const void*, const void*, const void*, const int*,
                              const int*, const int*, const int*, const int*, const int*,
                              float, int, int, int, int, int, int, int64_t, hipStream_t);
void launch_mfma_probe(float*, const void*, const void*, hipStream_t);
void launch_gemm_bf16(void*, const void*, const void*, int, int, int, hipStream_t);
}

#define CHECK_CUDA(x) TORCH_CHECK(x.is_cuda(), #x " must be a GPU tensor")
#define CHECK_CONT(x) TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")
#define CHECK_BF16(x) TORCH_CHECK(x.scalar_type() == at::kBFloat16, #x " must be bf16")


// q tensors may be strided views into the fused QKV GEMM output: dim-0
// stride is free, heads/dims must be dense
static int64_t q_token_stride(const torch::Tensor& t) {
  TORCH_CHECK(t.stride(2) == 1 && t.stride(1) == t.size(2),
              "head/dim axes must be dense");
  return t.stride(0);
}

static hipStream_t cur_stream() {
  return (hipStream_t)c10::hip::getCurrentHIPStream().stream();
}

static void rmsnorm(torch::Tensor out, torch::Tensor x, torch::Tensor weight,
                    double eps) {
  CHECK_CUDA(x); CHECK_CONT(x); CHECK_BF16(x); CHECK_CONT(out); CHECK_CONT(weight);
  const int h = x.size(-1);
  const int n = x.numel() / h;
  TORCH_CHECK(h % 8 == 0, "hidden size must be a multiple of 8");
  launch_rmsnorm(out.data_ptr(), x.data_ptr(), weight.data_ptr(), (float)eps, n, h,
                 cur_stream());
}

static void fused_add_rmsnorm(torch::Tensor out, torch::Tensor residual,
                              torch::Tensor x, torch::Tensor weight, double eps) {
  CHECK_CUDA(x); CHECK_CONT(x); CHECK_BF16(x); CHECK_CONT(out);
  CHECK_CONT(residual); CHECK_CONT(weight);
  const int h = x.size(-1);
  const int n = x.numel() / h;
  launch_fused_add_rmsnorm(out.data_ptr(), residual.data_ptr(), x.data_ptr(),
                           weight.data_ptr(), (float)eps, n, h, cur_stream());
}

static void swiglu(torch::Tensor out, torch::Tensor gate_up) {
  CHECK_CUDA(gate_up); CHECK_CONT(gate_up); CHECK_BF16(gate_up); CHECK_CONT(out);
  const int inter = gate_up.size(-1) / 2;
  const int64_t n = gate_up.numel() / (2 * inter);
  TORCH_CHECK(inter % 8 == 0, "intermediate size must be a multiple of 8");
  launch_swiglu(out.data_ptr(), gate_up.data_ptr(), n, inter, cur_stream());
}

static void rope_cache(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                       torch::Tensor positions, torch::Tensor slot_mapping,
                       torch::Tensor k_cache, torch::Tensor v_cache,
                       torch::Tensor cos_sin) {
  CHECK_CUDA(q);
  CHECK_CONT(k_cache); CHECK_CONT(v_cache); CHECK_CONT(cos_sin);
  const int64_t q_stride = q_token_stride(q);
  const int64_t kv_stride = q_token_stride(k);
  TORCH_CHECK(v.stride(0) == kv_stride, "k/v must share layout");
  TORCH_CHECK(positions.scalar_type() == at::kLong && slot_mapping.scalar_type() == at::kLong);
  TORCH_CHECK(cos_sin.scalar_type() == at::kFloat);
  const int N = q.size(0);
  const int Hq = q.size(1);
  const int Hkv = k.size(1);
  const int D = q.size(2);
  const int64_t P = cos_sin.size(1);
  TORCH_CHECK(D % 128 == 0 || D <= 512, "head_dim constraint");
  launch_rope_cache(q.data_ptr(), k.data_ptr(), v.data_ptr(), k_cache.data_ptr(),
                    v_cache.data_ptr(), positions.data_ptr<int64_t>(),
                    slot_mapping.data_ptr<int64_t>(), cos_sin.data_ptr<float>(), N,
                    Hq, Hkv, D, P, q_stride, kv_stride, cur_stream());
}

static void decode_attn(torch::Tensor out, torch::Tensor q, torch::Tensor k_cache,
                        torch::Tensor v_cache, torch::Tensor block_tables,
                        torch::Tensor seq_lens, double scale) {
  CHECK_CUDA(q); CHECK_BF16(q); CHECK_CONT(out);
  CHECK_CONT(k_cache); CHECK_CONT(v_cache);
  const int64_t q_stride = q_token_stride(q);
  TORCH_CHECK(block_tables.scalar_type() == at::kInt && seq_lens.scalar_type() == at::kInt);
  CHECK_CONT(block_tables); CHECK_CONT(seq_lens);
  const int B = q.size(0), Hq = q.size(1), D = q.size(2);
  const int Hkv = k_cache.size(2);
  const int kv_block = k_cache.size(1);
  const int max_blocks = block_tables.size(1);
  TORCH_CHECK(D == 128, "decode kernel requires head_dim 128");
  TORCH_CHECK(Hq % Hkv == 0 && Hq / Hkv <= 8, "GQA group must be <= 8");
  launch_decode_attn(out.data_ptr(), q.data_ptr(), k_cache.data_ptr(),
                     v_cache.data_ptr(), block_tables.data_ptr<int>(),
                     seq_lens.data_ptr<int>(), (float)scale, B, Hq, Hkv, D,
                     max_blocks, kv_block, q_stride, cur_stream());
}

static void prefill_attn_impl(torch::Tensor out, torch::Tensor q, torch::Tensor k_cache,
                              torch::Tensor v_cache, torch::Tensor block_tables,
                              torch::Tensor seq_lens, torch::Tensor ctx_lens,
                              torch::Tensor row_starts, torch::Tensor tile_seq,
                              torch::Tensor tile_q0, double scale, bool mfma) {
  CHECK_CUDA(q); CHECK_BF16(q); CHECK_CONT(out);
  const int64_t q_stride = q_token_stride(q);
  TORCH_CHECK(block_tables.scalar_type() == at::kInt);
  const int Hq = q.size(1), D = q.size(2);
  const int Hkv = k_cache.size(2);
  const int kv_block = k_cache.size(1);
  const int max_blocks = block_tables.size(1);
  const int num_tiles = tile_seq.size(0);
  TORCH_CHECK(D == 128, "prefill kernel requires head_dim 128");
  auto fn = mfma ? launch_prefill_attn_mfma : launch_prefill_attn;
  fn(out.data_ptr(), q.data_ptr(), k_cache.data_ptr(),
     v_cache.data_ptr(), block_tables.data_ptr<int>(),
     seq_lens.data_ptr<int>(), ctx_lens.data_ptr<int>(),
     row_starts.data_ptr<int>(), tile_seq.data_ptr<int>(),
     tile_q0.data_ptr<int>(), (float)scale, num_tiles, Hq, Hkv, D,
     max_blocks, kv_block, q_stride, cur_stream());
}

static void prefill_attn(torch::Tensor out, torch::Tensor q, torch::Tensor k_cache,
                         torch::Tensor v_cache, torch::Tensor block_tables,
                         torch::Tensor seq_lens, torch::Tensor ctx_lens,
                         torch::Tensor row_starts, torch::Tensor tile_seq,
                         torch::Tensor tile_q0, double scale) {
  prefill_attn_impl(out, q, k_cache, v_cache, block_tables, seq_lens, ctx_lens,
                    row_starts, tile_seq, tile_q0, scale, /*mfma=*/false);
}

static void prefill_attn_mfma(torch::Tensor out, torch::Tensor q, torch::Tensor k_cache,
                              torch::Tensor v_cache, torch::Tensor block_tables,
                              torch::Tensor seq_lens, torch::Tensor ctx_lens,
                              torch::Tensor row_starts, torch::Tensor tile_seq,
                              torch::Tensor tile_q0, double scale) {
  prefill_attn_impl(out, q, k_cache, v_cache, block_tables, seq_lens, ctx_lens,
                    row_starts, tile_seq, tile_q0, scale, /*mfma=*/true);
}

// C[M,N] = A[M,K] @ B[N,K]^T — the hand-written 256²-tile 8-phase MFMA
// GEMM (csrc/gemm_bf16.hip) for large-M bf16 projections
static void gemm_bf16(torch::Tensor C, torch::Tensor A, torch::Tensor B) {
  CHECK_CUDA(A); CHECK_CONT(A); CHECK_BF16(A);
  CHECK_CONT(B); CHECK_BF16(B); CHECK_CONT(C); CHECK_BF16(C);
  const int M = A.size(0), K = A.size(1), N = B.size(0);
  TORCH_CHECK(B.size(1) == K && C.size(0) == M && C.size(1) == N,
              "gemm_bf16 shape mismatch");
  TORCH_CHECK(K % 128 == 0, "gemm_bf16 requires K % 128 == 0");
  TORCH_CHECK(N % 256 == 0, "gemm_bf16 requires N % 256 == 0");
  launch_gemm_bf16(C.data_ptr(), A.data_ptr(), B.data_ptr(), M, N, K,
                   cur_stream());
}

static torch::Tensor mfma_probe(torch::Tensor a, torch::Tensor b) {
  CHECK_CUDA(a); CHECK_BF16(a); CHECK_CONT(a); CHECK_CONT(b);
  auto d = torch::empty({32, 32}, a.options().dtype(at::kFloat));
  launch_mfma_probe(d.data_ptr<float>(), a.data_ptr(), b.data_ptr(), cur_stream());
  return d;
}

static void sample(torch::Tensor out, torch::Tensor logits, torch::Tensor temps,
                   torch::Tensor top_ks, torch::Tensor top_ps, torch::Tensor uniforms,
                   c10::optional<torch::Tensor> mask) {
  CHECK_CUDA(logits); CHECK_CONT(logits);
  TORCH_CHECK(logits.scalar_type() == at::kFloat);
  const int B = logits.size(0), V = logits.size(1);
  TORCH_CHECK(V <= 512, "sampling kernel supports live vocab <= 512");
  const uint8_t* mptr = nullptr;
  if (mask.has_value()) {
    TORCH_CHECK(mask->scalar_type() == at::kBool && mask->is_contiguous());
    mptr = (const uint8_t*)mask->data_ptr();
  }
  launch_sample(out.data_ptr<int64_t>(), logits.data_ptr<float>(),
                temps.data_ptr<float>(), top_ks.data_ptr<int64_t>(),
                top_ps.data_ptr<float>(), uniforms.data_ptr<float>(), mptr, B, V,
                cur_stream());
}

static void sample_fullvocab(torch::Tensor out, torch::Tensor logits,
                             torch::Tensor temps, torch::Tensor top_ks,
                             torch::Tensor top_ps, torch::Tensor uniforms,
                             c10::optional<torch::Tensor> mask,
                             c10::optional<torch::Tensor> mask_map) {
  CHECK_CUDA(logits); CHECK_CONT(logits);
  const bool bf16 = logits.scalar_type() == at::kBFloat16;
  TORCH_CHECK(bf16 || logits.scalar_type() == at::kFloat,
              "sample_fullvocab wants bf16 or float logits");
  const int B = logits.size(0), V = logits.size(1);
  TORCH_CHECK(V <= (1 << 18), "sample_fullvocab supports vocab <= 256k");
  const uint8_t* mptr = nullptr;
  const int32_t* mapptr = nullptr;
  if (mask.has_value()) {
    TORCH_CHECK(mask->scalar_type() == at::kBool && mask->is_contiguous());
    TORCH_CHECK(mask_map.has_value(), "compact mask needs mask_map");
    TORCH_CHECK(mask_map->scalar_type() == at::kInt && mask_map->is_contiguous());
    TORCH_CHECK(mask_map->numel() == B);
    mptr = (const uint8_t*)mask->data_ptr();
    mapptr = mask_map->data_ptr<int32_t>();
  }
  launch_sample_fullvocab(out.data_ptr<int64_t>(), logits.data_ptr(), bf16,
                          temps.data_ptr<float>(), top_ks.data_ptr<int64_t>(),
                          top_ps.data_ptr<float>(), uniforms.data_ptr<float>(),
                          mptr, mapptr, B, V, cur_stream());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "MI355X-native runtime: paged-KV block manager + CDNA4 kernels";
  register_block_manager(m);
  m.def("rmsnorm", &rmsnorm);
  m.def("fused_add_rmsnorm", &fused_add_rmsnorm);
  m.def("swiglu", &swiglu);
  m.def("rope_cache", &rope_cache);
  m.def("decode_attn", &decode_attn);
  m.def("prefill_attn", &prefill_attn);
  m.def("prefill_attn_mfma", &prefill_attn_mfma);
  m.def("mfma_probe", &mfma_probe);
  m.def("gemm_bf16", &gemm_bf16);
  m.def("sample", &sample);
  m.def("sample_fullvocab", &sample_fullvocab);
}
