// PyTorch bindings for the MI355X HIP kernel library.
//
// Compiled natively with hipcc against libtorch's ROCm build (no hipify,
// no CUDA shims — the kernels are written directly for gfx950).

#include <torch/extension.h>

#include <ATen/hip/HIPContext.h>
#include <ATen/hip/impl/HIPGuardImplMasqueradingAsCUDA.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <hip/hip_runtime.h>

extern "C" void launch_als_solve(
    const long long* indptr, const int* indices, const float* values,
    const float* Y, const float* YtY, const float* V, float* X,
    int n_rows, int f, float lambda, float alpha,
    int implicit_mode, int wr_scale, int which,
    unsigned long long* prof, hipStream_t stream);

extern "C" void launch_topk_score(
    const float* Xq, const float* Y, const uint8_t* item_mask,
    const long long* ban_indptr, const int* ban_indices,
    float* out_val, int* out_idx,
    int B, long long N, int f, int K, int n_slices, int item_base,
    unsigned long long* prof, hipStream_t stream);

extern "C" void launch_topk_mfma(
    const unsigned short* Xq, const unsigned short* Y,
    const uint8_t* item_mask, const long long* ban_indptr,
    const int* ban_indices, float* out_val, int* out_idx,
    int B, long long N, int f, int K, int n_slices, int item_base,
    unsigned long long* prof, unsigned* th_g, hipStream_t stream);

namespace {

void check_cuda_f32(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be fp32");
}

bool supported_rank(int64_t f) {
  return f == 16 || f == 32 || f == 64 || f == 128;
}

// Solve one ALS half-iteration: for every CSR row r,
//   explicit: (sum_i y_i y_i^T + lambda*nnz_r*I) x_r = sum_i r_i y_i
//   implicit: (YtY + sum_i alpha*r * y_i y_i^T + lambda*I) x_r
//               = sum_i (1+alpha*r) y_i
// V = Y L^-T (L = chol(YtY + lambda I)) precomputed on the host per
// half-iteration enables the per-row Woodbury fast path in implicit mode
// (see als_woodbury_kernel). which: 0 = both passes, 1 = Woodbury rows
// only (emits Z), 2 = dense rows only (writes into `out`).
torch::Tensor als_solve(torch::Tensor indptr, torch::Tensor indices,
                        torch::Tensor values, torch::Tensor Y,
                        c10::optional<torch::Tensor> YtY,
                        c10::optional<torch::Tensor> V,
                        double lambda, double alpha,
                        bool implicit_mode, bool wr_scale,
                        int64_t which, c10::optional<torch::Tensor> out,
                        c10::optional<torch::Tensor> prof) {
  TORCH_CHECK(indptr.is_cuda() && indptr.scalar_type() == torch::kInt64 &&
                  indptr.is_contiguous(), "indptr must be contiguous i64 GPU");
  TORCH_CHECK(indices.is_cuda() && indices.scalar_type() == torch::kInt32 &&
                  indices.is_contiguous(), "indices must be contiguous i32 GPU");
  check_cuda_f32(values, "values");
  check_cuda_f32(Y, "Y");
  const int64_t f = Y.size(1);
  TORCH_CHECK(supported_rank(f), "rank must be one of 16/32/64/128, got ", f);
  const int64_t n_rows = indptr.size(0) - 1;
  TORCH_CHECK(indices.size(0) == values.size(0), "indices/values mismatch");
  const float* yty_ptr = nullptr;
  if (YtY.has_value()) {
    check_cuda_f32(*YtY, "YtY");
    TORCH_CHECK(YtY->size(0) == f && YtY->size(1) == f, "YtY must be f x f");
    yty_ptr = YtY->data_ptr<float>();
  }
  const float* v_ptr = nullptr;
  if (V.has_value()) {
    // V may be bf16 when PIO_ALS_STAGE_BF16 is set (the kernel stages
    // it as bf16; the pointer is reinterpreted device-side)
    TORCH_CHECK(V->is_cuda() && V->is_contiguous() &&
                    (V->scalar_type() == torch::kFloat32 ||
                     V->scalar_type() == torch::kBFloat16),
                "V must be contiguous fp32/bf16 GPU");
    TORCH_CHECK(V->sizes() == Y.sizes(), "V must match Y shape");
    const bool env_bf16 = [] {
      const char* e = getenv("PIO_ALS_STAGE_BF16");
      return e != nullptr && e[0] == '1';
    }();
    TORCH_CHECK((V->scalar_type() == torch::kBFloat16) == env_bf16,
                "V dtype must match PIO_ALS_STAGE_BF16");
    v_ptr = reinterpret_cast<const float*>(V->data_ptr());
  }
  unsigned long long* prof_ptr = nullptr;
  if (prof.has_value()) {
    TORCH_CHECK(prof->is_cuda() && prof->is_contiguous() &&
                prof->scalar_type() == torch::kUInt64 &&
                prof->numel() >= 5,
                "prof must be a contiguous u64 GPU tensor of >= 5");
    prof_ptr = reinterpret_cast<unsigned long long*>(
        prof->data_ptr());
  }
  torch::Tensor X;
  if (out.has_value()) {
    check_cuda_f32(*out, "out");
    TORCH_CHECK(out->size(0) == n_rows && out->size(1) == f,
                "out must be n_rows x f");
    X = *out;
  } else {
    X = torch::empty({n_rows, f}, Y.options());
  }
  c10::hip::HIPGuardMasqueradingAsCUDA guard(Y.device());
  hipStream_t stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  launch_als_solve(reinterpret_cast<const long long*>(indptr.data_ptr<int64_t>()), indices.data_ptr<int>(),
                   values.data_ptr<float>(), Y.data_ptr<float>(), yty_ptr,
                   v_ptr, X.data_ptr<float>(), (int)n_rows, (int)f, (float)lambda,
                   (float)alpha, implicit_mode ? 1 : 0, wr_scale ? 1 : 0,
                   (int)which, prof_ptr, stream);
  C10_HIP_CHECK(hipGetLastError());
  return X;
}

// Fused masked top-K scoring: per-slice candidate groups; caller merges
// with a small torch.topk. Returns (vals [B, n_slices*K], idx i32).
std::tuple<torch::Tensor, torch::Tensor> topk_score(
    torch::Tensor Xq, torch::Tensor Y, int64_t K, int64_t n_slices,
    c10::optional<torch::Tensor> item_mask,
    c10::optional<torch::Tensor> ban_indptr,
    c10::optional<torch::Tensor> ban_indices, int64_t item_base,
    c10::optional<torch::Tensor> prof) {
  check_cuda_f32(Xq, "Xq");
  check_cuda_f32(Y, "Y");
  const int64_t f = Y.size(1);
  TORCH_CHECK(supported_rank(f), "rank must be one of 16/32/64/128, got ", f);
  TORCH_CHECK(Xq.size(1) == f, "Xq/Y rank mismatch");
  TORCH_CHECK(K >= 1 && K <= 64, "K must be in [1, 64]");
  const int64_t B = Xq.size(0);
  const int64_t N = Y.size(0);
  TORCH_CHECK(n_slices >= 1);
  const uint8_t* mask_ptr = nullptr;
  if (item_mask.has_value()) {
    TORCH_CHECK(item_mask->is_cuda() && item_mask->is_contiguous() &&
                    item_mask->scalar_type() == torch::kUInt8,
                "item_mask must be contiguous u8 GPU");
    TORCH_CHECK(item_mask->size(0) == N, "item_mask size mismatch");
    mask_ptr = item_mask->data_ptr<uint8_t>();
  }
  const long long* bi_ptr = nullptr;
  const int* bx_ptr = nullptr;
  if (ban_indptr.has_value()) {
    TORCH_CHECK(ban_indices.has_value(), "ban_indices required with ban_indptr");
    TORCH_CHECK(ban_indptr->is_cuda() && ban_indptr->is_contiguous() &&
                ban_indptr->scalar_type() == torch::kInt64);
    TORCH_CHECK(ban_indices->is_cuda() && ban_indices->is_contiguous() &&
                ban_indices->scalar_type() == torch::kInt32);
    TORCH_CHECK(ban_indptr->size(0) == B + 1, "ban_indptr must be B+1");
    bi_ptr = reinterpret_cast<const long long*>(ban_indptr->data_ptr<int64_t>());
    bx_ptr = ban_indices->data_ptr<int>();
  }
  // the kernel emits TK_WAVES=4 independent candidate groups per slice
  auto out_val = torch::empty({B, n_slices * 4 * K}, Xq.options());
  auto out_idx = torch::empty({B, n_slices * 4 * K},
                              Xq.options().dtype(torch::kInt32));
  unsigned long long* prof_ptr = nullptr;
  if (prof.has_value()) {
    TORCH_CHECK(prof->is_cuda() && prof->is_contiguous() &&
                    prof->scalar_type() == torch::kUInt64 &&
                    prof->numel() >= 5,
                "prof must be a contiguous u64[5] GPU tensor");
    prof_ptr = reinterpret_cast<unsigned long long*>(
        prof->data_ptr<uint64_t>());
  }
  c10::hip::HIPGuardMasqueradingAsCUDA guard(Y.device());
  hipStream_t stream = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  launch_topk_score(Xq.data_ptr<float>(), Y.data_ptr<float>(), mask_ptr,
                    bi_ptr, bx_ptr, out_val.data_ptr<float>(),
                    out_idx.data_ptr<int>(), (int)B, (long long)N, (int)f,
                    (int)K, (int)n_slices, (int)item_base, prof_ptr, stream);
  C10_HIP_CHECK(hipGetLastError());
  return {out_val, out_idx};
}

// MFMA variant of topk_score: bf16 query/item factors (fp32 accumulate on
// the matrix cores). Same output layout as topk_score — candidate groups
// (n_slices*4) per query; the Python wrapper merges and fp32-rescores.
std::tuple<torch::Tensor, torch::Tensor> topk_score_mfma(
    torch::Tensor Xq, torch::Tensor Y, int64_t K, int64_t n_slices,
    c10::optional<torch::Tensor> item_mask,
    c10::optional<torch::Tensor> ban_indptr,
    c10::optional<torch::Tensor> ban_indices, int64_t item_base,
    c10::optional<torch::Tensor> prof) {
  TORCH_CHECK(Xq.is_cuda() && Xq.is_contiguous() &&
                  Xq.scalar_type() == torch::kBFloat16,
              "Xq must be contiguous bf16 GPU");
  TORCH_CHECK(Y.is_cuda() && Y.is_contiguous() &&
                  Y.scalar_type() == torch::kBFloat16,
              "Y must be contiguous bf16 GPU");
  const int64_t f = Y.size(1);
  TORCH_CHECK(f == 32 || f == 64 || f == 128,
              "mfma rank must be one of 32/64/128, got ", f);
  TORCH_CHECK(Xq.size(1) == f, "Xq/Y rank mismatch");
  TORCH_CHECK(K >= 1 && K <= 64, "K must be in [1, 64]");
  const int64_t B = Xq.size(0);
  const int64_t N = Y.size(0);
  TORCH_CHECK(n_slices >= 1);
  const uint8_t* mask_ptr = nullptr;
  if (item_mask.has_value()) {
    TORCH_CHECK(item_mask->is_cuda() && item_mask->is_contiguous() &&
                    item_mask->scalar_type() == torch::kUInt8,
                "item_mask must be contiguous u8 GPU");
    TORCH_CHECK(item_mask->size(0) == N, "item_mask size mismatch");
    mask_ptr = item_mask->data_ptr<uint8_t>();
  }
  const long long* bi_ptr = nullptr;
  const int* bx_ptr = nullptr;
  if (ban_indptr.has_value()) {
    TORCH_CHECK(ban_indices.has_value(), "ban_indices required");
    TORCH_CHECK(ban_indptr->is_cuda() && ban_indptr->is_contiguous() &&
                ban_indptr->scalar_type() == torch::kInt64);
    TORCH_CHECK(ban_indices->is_cuda() && ban_indices->is_contiguous() &&
                ban_indices->scalar_type() == torch::kInt32);
    TORCH_CHECK(ban_indptr->size(0) == B + 1, "ban_indptr must be B+1");
    bi_ptr = reinterpret_cast<const long long*>(
        ban_indptr->data_ptr<int64_t>());
    bx_ptr = ban_indices->data_ptr<int>();
  }
  // one candidate group per slice per query (the kernel keeps a single
  // shared top-K list per query per workgroup)
  auto out_val = torch::empty({B, n_slices * K},
                              Xq.options().dtype(torch::kFloat32));
  auto out_idx = torch::empty({B, n_slices * K},
                              Xq.options().dtype(torch::kInt32));
  unsigned long long* prof_ptr = nullptr;
  if (prof.has_value()) {
    TORCH_CHECK(prof->is_cuda() && prof->is_contiguous() &&
                    prof->scalar_type() == torch::kUInt64 &&
                    prof->numel() >= 5,
                "prof must be a contiguous u64[5] GPU tensor");
    prof_ptr = reinterpret_cast<unsigned long long*>(
        prof->data_ptr<uint64_t>());
  }
  c10::hip::HIPGuardMasqueradingAsCUDA guard(Y.device());
  hipStream_t stream =
      c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  // cross-slice threshold exchange (default ON; PIO_TOPK_GTH=0
  // disables): a [B] tm_enc-coded atomicMax cell per query, zero-init
  // (= below every real score), shared by all of a query's slice WGs
  // through L2. +17-29% for B in [64, 4096]; the kernel's
  // test-and-test-and-set read keeps the atomic queue short (the
  // unfiltered version serialized at small B —
  // profiles/serve_gth_ab_r2.log).
  torch::Tensor th_g;
  unsigned* th_g_ptr = nullptr;
  const char* e_gth = getenv("PIO_TOPK_GTH");
  const bool gth = !(e_gth != nullptr && e_gth[0] == '0');
  if (gth) {
    th_g = torch::zeros({B}, Xq.options().dtype(torch::kInt32));
    th_g_ptr = reinterpret_cast<unsigned*>(th_g.data_ptr<int>());
  }
  launch_topk_mfma(
      reinterpret_cast<const unsigned short*>(Xq.data_ptr()),
      reinterpret_cast<const unsigned short*>(Y.data_ptr()), mask_ptr,
      bi_ptr, bx_ptr, out_val.data_ptr<float>(), out_idx.data_ptr<int>(),
      (int)B, (long long)N, (int)f, (int)K, (int)n_slices, (int)item_base,
      prof_ptr, th_g_ptr, stream);
  C10_HIP_CHECK(hipGetLastError());
  return {out_val, out_idx};
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "MI355X (gfx950) HIP kernels for predictionio_amd";
  m.def("als_solve", &als_solve, "Fused ALS Gramian+Cholesky half-iteration",
        py::arg("indptr"), py::arg("indices"), py::arg("values"),
        py::arg("Y"), py::arg("YtY") = py::none(), py::arg("V") = py::none(),
        py::arg("lambda_") = 0.01,
        py::arg("alpha") = 1.0, py::arg("implicit_mode") = false,
        py::arg("wr_scale") = true, py::arg("which") = 0,
        py::arg("out") = py::none(), py::arg("prof") = py::none());
  m.def("topk_score", &topk_score, "Fused masked top-K scoring",
        py::arg("Xq"), py::arg("Y"), py::arg("K"), py::arg("n_slices") = 64,
        py::arg("item_mask") = py::none(), py::arg("ban_indptr") = py::none(),
        py::arg("ban_indices") = py::none(), py::arg("item_base") = 0,
        py::arg("prof") = py::none());
  m.def("topk_score_mfma", &topk_score_mfma,
        "Fused masked top-K scoring on MFMA (bf16 in, fp32 accumulate)",
        py::arg("Xq"), py::arg("Y"), py::arg("K"), py::arg("n_slices") = 64,
        py::arg("item_mask") = py::none(), py::arg("ban_indptr") = py::none(),
        py::arg("ban_indices") = py::none(), py::arg("item_base") = 0,
        py::arg("prof") = py::none());
}
