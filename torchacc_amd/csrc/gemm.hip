// Tuned bf16 GEMM on hipBLASLt with explicit algorithm selection.
//
// PyTorch's at::matmul takes hipBLASLt's heuristic top-1, which measured at
// ~52-60% of bf16 peak on the Llama-7B/70B training shapes
// (profiles/r01_7b_fsdp1.md) — 67% of step time. This module exposes the
// full solution space (hipblaslt_ext::getAllAlgos) so an offline harness
// (benchmarks/gemm_tune.py) can time every supported algorithm per shape and
// pin the winner; ops/linear.py routes the model projections through
// lt_gemm with the cached algo index.
//
// Row-major semantics throughout: out[M,N] = opA(a) @ opB(b), bf16 in/out,
// fp32 accumulate. Column-major mapping: D_cm(N,M) = opB(b_cm) * opA(a_cm)
// with operands swapped and trans flags carried over.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hipblaslt/hipblaslt.h>
#include <hipblaslt/hipblaslt-ext.hpp>

#include <memory>
#include <mutex>
#include <string>
#include <unordered_map>
#include <vector>

#define HIPBLASLT_CHECK(expr)                                               \
  do {                                                                      \
    hipblasStatus_t s_ = (expr);                                            \
    TORCH_CHECK(s_ == HIPBLAS_STATUS_SUCCESS, "hipblaslt error ", (int)s_,  \
                " at " #expr);                                              \
  } while (0)

namespace {

hipblasLtHandle_t lt_handle() {
  static hipblasLtHandle_t handle = [] {
    hipblasLtHandle_t h;
    HIPBLASLT_CHECK(hipblasLtCreate(&h));
    return h;
  }();
  return handle;
}

// workspace sized for split-k on the wgrad shapes; allocated through the
// torch caching allocator so it participates in memory accounting
torch::Tensor& lt_workspace() {
  static torch::Tensor ws;
  if (!ws.defined()) {
    const char* env = getenv("TA_GEMM_WORKSPACE_MB");
    int64_t mb = env ? atol(env) : 256;
    ws = torch::empty({mb * (1 << 20)},
                      torch::dtype(torch::kByte).device(torch::kCUDA));
  }
  return ws;
}

struct LtDesc {
  hipblasLtMatmulDesc_t op = nullptr;
  hipblasLtMatrixLayout_t la = nullptr, lb = nullptr, lc = nullptr;
  ~LtDesc() {
    if (op) hipblasLtMatmulDescDestroy(op);
    if (la) hipblasLtMatrixLayoutDestroy(la);
    if (lb) hipblasLtMatrixLayoutDestroy(lb);
    if (lc) hipblasLtMatrixLayoutDestroy(lc);
  }
};

// Build descriptors for row-major out[M,N] = opA(a[M,K or K,M]) @ opB(b).
// After the col-major swap: operand1 = b (op=tb), operand2 = a (op=ta),
// layouts are the physical row-major storages reinterpreted col-major.
void build_desc(LtDesc& d, int64_t m, int64_t n, int64_t k, bool ta, bool tb,
                int64_t lda, int64_t ldb) {
  HIPBLASLT_CHECK(hipblasLtMatmulDescCreate(&d.op, HIPBLAS_COMPUTE_32F,
                                            HIP_R_32F));
  hipblasOperation_t opA = tb ? HIPBLAS_OP_T : HIPBLAS_OP_N;
  hipblasOperation_t opB = ta ? HIPBLAS_OP_T : HIPBLAS_OP_N;
  HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
      d.op, HIPBLASLT_MATMUL_DESC_TRANSA, &opA, sizeof(opA)));
  HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
      d.op, HIPBLASLT_MATMUL_DESC_TRANSB, &opB, sizeof(opB)));
  // operand1 = b: row-major [k,n] (tb=N, ld=ldb) or [n,k] (tb=T, ld=ldb);
  // as col-major physical (ldb-major): (n,k) for N (ld=n... use ldb), etc.
  if (!tb) {  // b is [K,N] row-major -> col-major (N,K), op N
    HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&d.la, HIP_R_16BF, n, k, ldb));
  } else {  // b is [N,K] row-major -> col-major (K,N), op T
    HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&d.la, HIP_R_16BF, k, n, ldb));
  }
  if (!ta) {  // a is [M,K] row-major -> col-major (K,M), op N
    HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&d.lb, HIP_R_16BF, k, m, lda));
  } else {  // a is [K,M] row-major -> col-major (M,K), op T
    HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&d.lb, HIP_R_16BF, m, k, lda));
  }
  HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&d.lc, HIP_R_16BF, n, m, n));
}

void check_inputs(const torch::Tensor& a, const torch::Tensor& b, bool ta,
                  bool tb, int64_t& m, int64_t& n, int64_t& k) {
  TORCH_CHECK(a.is_cuda() && b.is_cuda(), "lt_gemm: GPU tensors required");
  TORCH_CHECK(a.scalar_type() == torch::kBFloat16 &&
                  b.scalar_type() == torch::kBFloat16,
              "lt_gemm: bf16 only");
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2, "lt_gemm: 2-D inputs");
  TORCH_CHECK(a.is_contiguous() && b.is_contiguous(),
              "lt_gemm: contiguous inputs");
  m = ta ? a.size(1) : a.size(0);
  k = ta ? a.size(0) : a.size(1);
  int64_t kb = tb ? b.size(1) : b.size(0);
  n = tb ? b.size(0) : b.size(1);
  TORCH_CHECK(k == kb, "lt_gemm: inner dims mismatch ", k, " vs ", kb);
}

}  // namespace

// Candidate algo indices for the problem, cheapest-estimated first.
// max_workspace_mb filters out algos needing more scratch than we keep.
std::vector<int64_t> lt_gemm_candidates(int64_t m, int64_t n, int64_t k,
                                        bool ta, bool tb,
                                        int64_t max_workspace_mb) {
  LtDesc d;
  int64_t lda = ta ? m : k;
  int64_t ldb = tb ? k : n;
  build_desc(d, m, n, k, ta, tb, lda, ldb);
  std::vector<hipblasLtMatmulHeuristicResult_t> all;
  HIPBLASLT_CHECK(hipblaslt_ext::getAllAlgos(
      lt_handle(), hipblaslt_ext::GemmType::HIPBLASLT_GEMM,
      tb ? HIPBLAS_OP_T : HIPBLAS_OP_N, ta ? HIPBLAS_OP_T : HIPBLAS_OP_N,
      HIP_R_16BF, HIP_R_16BF, HIP_R_16BF, HIP_R_16BF, HIPBLAS_COMPUTE_32F,
      all));
  std::vector<int64_t> out;
  float alpha = 1.0f, beta = 0.0f;
  size_t ws_cap = (size_t)max_workspace_mb << 20;
  for (auto& h : all) {
    size_t ws = 0;
    auto st = hipblaslt_ext::matmulIsAlgoSupported(
        lt_handle(), d.op, &alpha, d.la, d.lb, &beta, d.lc, d.lc, h.algo, ws);
    if (st == HIPBLAS_STATUS_SUCCESS && ws <= ws_cap) {
      out.push_back(hipblaslt_ext::getIndexFromAlgo(h.algo));
    }
  }
  return out;
}

namespace {

// Per-problem plan cache: descriptors + resolved algo. The algo lookup
// (getAlgosFromIndex / heuristic query) costs host-side milliseconds — at
// ~460 GEMM calls per training step an uncached lookup dominates the step.
struct LtPlan {
  LtDesc desc;
  hipblasLtMatmulAlgo_t algo;
};

std::unordered_map<std::string, std::unique_ptr<LtPlan>> plan_cache;
std::mutex plan_mutex;

LtPlan* get_plan(int64_t m, int64_t n, int64_t k, bool ta, bool tb,
                 int64_t algo_index) {
  char key[96];
  snprintf(key, sizeof(key), "%ld,%ld,%ld,%d%d,%ld", (long)m, (long)n,
           (long)k, (int)ta, (int)tb, (long)algo_index);
  std::lock_guard<std::mutex> lock(plan_mutex);
  auto it = plan_cache.find(key);
  if (it != plan_cache.end()) return it->second.get();

  auto plan = std::make_unique<LtPlan>();
  int64_t lda = ta ? m : k;
  int64_t ldb = tb ? k : n;
  build_desc(plan->desc, m, n, k, ta, tb, lda, ldb);
  float alpha = 1.0f, beta = 0.0f;
  auto& ws = lt_workspace();
  bool have_algo = false;
  if (algo_index >= 0) {
    std::vector<hipblasLtMatmulHeuristicResult_t> fetched;
    std::vector<int> idx{(int)algo_index};
    auto st = hipblaslt_ext::getAlgosFromIndex(lt_handle(), idx, fetched);
    if (st == HIPBLAS_STATUS_SUCCESS && !fetched.empty()) {
      size_t ws_need = 0;
      auto sup = hipblaslt_ext::matmulIsAlgoSupported(
          lt_handle(), plan->desc.op, &alpha, plan->desc.la, plan->desc.lb,
          &beta, plan->desc.lc, plan->desc.lc, fetched[0].algo, ws_need);
      if (sup == HIPBLAS_STATUS_SUCCESS && ws_need <= (size_t)ws.numel()) {
        plan->algo = fetched[0].algo;
        have_algo = true;
      }
    }
    TORCH_CHECK(have_algo, "lt_gemm: algo index ", algo_index,
                " unsupported for ", m, "x", n, "x", k);
  } else {
    hipblasLtMatmulPreference_t pref;
    HIPBLASLT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
    size_t ws_sz = (size_t)ws.numel();
    HIPBLASLT_CHECK(hipblasLtMatmulPreferenceSetAttribute(
        pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws_sz,
        sizeof(ws_sz)));
    hipblasLtMatmulHeuristicResult_t res;
    int returned = 0;
    HIPBLASLT_CHECK(hipblasLtMatmulAlgoGetHeuristic(
        lt_handle(), plan->desc.op, plan->desc.la, plan->desc.lb,
        plan->desc.lc, plan->desc.lc, pref, 1, &res, &returned));
    hipblasLtMatmulPreferenceDestroy(pref);
    TORCH_CHECK(returned > 0, "lt_gemm: no heuristic algo for ", m, "x", n,
                "x", k);
    plan->algo = res.algo;
  }
  LtPlan* raw = plan.get();
  plan_cache[key] = std::move(plan);
  return raw;
}

}  // namespace

// out[M,N] = opA(a) @ opB(b); algo_index < 0 -> heuristic top-1 (cached).
torch::Tensor lt_gemm(torch::Tensor a, torch::Tensor b, bool ta, bool tb,
                      int64_t algo_index, c10::optional<torch::Tensor> out_opt) {
  int64_t m, n, k;
  check_inputs(a, b, ta, tb, m, n, k);
  torch::Tensor out =
      out_opt.has_value() ? *out_opt : torch::empty({m, n}, a.options());
  TORCH_CHECK(out.is_contiguous() && out.size(0) == m && out.size(1) == n);
  LtPlan* plan = get_plan(m, n, k, ta, tb, algo_index);
  float alpha = 1.0f, beta = 0.0f;
  auto stream = at::hip::getCurrentHIPStream();
  auto& ws = lt_workspace();
  HIPBLASLT_CHECK(hipblasLtMatmul(
      lt_handle(), plan->desc.op, &alpha, b.data_ptr(), plan->desc.la,
      a.data_ptr(), plan->desc.lb, &beta, out.data_ptr(), plan->desc.lc,
      out.data_ptr(), plan->desc.lc, &plan->algo, ws.data_ptr(),
      (size_t)ws.numel(), stream));
  return out;
}

// Solution kernel name for profiling reports.
std::string lt_gemm_algo_name(int64_t algo_index) {
  std::vector<hipblasLtMatmulHeuristicResult_t> fetched;
  std::vector<int> idx{(int)algo_index};
  auto st = hipblaslt_ext::getAlgosFromIndex(lt_handle(), idx, fetched);
  if (st != HIPBLAS_STATUS_SUCCESS || fetched.empty()) return "";
  return hipblaslt_ext::getSolutionNameFromAlgo(lt_handle(), fetched[0].algo);
}

// ---------------------------------------------------------------------------
// Skinny GEMV for decode: y[M,N] = x[M,K] @ W[N,K]^T with M <= 8.
// hipBLASLt's m=1 kernels measure ~2.2 TB/s effective on the Llama decode
// projections; this wave-per-output-row kernel streams W once at wide
// vector width (16 B/lane) with the tiny x staying L2-hot, and amortizes
// the W read over all M rows.
// ---------------------------------------------------------------------------
#include "common.h"
#include "attn_common.h"

template <bool FP16, int M>
__global__ __launch_bounds__(256)
void gemv_kernel(const short* __restrict__ x, const short* __restrict__ w,
                 short* __restrict__ y, int n, int k) {
  const int wid = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int waves = gridDim.x * blockDim.x / WAVE;
  for (int row = wid; row < n; row += waves) {
    const short* wr = w + (long)row * k;
    float acc[M];
#pragma unroll
    for (int m = 0; m < M; ++m) acc[m] = 0.f;
    for (int i = lane * 8; i < k; i += WAVE * 8) {
      s16x8 wv = *reinterpret_cast<const s16x8*>(wr + i);
      float wf[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) wf[j] = AttnElem<FP16>::to_f32(wv[j]);
#pragma unroll
      for (int m = 0; m < M; ++m) {
        s16x8 xv = *reinterpret_cast<const s16x8*>(x + (long)m * k + i);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[m] += wf[j] * AttnElem<FP16>::to_f32(xv[j]);
      }
    }
#pragma unroll
    for (int m = 0; m < M; ++m) {
      float v = wave_reduce_sum(acc[m]);
      if (lane == 0) y[(long)m * n + row] = AttnElem<FP16>::from_f32(v);
    }
  }
}

// y = x @ w^T, x [M<=8, K], w [N, K]; bf16/fp16.
torch::Tensor lt_gemv(torch::Tensor x, torch::Tensor w) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && w.dim() == 2);
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous());
  const bool fp16 = x.scalar_type() == torch::kHalf;
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 || fp16);
  const int m = x.size(0), k = x.size(1), n = w.size(0);
  TORCH_CHECK(w.size(1) == k && m >= 1 && m <= 8 && k % 8 == 0);
  auto y = torch::empty({m, n}, x.options());
  auto stream = at::hip::getCurrentHIPStream();
  const int waves_wanted = std::min(n, 4096);
  dim3 grid((waves_wanted + 3) / 4), block(256);
#define GEMV_CASE(MM)                                                        \
  case MM:                                                                   \
    if (fp16)                                                                \
      hipLaunchKernelGGL((gemv_kernel<true, MM>), grid, block, 0, stream,    \
                         (const short*)x.data_ptr(),                         \
                         (const short*)w.data_ptr(), (short*)y.data_ptr(),   \
                         n, k);                                              \
    else                                                                     \
      hipLaunchKernelGGL((gemv_kernel<false, MM>), grid, block, 0, stream,   \
                         (const short*)x.data_ptr(),                         \
                         (const short*)w.data_ptr(), (short*)y.data_ptr(),   \
                         n, k);                                              \
    break;
  switch (m) {
    GEMV_CASE(1) GEMV_CASE(2) GEMV_CASE(3) GEMV_CASE(4)
    GEMV_CASE(5) GEMV_CASE(6) GEMV_CASE(7) GEMV_CASE(8)
  }
#undef GEMV_CASE
  HIP_CHECK_LAST();
  return y;
}
