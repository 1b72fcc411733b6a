// hipBLASLt fused-epilogue GEMMs for the transformer FFN hot path.
//
// MI355X design: fc1's bias+GELU and fc2-dgrad's dGELU+bias-grad run as
// hipBLASLt epilogues inside the Tensile GEMM kernels, so the 2x
// [tokens, 4h] bf16 activation tensor is never re-read by a separate
// elementwise pass (HBM is the bound at ~8 TB/s; the epilogue is free).
// Replaces the reference's fused_gemm_epilogue_kernel.cu
// (paddle/phi/kernels/fusion/gpu/, cublasLt path) with the ROCm-native
// equivalent.  GELU flavor is Tensile's tanh approximation.
#include <hipblaslt/hipblaslt.h>
#include <torch/extension.h>

#include <c10/hip/HIPStream.h>
#include <map>
#include <mutex>
#include <tuple>

#define LT_CHECK(expr)                                                        \
  do {                                                                        \
    hipblasStatus_t s_ = (expr);                                              \
    TORCH_CHECK(s_ == HIPBLAS_STATUS_SUCCESS, "hipblaslt error ", (int)s_,    \
                " at " #expr);                                                \
  } while (0)

namespace pa_lt {

using torch::Tensor;

static hipblasLtHandle_t handle() {
  static hipblasLtHandle_t h = [] {
    hipblasLtHandle_t t;
    LT_CHECK(hipblasLtCreate(&t));
    return t;
  }();
  return h;
}

static constexpr size_t kWorkspace = 64u << 20;

static Tensor workspace(const Tensor& like) {
  static Tensor ws;  // cached; device never changes in practice (1 proc/GPU)
  if (!ws.defined() || ws.device() != like.device())
    ws = torch::empty({(int64_t)kWorkspace},
                      like.options().dtype(torch::kUInt8));
  return ws;
}

struct AlgoKey {
  int64_t m, n, k;
  int epi;
  bool operator<(const AlgoKey& o) const {
    return std::tie(m, n, k, epi) < std::tie(o.m, o.n, o.k, o.epi);
  }
};
static std::map<AlgoKey, hipblasLtMatmulAlgo_t> g_algos;
static std::mutex g_mu;

// Run D[colmajor N x M] = op(A) * op(B) (+ epilogue) in fp32 compute over
// bf16 operands.  All the fused-FFN calls reduce to this.
static void lt_matmul(hipblasOperation_t opA, hipblasOperation_t opB,
                      int64_t rows, int64_t cols, int64_t kk,
                      const void* A, int64_t lda, const void* B, int64_t ldb,
                      void* D, int64_t ldd, hipblasLtEpilogue_t epi,
                      const void* bias, void* aux, int64_t aux_ld,
                      const Tensor& ref) {
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipblasLtMatmulDesc_t desc;
  LT_CHECK(hipblasLtMatmulDescCreate(&desc, HIPBLAS_COMPUTE_32F, HIP_R_32F));
  LT_CHECK(hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_TRANSA,
                                           &opA, sizeof(opA)));
  LT_CHECK(hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_TRANSB,
                                           &opB, sizeof(opB)));
  LT_CHECK(hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_EPILOGUE,
                                           &epi, sizeof(epi)));
  if (bias)
    LT_CHECK(hipblasLtMatmulDescSetAttribute(
        desc, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &bias, sizeof(bias)));
  if (aux) {
    LT_CHECK(hipblasLtMatmulDescSetAttribute(
        desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_POINTER, &aux, sizeof(aux)));
    LT_CHECK(hipblasLtMatmulDescSetAttribute(
        desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_LD, &aux_ld, sizeof(aux_ld)));
    int32_t auxdt = HIP_R_16BF;
    LT_CHECK(hipblasLtMatmulDescSetAttribute(
        desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_DATA_TYPE, &auxdt,
        sizeof(auxdt)));
  }

  // layouts: col-major; A is (rows x kk) after opA, B is (kk x cols) after opB
  int64_t a_r = opA == HIPBLAS_OP_N ? rows : kk;
  int64_t a_c = opA == HIPBLAS_OP_N ? kk : rows;
  int64_t b_r = opB == HIPBLAS_OP_N ? kk : cols;
  int64_t b_c = opB == HIPBLAS_OP_N ? cols : kk;
  hipblasLtMatrixLayout_t la, lb, ld;
  LT_CHECK(hipblasLtMatrixLayoutCreate(&la, HIP_R_16BF, a_r, a_c, lda));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&lb, HIP_R_16BF, b_r, b_c, ldb));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&ld, HIP_R_16BF, rows, cols, ldd));

  Tensor ws = workspace(ref);
  AlgoKey key{rows, cols, kk, (int)epi};
  hipblasLtMatmulAlgo_t algo;
  bool have = false;
  {
    std::lock_guard<std::mutex> g(g_mu);
    auto it = g_algos.find(key);
    if (it != g_algos.end()) { algo = it->second; have = true; }
  }
  if (!have) {
    hipblasLtMatmulPreference_t pref;
    LT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
    size_t wsz = kWorkspace;
    LT_CHECK(hipblasLtMatmulPreferenceSetAttribute(
        pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &wsz, sizeof(wsz)));
    hipblasLtMatmulHeuristicResult_t res[4];
    int found = 0;
    LT_CHECK(hipblasLtMatmulAlgoGetHeuristic(handle(), desc, la, lb, ld, ld,
                                             pref, 4, res, &found));
    LT_CHECK(hipblasLtMatmulPreferenceDestroy(pref));
    TORCH_CHECK(found > 0, "hipblaslt: no algo for epilogue GEMM ", rows, "x",
                cols, "x", kk, " epi=", (int)epi);
    algo = res[0].algo;
    std::lock_guard<std::mutex> g(g_mu);
    g_algos[key] = algo;
  }

  float alpha = 1.f, beta = 0.f;
  LT_CHECK(hipblasLtMatmul(handle(), desc, &alpha, A, la, B, lb, &beta, D, ld,
                           D, ld, &algo, ws.data_ptr(), kWorkspace, stream));
  LT_CHECK(hipblasLtMatrixLayoutDestroy(la));
  LT_CHECK(hipblasLtMatrixLayoutDestroy(lb));
  LT_CHECK(hipblasLtMatrixLayoutDestroy(ld));
  LT_CHECK(hipblasLtMatmulDescDestroy(desc));
}

// y, z = gelu(x @ w + b), (x @ w + b)   -- z is the pre-GELU aux saved
// for backward.  Paddle Linear layout: x [M,K] row-major, w [K,N] row-major
// (in_features x out_features), bias [N].
std::tuple<Tensor, Tensor> fc1_gelu_fwd(const Tensor& x, const Tensor& w,
                                        const Tensor& bias) {
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && bias.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 &&
              w.scalar_type() == torch::kBFloat16 &&
              bias.scalar_type() == torch::kBFloat16,
              "fc1_gelu_fwd: bf16 only");
  int64_t M = x.size(0), K = x.size(1), N = w.size(1);
  TORCH_CHECK(w.size(0) == K && bias.size(0) == N);
  auto y = torch::empty({M, N}, x.options());
  auto z = torch::empty({M, N}, x.options());
  // col-major D'[N,M] = A'(w storage viewed N x K, lda=N) * B'(x as K x M):
  // D'[n,m] = sum_k w[k,n] * x[m,k]
  lt_matmul(HIPBLAS_OP_N, HIPBLAS_OP_N, N, M, K, w.const_data_ptr(), N,
            x.const_data_ptr(), K, y.mutable_data_ptr(), N,
            HIPBLASLT_EPILOGUE_GELU_AUX_BIAS, bias.const_data_ptr(),
            z.mutable_data_ptr(), N, x);
  return {y, z};
}

// dg, db = dgelu(dy @ w2, z), colsum(dgelu(...))  -- fc2 dgrad with the
// dGELU+bias-grad epilogue: dy [M,H] row-major, w2 [H,N] row-major,
// z (pre-GELU aux from forward) [M,N].  Returns fc1's dZ and db in one GEMM.
std::tuple<Tensor, Tensor> fc2_dgrad_dgelu(const Tensor& dy, const Tensor& w2,
                                           const Tensor& z) {
  TORCH_CHECK(dy.is_contiguous() && w2.is_contiguous() && z.is_contiguous());
  // paddle layout: w2 [N, H] (in_features=N, out_features=H), dy [M, H]
  int64_t M = dy.size(0), H = dy.size(1), N = w2.size(0);
  TORCH_CHECK(w2.size(1) == H && z.size(0) == M && z.size(1) == N);
  auto dg = torch::empty({M, N}, dy.options());
  auto db = torch::empty({N}, dy.options());
  // col-major D'[N,M] = A'^T(w2 storage viewed H x N, lda=H) * B'(dy as
  // H x M): D'[n,m] = sum_h w2[n,h] dy[m,h]  ==  (dy @ w2^T)[m,n]
  lt_matmul(HIPBLAS_OP_T, HIPBLAS_OP_N, N, M, H, w2.const_data_ptr(), H,
            dy.const_data_ptr(), H, dg.mutable_data_ptr(), N,
            HIPBLASLT_EPILOGUE_DGELU_BGRAD, db.mutable_data_ptr(),
            const_cast<void*>(z.const_data_ptr()), N, dy);
  return {dg, db};
}

// heuristic-only support probe: how many algos exist for a bf16 GEMM of
// (m x n x k) with the given epilogue?  Lets Python pick the fused path at
// runtime instead of failing mid-autograd on builds without aux-epilogue
// Tensile kernels.
int64_t lt_epilogue_probe(int64_t m, int64_t n, int64_t k, int64_t epi) {
  hipblasLtMatmulDesc_t desc;
  LT_CHECK(hipblasLtMatmulDescCreate(&desc, HIPBLAS_COMPUTE_32F, HIP_R_32F));
  hipblasOperation_t opN = HIPBLAS_OP_N;
  LT_CHECK(hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_TRANSA,
                                           &opN, sizeof(opN)));
  LT_CHECK(hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_TRANSB,
                                           &opN, sizeof(opN)));
  hipblasLtEpilogue_t e = (hipblasLtEpilogue_t)epi;
  LT_CHECK(hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_EPILOGUE,
                                           &e, sizeof(e)));
  if (epi & 128) {  // aux-flavored epilogues
    int64_t ld = m;
    LT_CHECK(hipblasLtMatmulDescSetAttribute(
        desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_LD, &ld, sizeof(ld)));
    int32_t auxdt = HIP_R_16BF;
    LT_CHECK(hipblasLtMatmulDescSetAttribute(
        desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_DATA_TYPE, &auxdt,
        sizeof(auxdt)));
  }
  hipblasLtMatrixLayout_t la, lb, ld_;
  LT_CHECK(hipblasLtMatrixLayoutCreate(&la, HIP_R_16BF, m, k, m));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&lb, HIP_R_16BF, k, n, k));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&ld_, HIP_R_16BF, m, n, m));
  hipblasLtMatmulPreference_t pref;
  LT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
  size_t wsz = kWorkspace;
  LT_CHECK(hipblasLtMatmulPreferenceSetAttribute(
      pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &wsz, sizeof(wsz)));
  hipblasLtMatmulHeuristicResult_t res[8];
  int found = 0;
  hipblasStatus_t st = hipblasLtMatmulAlgoGetHeuristic(
      handle(), desc, la, lb, ld_, ld_, pref, 8, res, &found);
  hipblasLtMatmulPreferenceDestroy(pref);
  hipblasLtMatrixLayoutDestroy(la);
  hipblasLtMatrixLayoutDestroy(lb);
  hipblasLtMatrixLayoutDestroy(ld_);
  hipblasLtMatmulDescDestroy(desc);
  return st == HIPBLAS_STATUS_SUCCESS ? found : 0;
}

}  // namespace pa_lt
