// Fused transformer-MLP GEMMs via hipBLASLt epilogues (library GEMMs —
// the "plain GEMM" tier; the point-wise dGELU/bias-grad work that
// PyTorch runs as separate memory-bound kernels per MLP is folded into
// the GEMM epilogues where this hipBLASLt (1.2/gfx950) has algos —
// probed with scripts/probe_lt.cpp: GELU_AUX_BIAS and DGELU_BGRAD have
// NO algos; DGELU (NN) and BGRADB (NT wgrad) DO:
//   fwd:  z = x @ W1^T + b1        [BIAS]
//         h = gelu_tanh(z)         [one vectorized kernel here]
//         y = h @ W2^T + b2        [BIAS]
//   bwd:  dz = dGELU(dy @ W2, aux=z)            [DGELU]
//         dW2 = dy^T @ h, db2 = colsum(dy)      [BGRADB]
//         dW1 = dz^T @ x, db1 = colsum(dz)      [BGRADB]
//         dx = dz @ W1                          [plain]
// Replaces the GELU-backward kernel and both bias-grad reduce_kernels
// per MLP per step (GPT-2-XL profile: ~3% of step).
//
// All tensors are row-major torch bf16; hipBLASLt is column-major, so
// every call computes the transposed-view GEMM (the usual swap).
#include <hip/hip_runtime.h>
#include <hipblaslt/hipblaslt.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <cstdint>
#include <map>
#include <mutex>
#include <tuple>

#define CHECK_LT(x)                                                       \
  do {                                                                    \
    hipblasStatus_t st__ = (x);                                           \
    TORCH_CHECK(st__ == HIPBLAS_STATUS_SUCCESS, "hipBLASLt error ", st__, \
                " at ", __FILE__, ":", __LINE__);                         \
  } while (0)

namespace {

hipblasLtHandle_t lt_handle() {
  static hipblasLtHandle_t h = [] {
    hipblasLtHandle_t hh;
    CHECK_LT(hipblasLtCreate(&hh));
    return hh;
  }();
  return h;
}

constexpr size_t kWorkspace = 64u << 20;

void* workspace() {
  static void* ws = [] {
    void* p = nullptr;
    TORCH_CHECK(hipMalloc(&p, kWorkspace) == hipSuccess,
                "lt_mlp workspace alloc failed");
    return p;
  }();
  return ws;
}

struct PlanKey {
  int64_t m, n, k;
  int32_t epi;
  int32_t opA, opB;
  bool operator<(const PlanKey& o) const {
    return std::tie(m, n, k, epi, opA, opB) <
           std::tie(o.m, o.n, o.k, o.epi, o.opA, o.opB);
  }
};

struct Plan {
  hipblasLtMatmulDesc_t desc;
  hipblasLtMatrixLayout_t la, lb, lc;
  hipblasLtMatmulAlgo_t algo;
  bool has_algo = false;
};

std::map<PlanKey, Plan>& plan_cache() {
  static std::map<PlanKey, Plan> c;
  return c;
}
std::mutex g_mu;

// Column-major GEMM D[m,n] = opA(A) opB(B) in bf16 with fp32 compute.
// aux/bias/bgrad pointers are optional epilogue attachments.
void lt_gemm(int64_t m, int64_t n, int64_t k, hipblasOperation_t opA,
             hipblasOperation_t opB, const void* A, int64_t lda,
             const void* B, int64_t ldb, void* D, int64_t ldd,
             hipblasLtEpilogue_t epi, void* bias, void* aux,
             int64_t ld_aux, hipStream_t stream) {
  std::lock_guard<std::mutex> lk(g_mu);
  PlanKey key{m, n, k, (int32_t)epi, (int32_t)opA, (int32_t)opB};
  auto it = plan_cache().find(key);
  if (it == plan_cache().end()) {
    Plan p;
    CHECK_LT(hipblasLtMatmulDescCreate(&p.desc, HIPBLAS_COMPUTE_32F,
                                       HIP_R_32F));
    CHECK_LT(hipblasLtMatmulDescSetAttribute(
        p.desc, HIPBLASLT_MATMUL_DESC_TRANSA, &opA, sizeof(opA)));
    CHECK_LT(hipblasLtMatmulDescSetAttribute(
        p.desc, HIPBLASLT_MATMUL_DESC_TRANSB, &opB, sizeof(opB)));
    CHECK_LT(hipblasLtMatmulDescSetAttribute(
        p.desc, HIPBLASLT_MATMUL_DESC_EPILOGUE, &epi, sizeof(epi)));
    const int64_t ar = (opA == HIPBLAS_OP_N) ? m : k;
    const int64_t ac = (opA == HIPBLAS_OP_N) ? k : m;
    const int64_t br = (opB == HIPBLAS_OP_N) ? k : n;
    const int64_t bc = (opB == HIPBLAS_OP_N) ? n : k;
    CHECK_LT(hipblasLtMatrixLayoutCreate(&p.la, HIP_R_16BF, ar, ac,
                                         lda));
    CHECK_LT(hipblasLtMatrixLayoutCreate(&p.lb, HIP_R_16BF, br, bc,
                                         ldb));
    CHECK_LT(hipblasLtMatrixLayoutCreate(&p.lc, HIP_R_16BF, m, n, ldd));
    it = plan_cache().emplace(key, p).first;
  }
  Plan& p = it->second;
  // per-call epilogue pointers (not part of the cached key)
  if (bias) {
    CHECK_LT(hipblasLtMatmulDescSetAttribute(
        p.desc, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &bias,
        sizeof(bias)));
  }
  if (aux) {
    CHECK_LT(hipblasLtMatmulDescSetAttribute(
        p.desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_POINTER, &aux,
        sizeof(aux)));
    CHECK_LT(hipblasLtMatmulDescSetAttribute(
        p.desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_LD, &ld_aux,
        sizeof(ld_aux)));
  }
  const float alpha = 1.f, beta = 0.f;
  if (!p.has_algo) {
    // NOTE: a 16-candidate timing tuner was tried here and REVERTED:
    // several heuristic candidates on this hipBLASLt build return
    // wrong results for epilogue GEMMs (fast-but-broken kernels win
    // the timing race). heuristic[0] is correct; the path is
    // experimental anyway (see ops/lt_mlp.py verdict).
    hipblasLtMatmulPreference_t pref;
    CHECK_LT(hipblasLtMatmulPreferenceCreate(&pref));
    size_t ws = kWorkspace;
    CHECK_LT(hipblasLtMatmulPreferenceSetAttribute(
        pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws,
        sizeof(ws)));
    hipblasLtMatmulHeuristicResult_t res[1];
    int found = 0;
    CHECK_LT(hipblasLtMatmulAlgoGetHeuristic(
        lt_handle(), p.desc, p.la, p.lb, p.lc, p.lc, pref, 1, res,
        &found));
    hipblasLtMatmulPreferenceDestroy(pref);
    TORCH_CHECK(found > 0, "no hipBLASLt algo for m=", m, " n=", n,
                " k=", k, " epi=", (int)epi);
    p.algo = res[0].algo;
    p.has_algo = true;
  }
  CHECK_LT(hipblasLtMatmul(lt_handle(), p.desc, &alpha, A, p.la, B,
                           p.lb, &beta, D, p.lc, D, p.lc, &p.algo,
                           workspace(), kWorkspace, stream));
}

inline const void* dp(const torch::Tensor& t) { return t.data_ptr(); }
inline void* dp(torch::Tensor& t) { return t.data_ptr(); }

using bf16x8v = __attribute__((ext_vector_type(8))) __bf16;

// h = gelu_tanh(z), 8-wide bf16
__global__ __launch_bounds__(256) void gelu_fwd_kernel(
    const __hip_bfloat16* __restrict__ z, __hip_bfloat16* __restrict__ h,
    long n8) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n8) return;
  bf16x8v zv = *reinterpret_cast<const bf16x8v*>(z + i * 8);
  bf16x8v hv;
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    const float x = (float)zv[j];
    const float c = 0.7978845608028654f * (x + 0.044715f * x * x * x);
    hv[j] = (__bf16)(0.5f * x * (1.0f + tanhf(c)));
  }
  *reinterpret_cast<bf16x8v*>(h + i * 8) = hv;
}

}  // namespace

// fwd: returns {y, h, z}
std::vector<torch::Tensor> lt_mlp_fwd(torch::Tensor x,
                                      torch::Tensor W1,
                                      torch::Tensor b1,
                                      torch::Tensor W2,
                                      torch::Tensor b2) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kBFloat16 &&
              x.dim() == 2 && x.is_contiguous());
  const int64_t M = x.size(0), C = x.size(1), F = W1.size(0);
  TORCH_CHECK(W1.size(1) == C && W2.size(0) == C && W2.size(1) == F);
  TORCH_CHECK((M * F) % 8 == 0);
  auto h = torch::empty({M, F}, x.options());
  auto z = torch::empty({M, F}, x.options());
  auto y = torch::empty({M, C}, x.options());
  auto stream = at::hip::getCurrentHIPStream().stream();
  // z_cm[F,M] = W1_cm[C,F]^T @ x_cm[C,M] + b1
  lt_gemm(F, M, C, HIPBLAS_OP_T, HIPBLAS_OP_N, dp(W1), C, dp(x), C,
          dp(z), F, HIPBLASLT_EPILOGUE_BIAS, dp(b1), nullptr, 0,
          stream);
  const long n8 = (long)M * F / 8;
  hipLaunchKernelGGL(gelu_fwd_kernel, dim3((n8 + 255) / 256), dim3(256),
                     0, stream,
                     reinterpret_cast<const __hip_bfloat16*>(z.data_ptr()),
                     reinterpret_cast<__hip_bfloat16*>(h.data_ptr()),
                     n8);
  // y_cm[C,M] = W2_cm[F,C]^T @ h_cm[F,M]
  lt_gemm(C, M, F, HIPBLAS_OP_T, HIPBLAS_OP_N, dp(W2), F, dp(h), F,
          dp(y), C, HIPBLASLT_EPILOGUE_BIAS, dp(b2), nullptr, 0,
          stream);
  return {y, h, z};
}

// bwd: returns {dx, dW1, db1, dW2, db2}
std::vector<torch::Tensor> lt_mlp_bwd(torch::Tensor dy, torch::Tensor x,
                                      torch::Tensor W1,
                                      torch::Tensor W2, torch::Tensor h,
                                      torch::Tensor z) {
  const int64_t M = x.size(0), C = x.size(1), F = W1.size(0);
  dy = dy.contiguous();
  auto dz = torch::empty({M, F}, x.options());
  auto dx = torch::empty({M, C}, x.options());
  auto dW1 = torch::empty({F, C}, x.options());
  auto dW2 = torch::empty({C, F}, x.options());
  auto db1 = torch::empty({F}, x.options());
  auto db2 = torch::empty({C}, x.options());
  auto stream = at::hip::getCurrentHIPStream().stream();
  // dz_cm[F,M] = W2_cm[F,C] @ dy_cm[C,M], then dGELU with aux=z
  lt_gemm(F, M, C, HIPBLAS_OP_N, HIPBLAS_OP_N, dp(W2), F, dp(dy), C,
          dp(dz), F, HIPBLASLT_EPILOGUE_DGELU, nullptr, dp(z), F,
          stream);
  // dW2_rm[C,F] -> cm[F,C] = h_cm[F,M] @ dy_cm[C,M]^T; db2 from B=dy
  lt_gemm(F, C, M, HIPBLAS_OP_N, HIPBLAS_OP_T, dp(h), F, dp(dy), C,
          dp(dW2), F, HIPBLASLT_EPILOGUE_BGRADB, dp(db2), nullptr, 0,
          stream);
  // dW1_rm[F,C] -> cm[C,F] = x_cm[C,M] @ dz_cm[F,M]^T; db1 from B=dz
  lt_gemm(C, F, M, HIPBLAS_OP_N, HIPBLAS_OP_T, dp(x), C, dp(dz), F,
          dp(dW1), C, HIPBLASLT_EPILOGUE_BGRADB, dp(db1), nullptr, 0,
          stream);
  // dx_cm[C,M] = W1_cm[C,F] @ dz_cm[F,M]
  lt_gemm(C, M, F, HIPBLAS_OP_N, HIPBLAS_OP_N, dp(W1), C, dp(dz), F,
          dp(dx), C, HIPBLASLT_EPILOGUE_DEFAULT, nullptr, nullptr, 0,
          stream);
  return {dx, dW1, db1, dW2, db2};
}


// General Linear: y = x @ W^T + b with BIAS fwd epilogue; backward
// fuses db = colsum(dy) into the wgrad GEMM via BGRADB.
std::vector<torch::Tensor> lt_linear_fwd(torch::Tensor x,
                                         torch::Tensor W,
                                         torch::Tensor b) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kBFloat16 &&
              x.dim() == 2 && x.is_contiguous());
  const int64_t M = x.size(0), C = x.size(1), N = W.size(0);
  TORCH_CHECK(W.size(1) == C);
  auto y = torch::empty({M, N}, x.options());
  auto stream = at::hip::getCurrentHIPStream().stream();
  lt_gemm(N, M, C, HIPBLAS_OP_T, HIPBLAS_OP_N, dp(W), C, dp(x), C,
          dp(y), N, HIPBLASLT_EPILOGUE_BIAS, dp(b), nullptr, 0,
          stream);
  return {y};
}

std::vector<torch::Tensor> lt_linear_bwd(torch::Tensor dy,
                                         torch::Tensor x,
                                         torch::Tensor W) {
  const int64_t M = x.size(0), C = x.size(1), N = W.size(0);
  dy = dy.contiguous();
  auto dx = torch::empty({M, C}, x.options());
  auto dW = torch::empty({N, C}, x.options());
  auto db = torch::empty({N}, x.options());
  auto stream = at::hip::getCurrentHIPStream().stream();
  // dW_rm[N,C] -> cm[C,N] = x_cm[C,M] @ dy_cm[N,M]^T; db from B=dy
  lt_gemm(C, N, M, HIPBLAS_OP_N, HIPBLAS_OP_T, dp(x), C, dp(dy), N,
          dp(dW), C, HIPBLASLT_EPILOGUE_BGRADB, dp(db), nullptr, 0,
          stream);
  // dx_cm[C,M] = W_cm[C,N] @ dy_cm[N,M]
  lt_gemm(C, M, N, HIPBLAS_OP_N, HIPBLAS_OP_N, dp(W), C, dp(dy), N,
          dp(dx), C, HIPBLASLT_EPILOGUE_DEFAULT, nullptr, nullptr, 0,
          stream);
  return {dx, dW, db};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("lt_mlp_fwd", &lt_mlp_fwd,
        "fused MLP forward (GELU_AUX_BIAS + BIAS epilogues)");
  m.def("lt_linear_fwd", &lt_linear_fwd, "Linear fwd (BIAS epilogue)");
  m.def("lt_linear_bwd", &lt_linear_bwd,
        "Linear bwd (BGRADB-fused bias grad)");
  m.def("lt_mlp_bwd", &lt_mlp_bwd,
        "fused MLP backward (DGELU_BGRAD + BGRADB epilogues)");
}
