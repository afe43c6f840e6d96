// hipBLASLt epilogue-fused GEMMs for the GPT MLP (fc1 path).
//
// Analysis (BENCHMARKS.md, TODO.md): the hand-written MFMA GEMM tops out
// ~855 TF/s vs hipBLASLt's 1373 at 4096^3, so fusing GELU into OUR tile
// loses end-to-end; hipBLASLt's epilogues keep library GEMM speed and
// delete the separate bias+gelu memory pass instead:
//   forward : D = GELU(bias + x @ W^T), pre-GELU saved to aux
//             (HIPBLASLT_EPILOGUE_GELU_AUX_BIAS)
//   backward: dPre = dgelu(aux) o (dy @ W2) and db in the SAME GEMM that
//             computes fc2's data grad (HIPBLASLT_EPILOGUE_DGELU_BGRAD)
// Exposed as ext.lt_fc1_forward / ext.lt_matmul_dgelu_bgrad; wired into
// the model only behind METIS_FC1_EPILOGUE=1 until GPU-validated.
//
// Layout note: torch tensors are row-major; hipBLASLt is column-major.
// A row-major [M, N] buffer is the column-major [N, M] matrix, so
// D_rm[M, N] = X_rm[M, K] @ W_rm[N, K]^T is computed as the col-major
// product D_cm[N, M] = op_T(W_cm[K, N]) * op_N(X_cm[K, M]); the bias
// vector (length N = rows of D_cm) broadcasts over columns = over M,
// which is exactly per-output-feature bias.

#include <torch/extension.h>

#include <hipblaslt/hipblaslt.h>

#include <c10/hip/HIPStream.h>

#include <mutex>
#include <stdexcept>
#include <string>
#include <unordered_map>

namespace {

#define LT_CHECK(expr)                                                        \
    do {                                                                      \
        hipblasStatus_t s_ = (expr);                                          \
        TORCH_CHECK(s_ == HIPBLAS_STATUS_SUCCESS, "hipblaslt error ",         \
                    (int)s_, " at ", #expr);                                  \
    } while (0)

hipblasLtHandle_t lt_handle() {
    static hipblasLtHandle_t handle = [] {
        hipblasLtHandle_t h;
        LT_CHECK(hipblasLtCreate(&h));
        return h;
    }();
    return handle;
}

constexpr size_t kWorkspaceBytes = 64ull << 20;

torch::Tensor workspace(const torch::TensorOptions& opt) {
    static torch::Tensor ws;
    if (!ws.defined() || ws.numel() < (long)kWorkspaceBytes)
        ws = torch::empty({(long)kWorkspaceBytes},
                          opt.dtype(torch::kUInt8));
    return ws;
}

struct LtPlan {
    hipblasLtMatmulDesc_t desc{};
    hipblasLtMatrixLayout_t a{}, b{}, d{};
    hipblasLtMatmulAlgo_t algo{};
    bool has_algo = false;
};

// One cached plan per (epilogue, M, N, K): the heuristic query costs ~100us
// and the training loop reuses a handful of shapes.
std::unordered_map<std::string, LtPlan>& plan_cache() {
    static std::unordered_map<std::string, LtPlan> cache;
    return cache;
}
std::mutex cache_mutex;

// transa=true : D_cm[N,M] = op_T(A_cm[K,N]) * B_cm[K,M]
//               (A is a torch row-major [N,K] weight — the y = x @ W^T form)
// transa=false: D_cm[N,M] = op_N(A_cm[N,K]) * B_cm[K,M]
//               (A is a torch row-major [K,N] weight — the y = x @ W form)
LtPlan& get_plan(hipblasLtEpilogue_t epi, bool transa, long M, long N, long K,
                 const void* bias, const void* aux, long aux_ld) {
    std::lock_guard<std::mutex> lock(cache_mutex);
    std::string key = std::to_string((int)epi) + (transa ? "t" : "n") + "_" +
                      std::to_string(M) + "_" + std::to_string(N) + "_" +
                      std::to_string(K);
    auto it = plan_cache().find(key);
    if (it != plan_cache().end()) return it->second;

    LtPlan plan;
    LT_CHECK(hipblasLtMatmulDescCreate(&plan.desc, HIPBLAS_COMPUTE_32F,
                                       HIP_R_32F));
    hipblasOperation_t opA = transa ? HIPBLAS_OP_T : HIPBLAS_OP_N;
    hipblasOperation_t opN = HIPBLAS_OP_N;
    LT_CHECK(hipblasLtMatmulDescSetAttribute(
        plan.desc, HIPBLASLT_MATMUL_DESC_TRANSA, &opA, sizeof(opA)));
    LT_CHECK(hipblasLtMatmulDescSetAttribute(
        plan.desc, HIPBLASLT_MATMUL_DESC_TRANSB, &opN, sizeof(opN)));
    LT_CHECK(hipblasLtMatmulDescSetAttribute(
        plan.desc, HIPBLASLT_MATMUL_DESC_EPILOGUE, &epi, sizeof(epi)));
    if (bias != nullptr)
        LT_CHECK(hipblasLtMatmulDescSetAttribute(
            plan.desc, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &bias,
            sizeof(bias)));
    if (aux != nullptr) {
        LT_CHECK(hipblasLtMatmulDescSetAttribute(
            plan.desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_POINTER, &aux,
            sizeof(aux)));
        LT_CHECK(hipblasLtMatmulDescSetAttribute(
            plan.desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_LD, &aux_ld,
            sizeof(aux_ld)));
    }

    // col-major views (see layout note)
    if (transa)
        LT_CHECK(hipblasLtMatrixLayoutCreate(&plan.a, HIP_R_16BF, K, N, K));
    else
        LT_CHECK(hipblasLtMatrixLayoutCreate(&plan.a, HIP_R_16BF, N, K, N));
    LT_CHECK(hipblasLtMatrixLayoutCreate(&plan.b, HIP_R_16BF, K, M, K));
    LT_CHECK(hipblasLtMatrixLayoutCreate(&plan.d, HIP_R_16BF, N, M, N));

    hipblasLtMatmulPreference_t pref;
    LT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
    uint64_t ws_bytes = kWorkspaceBytes;
    LT_CHECK(hipblasLtMatmulPreferenceSetAttribute(
        pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws_bytes,
        sizeof(ws_bytes)));
    hipblasLtMatmulHeuristicResult_t result{};
    int found = 0;
    LT_CHECK(hipblasLtMatmulAlgoGetHeuristic(
        lt_handle(), plan.desc, plan.a, plan.b, plan.d, plan.d, pref, 1,
        &result, &found));
    LT_CHECK(hipblasLtMatmulPreferenceDestroy(pref));
    TORCH_CHECK(found > 0, "hipblaslt: no algorithm for epilogue ", (int)epi,
                " at M=", M, " N=", N, " K=", K);
    plan.algo = result.algo;
    plan.has_algo = true;
    return plan_cache().emplace(key, plan).first->second;
}

void run_matmul(LtPlan& plan, const void* wA, const void* xB, void* outD,
                const void* bias, const void* aux, long aux_ld,
                hipblasLtEpilogue_t epi, long M, long N, long K,
                const torch::TensorOptions& opt) {
    // bias/aux pointers live in the cached desc: refresh them per call
    if (bias != nullptr)
        LT_CHECK(hipblasLtMatmulDescSetAttribute(
            plan.desc, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &bias,
            sizeof(bias)));
    if (aux != nullptr) {
        LT_CHECK(hipblasLtMatmulDescSetAttribute(
            plan.desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_POINTER, &aux,
            sizeof(aux)));
        LT_CHECK(hipblasLtMatmulDescSetAttribute(
            plan.desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_LD, &aux_ld,
            sizeof(aux_ld)));
    }
    float alpha = 1.0f, beta = 0.0f;
    auto ws = workspace(opt);
    auto stream = c10::hip::getCurrentHIPStream().stream();
    LT_CHECK(hipblasLtMatmul(
        lt_handle(), plan.desc, &alpha, wA, plan.a, xB, plan.b, &beta, outD,
        plan.d, outD, plan.d, plan.has_algo ? &plan.algo : nullptr,
        ws.data_ptr(), kWorkspaceBytes, stream));
}

}  // namespace

// forward: (y, pre_gelu) = GELU(bias + x @ w^T), both [M, N] bf16
std::vector<torch::Tensor> lt_fc1_forward(torch::Tensor x, torch::Tensor w,
                                          torch::Tensor bias) {
    TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16);
    TORCH_CHECK(x.dim() == 2 && w.dim() == 2 && x.size(1) == w.size(1),
                "x [M,K], w [N,K]");
    auto xc = x.contiguous(), wc = w.contiguous(), bc = bias.contiguous();
    const long M = x.size(0), K = x.size(1), N = w.size(0);
    auto y = torch::empty({M, N}, x.options());
    auto aux = torch::empty({M, N}, x.options());

    auto& plan = get_plan(HIPBLASLT_EPILOGUE_GELU_AUX_BIAS, true, M, N, K,
                          bc.data_ptr(), aux.data_ptr(), N);
    run_matmul(plan, wc.data_ptr(), xc.data_ptr(), y.data_ptr(),
               bc.data_ptr(), aux.data_ptr(), N,
               HIPBLASLT_EPILOGUE_GELU_AUX_BIAS, M, N, K, x.options());
    return {y, aux};
}

// backward: dpre = dgelu(pre_gelu) o (dy @ w2), dbias1 = colsum(dpre)
// (the fc2 data-grad GEMM with the dGELU + bias-grad epilogue)
std::vector<torch::Tensor> lt_matmul_dgelu_bgrad(torch::Tensor dy,
                                                 torch::Tensor w2,
                                                 torch::Tensor pre_gelu) {
    TORCH_CHECK(dy.is_cuda() && dy.dtype() == torch::kBFloat16);
    TORCH_CHECK(dy.dim() == 2 && w2.dim() == 2 && dy.size(1) == w2.size(0),
                "dy [M,H], w2 [H,N]");
    auto dyc = dy.contiguous(), w2c = w2.contiguous();
    auto auxc = pre_gelu.contiguous();
    const long M = dy.size(0), H = dy.size(1), N = w2.size(1);
    TORCH_CHECK(pre_gelu.size(0) == M && pre_gelu.size(1) == N);
    auto dpre = torch::empty({M, N}, dy.options());
    auto dbias = torch::empty({N}, dy.options());

    // col-major: dpre_cm[N, M] = op_N(w2_cm[N, H]) * op_N(dy_cm[H, M])
    // (w2 row-major [H, N] IS the col-major [N, H]); BGRAD reduces over
    // columns (= M) into a length-N (= ffn features) vector
    auto& plan = get_plan(HIPBLASLT_EPILOGUE_DGELU_BGRAD, false, M, N, H,
                          dbias.data_ptr(), auxc.data_ptr(), N);
    run_matmul(plan, w2c.data_ptr(), dyc.data_ptr(), dpre.data_ptr(),
               dbias.data_ptr(), auxc.data_ptr(), N,
               HIPBLASLT_EPILOGUE_DGELU_BGRAD, M, N, H, dy.options());
    return {dpre, dbias};
}

// fp8 (OCP e4m3) GEMM: D_bf16 = scale_a * scale_b * (x_fp8 @ w_fp8^T).
// MI355X's fp8 MFMA rate is 2x bf16 (~5 PFLOP/s dense), so running the
// big projection/FFN GEMMs in fp8 with per-tensor scales is the largest
// single perf lever left after kernel fusion; exposed for the round-2
// validated rollout (opt-in wiring, bf16 backward).
torch::Tensor lt_fp8_matmul(torch::Tensor x, torch::Tensor w,
                            torch::Tensor scale_x, torch::Tensor scale_w) {
    TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kFloat8_e4m3fn);
    TORCH_CHECK(w.dtype() == torch::kFloat8_e4m3fn);
    TORCH_CHECK(x.dim() == 2 && w.dim() == 2 && x.size(1) == w.size(1));
    TORCH_CHECK(scale_x.dtype() == torch::kFloat32 && scale_x.numel() == 1);
    auto xc = x.contiguous(), wc = w.contiguous();
    const long M = x.size(0), K = x.size(1), N = w.size(0);
    auto y = torch::empty({M, N},
                          x.options().dtype(torch::kBFloat16));

    std::lock_guard<std::mutex> lock(cache_mutex);
    std::string key = "fp8_" + std::to_string(M) + "_" + std::to_string(N) +
                      "_" + std::to_string(K);
    auto it = plan_cache().find(key);
    if (it == plan_cache().end()) {
        LtPlan plan;
        LT_CHECK(hipblasLtMatmulDescCreate(&plan.desc, HIPBLAS_COMPUTE_32F,
                                           HIP_R_32F));
        hipblasOperation_t opT = HIPBLAS_OP_T, opN = HIPBLAS_OP_N;
        LT_CHECK(hipblasLtMatmulDescSetAttribute(
            plan.desc, HIPBLASLT_MATMUL_DESC_TRANSA, &opT, sizeof(opT)));
        LT_CHECK(hipblasLtMatmulDescSetAttribute(
            plan.desc, HIPBLASLT_MATMUL_DESC_TRANSB, &opN, sizeof(opN)));
        LT_CHECK(hipblasLtMatrixLayoutCreate(&plan.a, HIP_R_8F_E4M3, K, N, K));
        LT_CHECK(hipblasLtMatrixLayoutCreate(&plan.b, HIP_R_8F_E4M3, K, M, K));
        LT_CHECK(hipblasLtMatrixLayoutCreate(&plan.d, HIP_R_16BF, N, M, N));
        hipblasLtMatmulPreference_t pref;
        LT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
        uint64_t ws_bytes = kWorkspaceBytes;
        LT_CHECK(hipblasLtMatmulPreferenceSetAttribute(
            pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws_bytes,
            sizeof(ws_bytes)));
        hipblasLtMatmulHeuristicResult_t result{};
        int found = 0;
        LT_CHECK(hipblasLtMatmulAlgoGetHeuristic(
            lt_handle(), plan.desc, plan.a, plan.b, plan.d, plan.d, pref, 1,
            &result, &found));
        LT_CHECK(hipblasLtMatmulPreferenceDestroy(pref));
        TORCH_CHECK(found > 0, "hipblaslt: no fp8 algorithm at M=", M,
                    " N=", N, " K=", K);
        plan.algo = result.algo;
        plan.has_algo = true;
        it = plan_cache().emplace(key, plan).first;
    }
    LtPlan& plan = it->second;
    // per-tensor scales: A = w (scale_w), B = x (scale_x)
    const void* sa = scale_w.data_ptr();
    const void* sb = scale_x.data_ptr();
    LT_CHECK(hipblasLtMatmulDescSetAttribute(
        plan.desc, HIPBLASLT_MATMUL_DESC_A_SCALE_POINTER, &sa, sizeof(sa)));
    LT_CHECK(hipblasLtMatmulDescSetAttribute(
        plan.desc, HIPBLASLT_MATMUL_DESC_B_SCALE_POINTER, &sb, sizeof(sb)));
    float alpha = 1.0f, beta = 0.0f;
    auto ws = workspace(y.options());
    auto stream = c10::hip::getCurrentHIPStream().stream();
    LT_CHECK(hipblasLtMatmul(
        lt_handle(), plan.desc, &alpha, wc.data_ptr(), plan.a, xc.data_ptr(),
        plan.b, &beta, y.data_ptr(), plan.d, y.data_ptr(), plan.d,
        &plan.algo, ws.data_ptr(), kWorkspaceBytes, stream));
    return y;
}
