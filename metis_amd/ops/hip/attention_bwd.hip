// Flash attention backward for gfx950 (causal, GQA) — bf16 I/O, fp32
// accumulation, MFMA 16x16x32, LSE-based recompute.
//
// Split-kernel design (deterministic, no atomics):
//   * dQ kernel: each wave owns 16 q rows, loops kv tiles <= its rows:
//       P  = exp(scale * Q K^T - lse)            (recompute, natural A/B)
//       dP = dO V^T                               (both natural layouts)
//       dS = scale * P o (dP - delta)
//       dQ += dS @ K      (dS via per-wave LDS relayout; K^T staged in LDS)
//   * dKV kernel: each wave owns 16 kv rows, loops q tiles >= its rows:
//       P^T  = exp(scale * K Q^T - lse[col])
//       dP^T = V dO^T
//       dV  += P^T  @ dO  (P^T via LDS relayout; dO^T staged in LDS)
//       dK  += dS^T @ Q   (dS^T via LDS relayout; Q^T staged in LDS)
//     GQA: the kernel runs per q head into per-head dK/dV buffers; the
//     wrapper sums each KV group (deterministic, no atomics).
//   * delta = rowsum(dO o O) is computed by the wrapper in torch.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int THREADS = 256;
constexpr int RBLK = 64;     // rows (q or kv) per block, 16 per wave
constexpr int CTILE = 32;    // opposing-side tile width
constexpr int DMAX = 128;

// load one natural A/B fragment: 8 contiguous bf16 of `row`, chunk c
template <int D>
__device__ __forceinline__ bf16x8 frag8(const bf16* base, long row, int d0) {
    if (d0 < D)
        return *reinterpret_cast<const bf16x8*>(base + row * D + d0);
    bf16x8 z;
    #pragma unroll
    for (int i = 0; i < 8; ++i) z[i] = 0;
    return z;
}

template <int D>
__global__ __launch_bounds__(THREADS, 2) void attn_bwd_dq_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ V, const bf16* __restrict__ dO,
    const float* __restrict__ LSE, const float* __restrict__ Delta,
    bf16* __restrict__ dQ,
    int B, int H, int Hkv, int S, float scale) {
    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
    const int col16 = lane & 15;
    const int k8 = lane >> 4;

    const int qtile = blockIdx.x % (S / RBLK);
    const int head = (blockIdx.x / (S / RBLK)) % H;
    const int batch = blockIdx.x / (S / RBLK) / H;
    const int kv_head = head / (H / Hkv);
    const long q_base = (((long)batch * H + head) * S) * D;
    const long kv_base = (((long)batch * Hkv + kv_head) * S) * D;
    const long row_base = ((long)batch * H + head) * S;

    const int q0 = qtile * RBLK + wave * 16;
    constexpr int dchunks = (D + 31) / 32;
    constexpr int djtiles = D / 16;

    extern __shared__ __attribute__((aligned(16))) char smem[];
    short* Kt = reinterpret_cast<short*>(smem);                     // [D][32]
    short* Sw = reinterpret_cast<short*>(smem + DMAX * CTILE * 2)
                + wave * 16 * CTILE;                                 // [16][32]

    // per-lane row state (rows k8*4 + r)
    float lse_r[4], delta_r[4];
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
        const int qrow = q0 + k8 * 4 + r;
        lse_r[r] = LSE[row_base + qrow];
        delta_r[r] = Delta[row_base + qrow];
    }

    // Q and dO fragments for this wave's rows (A layout, m = col16)
    bf16x8 q_frag[dchunks], do_frag[dchunks];
    #pragma unroll
    for (int c = 0; c < dchunks; ++c) {
        q_frag[c] = frag8<D>(Q + q_base, q0 + col16, c * 32 + k8 * 8);
        do_frag[c] = frag8<D>(dO + q_base, q0 + col16, c * 32 + k8 * 8);
    }

    floatx4 dq_acc[djtiles];
    #pragma unroll
    for (int jd = 0; jd < djtiles; ++jd) dq_acc[jd] = floatx4{0.f, 0.f, 0.f, 0.f};

    const int kv_end = qtile * RBLK + RBLK;
    for (int kv0 = 0; kv0 < kv_end; kv0 += CTILE) {
        // stage K^T tile into LDS (block-wide); lanes walk KV ROWS so the
        // scalar transpose writes hit contiguous addresses across the wave
        // (d-major mapping = 16-way bank conflict, see attention.hip)
        {
            constexpr int chunks = CTILE * D / 8;
            for (int c = threadIdx.x; c < chunks; c += THREADS) {
                const int row = c % CTILE;
                const int d0 = (c / CTILE) * 8;
                bf16x8 kv8 = *reinterpret_cast<const bf16x8*>(
                    K + kv_base + (long)(kv0 + row) * D + d0);
                #pragma unroll
                for (int i = 0; i < 8; ++i)
                    Kt[(d0 + i) * CTILE + row] = kv8[i];
            }
        }
        __syncthreads();
        // inactive waves skip compute but keep barriers uniform
        const bool active = kv0 <= q0 + 15;

        if (active) {
        // S and dP for two 16-col subtiles
        floatx4 s_acc[2], dp_acc[2];
        #pragma unroll
        for (int j = 0; j < 2; ++j) {
            s_acc[j] = floatx4{0.f, 0.f, 0.f, 0.f};
            dp_acc[j] = floatx4{0.f, 0.f, 0.f, 0.f};
            const long kvrow = kv0 + j * 16 + col16;
            #pragma unroll
            for (int c = 0; c < dchunks; ++c) {
                const int d0 = c * 32 + k8 * 8;
                bf16x8 k_frag = frag8<D>(K + kv_base, kvrow, d0);
                s_acc[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    q_frag[c], k_frag, s_acc[j], 0, 0, 0);
                bf16x8 v_frag = frag8<D>(V + kv_base, kvrow, d0);
                dp_acc[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    do_frag[c], v_frag, dp_acc[j], 0, 0, 0);
            }
        }

        // dS = scale * P o (dP - delta); P = exp(scale*s - lse); mask col>row
        #pragma unroll
        for (int j = 0; j < 2; ++j)
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int qrow = q0 + k8 * 4 + r;
                const int kvcol = kv0 + j * 16 + col16;
                float p = (kvcol > qrow)
                              ? 0.f
                              : __expf(s_acc[j][r] * scale - lse_r[r]);
                float ds = scale * p * (dp_acc[j][r] - delta_r[r]);
                Sw[(k8 * 4 + r) * CTILE + j * 16 + col16] = float_to_bf16_bits(ds);
            }
        }  // active

        // barrier orders the scalar dS stores against the vector re-read
        // (different pointer types: TBAA would otherwise allow hoisting)
        __syncthreads();

        if (active) {
        // dQ += dS @ K : A = dS (LDS relayout), B = K^T from LDS
        bf16x8 ds_frag = *reinterpret_cast<const bf16x8*>(
            Sw + col16 * CTILE + k8 * 8);
        #pragma unroll
        for (int jd = 0; jd < djtiles; ++jd) {
            bf16x8 kt_frag = *reinterpret_cast<const bf16x8*>(
                Kt + (jd * 16 + col16) * CTILE + k8 * 8);
            dq_acc[jd] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                ds_frag, kt_frag, dq_acc[jd], 0, 0, 0);
        }
        }  // active
        __syncthreads();
    }

    #pragma unroll
    for (int r = 0; r < 4; ++r) {
        const long qrow = q0 + k8 * 4 + r;
        #pragma unroll
        for (int jd = 0; jd < djtiles; ++jd)
            dQ[q_base + qrow * D + jd * 16 + col16] =
                __float2bfloat16(dq_acc[jd][r]);
    }
}

template <int D>
__global__ __launch_bounds__(THREADS, 2) void attn_bwd_dkv_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ V, const bf16* __restrict__ dO,
    const float* __restrict__ LSE, const float* __restrict__ Delta,
    bf16* __restrict__ dK,   // [B, H, S, D] per q-head (wrapper reduces GQA)
    bf16* __restrict__ dV,
    int B, int H, int Hkv, int S, float scale) {
    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
    const int col16 = lane & 15;
    const int k8 = lane >> 4;

    const int kvtile = blockIdx.x % (S / RBLK);
    const int head = (blockIdx.x / (S / RBLK)) % H;
    const int batch = blockIdx.x / (S / RBLK) / H;
    const int kv_head = head / (H / Hkv);
    const long q_base = (((long)batch * H + head) * S) * D;
    const long kv_base = (((long)batch * Hkv + kv_head) * S) * D;
    const long out_base = (((long)batch * H + head) * S) * D;
    const long row_base = ((long)batch * H + head) * S;

    const int kv0 = kvtile * RBLK + wave * 16;   // wave's first kv row
    constexpr int dchunks = (D + 31) / 32;
    constexpr int djtiles = D / 16;

    extern __shared__ __attribute__((aligned(16))) char smem[];
    short* Qt = reinterpret_cast<short*>(smem);                      // [D][32]
    short* dOt = reinterpret_cast<short*>(smem + DMAX * CTILE * 2);   // [D][32]
    short* Sw = reinterpret_cast<short*>(smem + 2 * DMAX * CTILE * 2)
                + wave * 16 * CTILE;                                  // [16][32]
    short* Pw = reinterpret_cast<short*>(smem + 2 * DMAX * CTILE * 2
                + 4 * 16 * CTILE * 2) + wave * 16 * CTILE;            // [16][32]

    // K and V fragments for this wave's rows (A layout, m = col16)
    bf16x8 k_frag[dchunks], v_frag[dchunks];
    #pragma unroll
    for (int c = 0; c < dchunks; ++c) {
        k_frag[c] = frag8<D>(K + kv_base, kv0 + col16, c * 32 + k8 * 8);
        v_frag[c] = frag8<D>(V + kv_base, kv0 + col16, c * 32 + k8 * 8);
    }

    floatx4 dk_acc[djtiles], dv_acc[djtiles];
    #pragma unroll
    for (int jd = 0; jd < djtiles; ++jd) {
        dk_acc[jd] = floatx4{0.f, 0.f, 0.f, 0.f};
        dv_acc[jd] = floatx4{0.f, 0.f, 0.f, 0.f};
    }

    // causal: only q tiles overlapping [block kv start, S)
    const int q_start = (kvtile * RBLK) / CTILE * CTILE;
    for (int q0 = q_start; q0 < S; q0 += CTILE) {
        // stage Q^T and dO^T tiles (block-wide, kv-row-major lane walk —
        // the d-major mapping is a 16-way bank conflict on the writes)
        {
            constexpr int chunks = CTILE * D / 8;
            for (int c = threadIdx.x; c < chunks; c += THREADS) {
                const int row = c % CTILE;
                const int d0 = (c / CTILE) * 8;
                bf16x8 qv = *reinterpret_cast<const bf16x8*>(
                    Q + q_base + (long)(q0 + row) * D + d0);
                bf16x8 dov = *reinterpret_cast<const bf16x8*>(
                    dO + q_base + (long)(q0 + row) * D + d0);
                #pragma unroll
                for (int i = 0; i < 8; ++i) {
                    Qt[(d0 + i) * CTILE + row] = qv[i];
                    dOt[(d0 + i) * CTILE + row] = dov[i];
                }
            }
        }
        __syncthreads();
        // entire q tile above the diagonal -> inactive (barriers uniform)
        const bool active = q0 + CTILE - 1 >= kv0;

        float pt[2][4];
        if (active) {
        // S^T = K Q^T and dP^T = V dO^T for two 16-col (q) subtiles
        floatx4 st_acc[2], dpt_acc[2];
        #pragma unroll
        for (int j = 0; j < 2; ++j) {
            st_acc[j] = floatx4{0.f, 0.f, 0.f, 0.f};
            dpt_acc[j] = floatx4{0.f, 0.f, 0.f, 0.f};
            const long qrow = q0 + j * 16 + col16;
            #pragma unroll
            for (int c = 0; c < dchunks; ++c) {
                const int d0 = c * 32 + k8 * 8;
                bf16x8 qf = frag8<D>(Q + q_base, qrow, d0);
                st_acc[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    k_frag[c], qf, st_acc[j], 0, 0, 0);
                bf16x8 dof = frag8<D>(dO + q_base, qrow, d0);
                dpt_acc[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    v_frag[c], dof, dpt_acc[j], 0, 0, 0);
            }
        }

        // P^T and dS^T (C layout: row = kv = k8*4+r, col = q)
        #pragma unroll
        for (int j = 0; j < 2; ++j)
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int kvrow = kv0 + k8 * 4 + r;
                const int qcol = q0 + j * 16 + col16;
                float p = (qcol < kvrow)
                              ? 0.f
                              : __expf(st_acc[j][r] * scale
                                       - LSE[row_base + qcol]);
                pt[j][r] = p;
                float ds = scale * p
                           * (dpt_acc[j][r] - Delta[row_base + qcol]);
                Sw[(k8 * 4 + r) * CTILE + j * 16 + col16] = float_to_bf16_bits(ds);
                Pw[(k8 * 4 + r) * CTILE + j * 16 + col16] =
                    float_to_bf16_bits(pt[j][r]);
            }
        }  // active

        // barrier orders the scalar dS^T / P^T stores against the vector
        // re-reads (TBAA) and keeps control flow uniform
        __syncthreads();

        if (active) {
        // dK += dS^T @ Q : A = dS^T (LDS relayout), B = Q^T (LDS)
        {
            bf16x8 dst_frag = *reinterpret_cast<const bf16x8*>(
                Sw + col16 * CTILE + k8 * 8);
            #pragma unroll
            for (int jd = 0; jd < djtiles; ++jd) {
                bf16x8 qt_frag = *reinterpret_cast<const bf16x8*>(
                    Qt + (jd * 16 + col16) * CTILE + k8 * 8);
                dk_acc[jd] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    dst_frag, qt_frag, dk_acc[jd], 0, 0, 0);
            }
        }

        // dV += P^T @ dO : A = P^T (LDS relayout), B = dO^T (LDS)
        {
            bf16x8 pt_frag = *reinterpret_cast<const bf16x8*>(
                Pw + col16 * CTILE + k8 * 8);
            #pragma unroll
            for (int jd = 0; jd < djtiles; ++jd) {
                bf16x8 dot_frag = *reinterpret_cast<const bf16x8*>(
                    dOt + (jd * 16 + col16) * CTILE + k8 * 8);
                dv_acc[jd] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    pt_frag, dot_frag, dv_acc[jd], 0, 0, 0);
            }
        }
        }  // active
        __syncthreads();
    }

    #pragma unroll
    for (int r = 0; r < 4; ++r) {
        const long kvrow = kv0 + k8 * 4 + r;
        #pragma unroll
        for (int jd = 0; jd < djtiles; ++jd) {
            dK[out_base + kvrow * D + jd * 16 + col16] =
                __float2bfloat16(dk_acc[jd][r]);
            dV[out_base + kvrow * D + jd * 16 + col16] =
                __float2bfloat16(dv_acc[jd][r]);
        }
    }
}

}  // namespace

std::vector<torch::Tensor> attn_bwd(
    torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor d_o, torch::Tensor lse, torch::Tensor delta,
    double scale) {
    TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16);
    const long B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
    const long Hkv = k.size(1);
    TORCH_CHECK(S % RBLK == 0 && D % 16 == 0 && D <= DMAX);
    auto qc = q.contiguous(), kc = k.contiguous(), vc = v.contiguous();
    auto doc = d_o.contiguous();
    auto lsec = lse.contiguous(), dc = delta.contiguous();

    auto dq = torch::empty_like(qc);
    // per q-head dK/dV; the wrapper reduces the GQA group
    auto dk = torch::empty({B, H, S, D}, q.options());
    auto dv = torch::empty({B, H, S, D}, q.options());

    auto stream = c10::hip::getCurrentHIPStream().stream();
    const int grid = (int)(B * H * (S / RBLK));
    const int lds_dq = DMAX * CTILE * 2 + 4 * 16 * CTILE * 2;
    const int lds_dkv = 2 * DMAX * CTILE * 2 + 2 * 4 * 16 * CTILE * 2;

    #define LAUNCH_BWD(DD)                                                    \
        hipLaunchKernelGGL(attn_bwd_dq_kernel<DD>, dim3(grid), dim3(THREADS), \
            lds_dq, stream,                                                   \
            reinterpret_cast<const bf16*>(qc.data_ptr()),                     \
            reinterpret_cast<const bf16*>(kc.data_ptr()),                     \
            reinterpret_cast<const bf16*>(vc.data_ptr()),                     \
            reinterpret_cast<const bf16*>(doc.data_ptr()),                    \
            lsec.data_ptr<float>(), dc.data_ptr<float>(),                     \
            reinterpret_cast<bf16*>(dq.data_ptr()),                           \
            (int)B, (int)H, (int)Hkv, (int)S, (float)scale);                  \
        hipLaunchKernelGGL(attn_bwd_dkv_kernel<DD>, dim3(grid),               \
            dim3(THREADS), lds_dkv, stream,                                   \
            reinterpret_cast<const bf16*>(qc.data_ptr()),                     \
            reinterpret_cast<const bf16*>(kc.data_ptr()),                     \
            reinterpret_cast<const bf16*>(vc.data_ptr()),                     \
            reinterpret_cast<const bf16*>(doc.data_ptr()),                    \
            lsec.data_ptr<float>(), dc.data_ptr<float>(),                     \
            reinterpret_cast<bf16*>(dk.data_ptr()),                           \
            reinterpret_cast<bf16*>(dv.data_ptr()),                           \
            (int)B, (int)H, (int)Hkv, (int)S, (float)scale)
    switch ((int)D) {
        case 64: LAUNCH_BWD(64); break;
        case 80: LAUNCH_BWD(80); break;
        case 96: LAUNCH_BWD(96); break;
        case 128: LAUNCH_BWD(128); break;
        default: TORCH_CHECK(false, "head dim must be 64/80/96/128, got ", D);
    }
    #undef LAUNCH_BWD
    HIP_CHECK_LAST();

    return {dq, dk, dv};
}
