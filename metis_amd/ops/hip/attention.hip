// Flash attention forward (causal, GQA) for gfx950 — bf16 I/O, fp32
// online softmax, MFMA (v_mfma_f32_16x16x32_bf16) for QK^T and PV.
//
// v1 structure (correctness-first; the optimization ladder of
// cdna_hip_programming.md Appendix B — K-LDS XOR swizzle, tr_b16 V reads,
// async staging — lands on top of this):
//   * block = 256 threads = 4 waves, each wave owns a 16-row Q strip
//     (64 q rows per block), KV tiles of 32 columns;
//   * Q fragments preloaded to registers (one bf16x8 per 32-wide D chunk);
//   * QK^T B-fragments read straight from K (row-major [S, D] gives each
//     lane 8 contiguous elements of one K row; the 8 KB tile stays L2-hot);
//   * V is staged transposed into LDS once per block per tile, so PV's
//     B-fragment (fixed d, 8 contiguous kv) is one ds_read_b128;
//   * P redistributes C-layout -> A-layout through a per-wave LDS tile;
//   * per-row running (m, l) in registers; O accumulates in C-layout
//     fragments, rescaled on max growth, normalized once at the end.
//
// Saves the row LSE (m + log l) for the backward pass.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int THREADS = 256;
constexpr int QBLK = 64;      // q rows per block (16 per wave)
constexpr int KVBLK = 32;     // kv columns per tile
constexpr int DMAX = 128;

template <int D>
__global__ __launch_bounds__(THREADS, 2) void attn_fwd_kernel(
    const bf16* __restrict__ Q,   // [B, H, S, D]
    const bf16* __restrict__ K,   // [B, Hkv, S, D]
    const bf16* __restrict__ V,   // [B, Hkv, S, D]
    bf16* __restrict__ O,         // [B, H, S, D]
    float* __restrict__ LSE,      // [B, H, S]
    int B, int H, int Hkv, int S,
    float scale) {
    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
    const int col16 = lane & 15;       // 16-lane group position
    const int k8 = lane >> 4;          // which 8-element K chunk (0..3)

    const int qtile = blockIdx.x % (S / QBLK);
    const int head = (blockIdx.x / (S / QBLK)) % H;
    const int batch = blockIdx.x / (S / QBLK) / H;
    const int kv_head = head / (H / Hkv);

    const long q_base = (((long)batch * H + head) * S) * D;
    const long kv_base = (((long)batch * Hkv + kv_head) * S) * D;

    const int q0 = qtile * QBLK + wave * 16;   // this wave's first q row
    constexpr int dchunks = (D + 31) / 32;     // 32-wide D chunks
    constexpr int djtiles = D / 16;            // 16-wide output column tiles

    extern __shared__ __attribute__((aligned(16))) char smem[];
    // LDS tiles hold raw bf16 BITS as shorts: assigning a short into a
    // __hip_bfloat16 struct would numerically convert (16256 -> 16256.0f),
    // not reinterpret — the classic trap this kernel once had.
    short* Vt = reinterpret_cast<short*>(smem);             // [D][KVBLK]
    short* Pw = reinterpret_cast<short*>(smem + DMAX * KVBLK * 2)
                + wave * 16 * KVBLK;                         // per-wave [16][KVBLK]

    // ---- preload this wave's Q fragments (A-layout per 32-chunk) --------
    bf16x8 q_frag[4];
    {
        const int qrow = q0 + col16;
        #pragma unroll
        for (int c = 0; c < dchunks; ++c) {
            const int d0 = c * 32 + k8 * 8;
            if (d0 < D) {
                q_frag[c] = *reinterpret_cast<const bf16x8*>(
                    Q + q_base + (long)qrow * D + d0);
            } else {
                #pragma unroll
                for (int i = 0; i < 8; ++i) q_frag[c][i] = 0;
            }
        }
    }

    // ---- running state: rows (k8*4 + r) of the C layout -----------------
    float m_run[4], l_run[4];
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
        m_run[r] = -1e30f;
        l_run[r] = 0.f;
    }
    floatx4 o_acc[djtiles];
    #pragma unroll
    for (int jd = 0; jd < djtiles; ++jd) o_acc[jd] = floatx4{0.f, 0.f, 0.f, 0.f};

    const int kv_end = qtile * QBLK + QBLK;    // causal bound for the block
    for (int kv0 = 0; kv0 < kv_end; kv0 += KVBLK) {
        // ---- stage V^T into LDS (whole block cooperates) ----------------
        {
            // 256 threads x bf16x8: covers KVBLK * D / 8 chunks
            constexpr int chunks = KVBLK * D / 8;
            for (int c = threadIdx.x; c < chunks; c += THREADS) {
                const int row = c / (D / 8);        // kv row in tile
                const int d0 = (c % (D / 8)) * 8;
                bf16x8 v = *reinterpret_cast<const bf16x8*>(
                    V + kv_base + (long)(kv0 + row) * D + d0);
                #pragma unroll
                for (int i = 0; i < 8; ++i)
                    Vt[(d0 + i) * KVBLK + row] = v[i];
            }
        }
        __syncthreads();

        // Causal: a tile starting past this wave's last row (q0+15) is
        // fully masked — with m_run still at -1e30 its exp(s - m) would be
        // exp(0) = 1. Inactive waves skip the compute but keep every
        // barrier (uniform control flow).
        const bool active = kv0 <= q0 + 15;

        // ---- S = scale * Q @ K^T for two 16-col subtiles ----------------
        float p[2][4];   // exp(S - m) per (j, r)
        if (active) {
        floatx4 s_acc[2];
        #pragma unroll
        for (int j = 0; j < 2; ++j) {
            s_acc[j] = floatx4{0.f, 0.f, 0.f, 0.f};
            const int kvrow = kv0 + j * 16 + col16;
            #pragma unroll
            for (int c = 0; c < dchunks; ++c) {
                bf16x8 k_frag;
                const int d0 = c * 32 + k8 * 8;
                if (d0 < D) {
                    k_frag = *reinterpret_cast<const bf16x8*>(
                        K + kv_base + (long)kvrow * D + d0);
                } else {
                    #pragma unroll
                    for (int i = 0; i < 8; ++i) k_frag[i] = 0;
                }
                s_acc[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    q_frag[c], k_frag, s_acc[j], 0, 0, 0);
            }
        }

        // ---- causal mask + online softmax -------------------------------
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int qrow = q0 + k8 * 4 + r;
            float tile_max = -1e30f;
            #pragma unroll
            for (int j = 0; j < 2; ++j) {
                const int kvcol = kv0 + j * 16 + col16;
                float s = s_acc[j][r] * scale;
                if (kvcol > qrow) s = -1e30f;
                s_acc[j][r] = s;
                tile_max = fmaxf(tile_max, s);
            }
            // row max across the 16-lane group
            #pragma unroll
            for (int off = 8; off > 0; off >>= 1)
                tile_max = fmaxf(tile_max, __shfl_xor(tile_max, off, 16));

            const float m_new = fmaxf(m_run[r], tile_max);
            const float alpha = __expf(m_run[r] - m_new);
            float row_sum = 0.f;
            #pragma unroll
            for (int j = 0; j < 2; ++j) {
                p[j][r] = __expf(s_acc[j][r] - m_new);
                row_sum += p[j][r];
            }
            #pragma unroll
            for (int off = 8; off > 0; off >>= 1)
                row_sum += __shfl_xor(row_sum, off, 16);

            l_run[r] = l_run[r] * alpha + row_sum;
            m_run[r] = m_new;
            // rescale O rows r
            #pragma unroll
            for (int jd = 0; jd < djtiles; ++jd)
                o_acc[jd][r] *= alpha;
        }

        // ---- P (C layout) -> LDS ----------------------------------------
        #pragma unroll
        for (int j = 0; j < 2; ++j)
            #pragma unroll
            for (int r = 0; r < 4; ++r)
                Pw[(k8 * 4 + r) * KVBLK + j * 16 + col16] =
                    float_to_bf16_bits(p[j][r]);
        }  // active

        // The barrier both keeps control flow uniform and orders the
        // scalar P stores against the vector re-read below: the two go
        // through different pointer types, and without a barrier the
        // compiler may hoist the ds_read above the ds_writes (TBAA).
        __syncthreads();

        if (active) {
        // ---- O += P @ V (P re-read in A layout) -------------------------
        bf16x8 p_frag = *reinterpret_cast<const bf16x8*>(
            Pw + col16 * KVBLK + k8 * 8);
        #pragma unroll
        for (int jd = 0; jd < djtiles; ++jd) {
            bf16x8 v_frag = *reinterpret_cast<const bf16x8*>(
                Vt + (jd * 16 + col16) * KVBLK + k8 * 8);
            o_acc[jd] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                p_frag, v_frag, o_acc[jd], 0, 0, 0);
        }
        }  // active
        __syncthreads();   // Vt will be overwritten next tile
    }

    // ---- normalize + store ----------------------------------------------
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
        const int qrow = q0 + k8 * 4 + r;
        const float inv_l = 1.f / l_run[r];
        #pragma unroll
        for (int jd = 0; jd < djtiles; ++jd)
            O[q_base + (long)qrow * D + jd * 16 + col16] =
                __float2bfloat16(o_acc[jd][r] * inv_l);
        if (col16 == 0)
            LSE[((long)batch * H + head) * S + qrow] = m_run[r] + __logf(l_run[r]);
    }
}

}  // namespace

std::vector<torch::Tensor> attn_fwd(
    torch::Tensor q, torch::Tensor k, torch::Tensor v, double scale) {
    TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16, "q must be CUDA bf16");
    TORCH_CHECK(q.dim() == 4, "q must be [B, H, S, D]");
    const long B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
    const long Hkv = k.size(1);
    TORCH_CHECK(k.size(2) == S && v.size(2) == S, "kv length mismatch");
    TORCH_CHECK(D % 16 == 0 && D <= DMAX, "head dim must be /16 and <= 128");
    TORCH_CHECK(S % QBLK == 0, "sequence length must be a multiple of 64");
    TORCH_CHECK(H % Hkv == 0, "GQA requires H % Hkv == 0");
    auto qc = q.contiguous(), kc = k.contiguous(), vc = v.contiguous();

    auto o = torch::empty_like(qc);
    auto lse = torch::empty({B, H, S}, q.options().dtype(torch::kFloat32));

    const int grid = (int)(B * H * (S / QBLK));
    const int lds = DMAX * KVBLK * 2 + 4 * 16 * KVBLK * 2;
    auto stream = c10::hip::getCurrentHIPStream().stream();
    #define LAUNCH_D(DD)                                                     \
        hipLaunchKernelGGL(attn_fwd_kernel<DD>, dim3(grid), dim3(THREADS),   \
            lds, stream,                                                     \
            reinterpret_cast<const bf16*>(qc.data_ptr()),                    \
            reinterpret_cast<const bf16*>(kc.data_ptr()),                    \
            reinterpret_cast<const bf16*>(vc.data_ptr()),                    \
            reinterpret_cast<bf16*>(o.data_ptr()),                           \
            lse.data_ptr<float>(),                                           \
            (int)B, (int)H, (int)Hkv, (int)S, (float)scale)
    switch (D) {
        case 64: LAUNCH_D(64); break;
        case 80: LAUNCH_D(80); break;
        case 96: LAUNCH_D(96); break;
        case 128: LAUNCH_D(128); break;
        default: TORCH_CHECK(false, "head dim must be 64/80/96/128, got ", D);
    }
    #undef LAUNCH_D
    HIP_CHECK_LAST();
    return {o, lse};
}
