// Flash attention forward (causal, GQA) for gfx950 — bf16 I/O, fp32
// online softmax, MFMA (v_mfma_f32_16x16x32_bf16) for QK^T and PV.
//
// v2 structure:
//   * block = 256 threads = 4 waves covering 128 q rows; each wave owns
//     TWO 16-row blocks paired as (w, 7-w) so causal work is balanced
//     across waves (early rows pair with late rows);
//   * KV tiles of 64 columns; K is staged into LDS in natural [64][D]
//     layout with one 16-B slot of row padding (breaks the power-of-2
//     row-stride bank conflict), V transposed into LDS [D][64+8];
//   * Q fragments preloaded to registers; per-row (m, l) online softmax;
//     P goes through a per-wave LDS tile (C layout -> A layout), published
//     under the block barrier that also keeps control flow uniform;
//   * D in {64, 80, 96, 128} as template instantiations so every
//     accumulator array is compile-time indexed (registers, not scratch).
//
// Saves the row LSE (m + log l) for the backward pass.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

typedef short short4v __attribute__((ext_vector_type(4)));

constexpr int THREADS = 256;
constexpr int QBLK = 128;     // q rows per block (2 x 16 per wave)
constexpr int VPAD = 8;       // Vt row padding (bf16 elements)

// KV tile width: 128 for D <= 80 (the K/V^T/P LDS images still fit two
// blocks per CU), 64 above. Wider tiles amortize the two barriers per
// tile over twice the MFMA work.
template <int D>
constexpr int kvblk_for() { return D <= 80 ? 128 : 64; }

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
    const int col16 = lane & 15;
    const int k8 = lane >> 4;

    constexpr int KVBLK = kvblk_for<D>();    // kv columns per tile
    constexpr int JSUB = KVBLK / 16;         // 16-col S subtiles per tile
    constexpr int dchunks = (D + 31) / 32;   // 32-wide K-dim chunks of D
    constexpr int djtiles = D / 16;          // 16-wide output column tiles
    constexpr int KSLOT = D / 8 + 1;         // K LDS slots per row (padded)
    constexpr int VROW = KVBLK + VPAD;       // Vt LDS row length

    const int qtile = blockIdx.x % (S / QBLK);
    const int head = (blockIdx.x / (S / QBLK)) % H;
    const int batch = blockIdx.x / (S / QBLK) / H;
    const int kv_head = head / (H / Hkv);

    const long q_base = (((long)batch * H + head) * S) * D;
    const long kv_base = (((long)batch * Hkv + kv_head) * S) * D;

    // this wave's two 16-row blocks: rb index within the 128-row block
    const int rbid[2] = {wave, 7 - wave};
    const int qb = qtile * QBLK;

    extern __shared__ __attribute__((aligned(16))) char smem[];
    // raw bf16 BITS as shorts (a short into __hip_bfloat16 converts
    // numerically — never store element-wise through the struct type).
    // V stays in its NATURAL [kv][D] image: PV B-fragments read it with
    // ds_read_b64_tr_b16 (hardware transpose), so no scalar transpose
    // staging exists anywhere in this kernel.
    short* Ks = reinterpret_cast<short*>(smem);                 // [KVBLK][KSLOT*8]
    short* Vs = Ks + KVBLK * KSLOT * 8;                         // [KVBLK][KSLOT*8]
    short* Pw = Vs + KVBLK * KSLOT * 8 + wave * 2 * 16 * VROW;  // [2][16][VROW]

    // ---- preload Q fragments for both row blocks ------------------------
    bf16x8 q_frag[2][dchunks];
    #pragma unroll
    for (int rb = 0; rb < 2; ++rb) {
        const int qrow = qb + rbid[rb] * 16 + col16;
        #pragma unroll
        for (int c = 0; c < dchunks; ++c) {
            const int d0 = c * 32 + k8 * 8;
            if (d0 < D) {
                q_frag[rb][c] = *reinterpret_cast<const bf16x8*>(
                    Q + q_base + (long)qrow * D + d0);
            } else {
                #pragma unroll
                for (int i = 0; i < 8; ++i) q_frag[rb][c][i] = 0;
            }
        }
    }

    float m_run[2][4], l_run[2][4];
    #pragma unroll
    for (int rb = 0; rb < 2; ++rb)
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            m_run[rb][r] = -1e30f;
            l_run[rb][r] = 0.f;
        }
    floatx4 o_acc[2][djtiles];
    #pragma unroll
    for (int rb = 0; rb < 2; ++rb)
        #pragma unroll
        for (int jd = 0; jd < djtiles; ++jd)
            o_acc[rb][jd] = floatx4{0.f, 0.f, 0.f, 0.f};

    // T14 async-stage split: each tile's K/V chunks are loaded to
    // registers one tile AHEAD (HBM latency hides under the previous
    // tile's compute); the LDS write happens at loop top from registers.
    // 256 threads x (K: 4-row d-walk, V: kv-row walk) per the conflict
    // analysis below.
    constexpr int chunks = KVBLK * D / 8;           // 16-B chunks per tile
    constexpr int per_thread = (chunks + THREADS - 1) / THREADS;
    bf16x8 k_stage[per_thread], v_stage[per_thread];

    auto issue_loads = [&](int kv0) {
        #pragma unroll
        for (int u = 0; u < per_thread; ++u) {
            const int c = threadIdx.x + u * THREADS;
            if (c >= chunks) break;
            // K: lanes walk d-chunks within a row (coalesced reads,
            // conflict-free vector LDS writes at row stride 272 B)
            const int krow = c / (D / 8);
            const int ks8 = c % (D / 8);
            k_stage[u] = *reinterpret_cast<const bf16x8*>(
                K + kv_base + (long)(kv0 + krow) * D + ks8 * 8);
            // V: lanes walk KV ROWS so the scalar transpose writes hit
            // contiguous LDS addresses (d-major mapping was a 16-way
            // bank conflict: 5.5e9 measured conflict cycles)
            const int vrow = c % KVBLK;
            const int vs8 = c / KVBLK;
            v_stage[u] = *reinterpret_cast<const bf16x8*>(
                V + kv_base + (long)(kv0 + vrow) * D + vs8 * 8);
        }
    };
    auto write_stage = [&]() {
        #pragma unroll
        for (int u = 0; u < per_thread; ++u) {
            const int c = threadIdx.x + u * THREADS;
            if (c >= chunks) break;
            const int krow = c / (D / 8);
            const int ks8 = c % (D / 8);
            *reinterpret_cast<bf16x8*>(Ks + (krow * KSLOT + ks8) * 8) = k_stage[u];
            const int vrow = c % KVBLK;
            const int vs8 = c / KVBLK;
            *reinterpret_cast<bf16x8*>(Vs + (vrow * KSLOT + vs8) * 8) = v_stage[u];
        }
    };

    const int kv_end = qb + QBLK;
    issue_loads(0);
    for (int kv0 = 0; kv0 < kv_end; kv0 += KVBLK) {
        write_stage();
        if (kv0 + KVBLK < kv_end) issue_loads(kv0 + KVBLK);
        __syncthreads();

        // ---- per row block: S = scale * Q K^T, online softmax, P -> LDS -
        bool rb_active[2];
        float p[2][JSUB][4];   // [rb][j][r]
        #pragma unroll
        for (int rb = 0; rb < 2; ++rb) {
            const int q0 = qb + rbid[rb] * 16;
            rb_active[rb] = kv0 <= q0 + 15;
            if (!rb_active[rb]) continue;

            __builtin_amdgcn_s_setprio(1);   // T5: favor the MFMA cluster
            #pragma unroll
            for (int j = 0; j < JSUB; ++j) {    // 16-col subtiles
                floatx4 s_acc = floatx4{0.f, 0.f, 0.f, 0.f};
                const int kvrow = j * 16 + col16;
                #pragma unroll
                for (int c = 0; c < dchunks; ++c) {
                    const int d0 = c * 32 + k8 * 8;
                    bf16x8 k_frag;
                    if (d0 < D) {
                        k_frag = *reinterpret_cast<const bf16x8*>(
                            Ks + (kvrow * KSLOT + d0 / 8) * 8);
                    } else {
                        #pragma unroll
                        for (int i = 0; i < 8; ++i) k_frag[i] = 0;
                    }
                    s_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        q_frag[rb][c], k_frag, s_acc, 0, 0, 0);
                }
                #pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int qrow = q0 + k8 * 4 + r;
                    const int kvcol = kv0 + j * 16 + col16;
                    float sv = s_acc[r] * scale;
                    p[rb][j][r] = (kvcol > qrow) ? -1e30f : sv;
                }
            }
            __builtin_amdgcn_s_setprio(0);

            #pragma unroll
            for (int r = 0; r < 4; ++r) {
                float tile_max = -1e30f;
                #pragma unroll
                for (int j = 0; j < JSUB; ++j)
                    tile_max = fmaxf(tile_max, p[rb][j][r]);
                #pragma unroll
                for (int off = 8; off > 0; off >>= 1)
                    tile_max = fmaxf(tile_max, __shfl_xor(tile_max, off, 16));

                const float m_new = fmaxf(m_run[rb][r], tile_max);
                const float alpha = __expf(m_run[rb][r] - m_new);
                float row_sum = 0.f;
                #pragma unroll
                for (int j = 0; j < JSUB; ++j) {
                    p[rb][j][r] = __expf(p[rb][j][r] - m_new);
                    row_sum += p[rb][j][r];
                }
                #pragma unroll
                for (int off = 8; off > 0; off >>= 1)
                    row_sum += __shfl_xor(row_sum, off, 16);

                l_run[rb][r] = l_run[rb][r] * alpha + row_sum;
                m_run[rb][r] = m_new;
                #pragma unroll
                for (int jd = 0; jd < djtiles; ++jd)
                    o_acc[rb][jd][r] *= alpha;
            }

            short* Prb = Pw + rb * 16 * VROW;
            #pragma unroll
            for (int j = 0; j < JSUB; ++j)
                #pragma unroll
                for (int r = 0; r < 4; ++r)
                    Prb[(k8 * 4 + r) * VROW + j * 16 + col16] =
                        float_to_bf16_bits(p[rb][j][r]);
        }

        // P is per-wave private: DS ops of one wave complete in order, so
        // a compiler-level fence (no barrier) suffices to keep the vector
        // re-read below the scalar writes above.
        asm volatile("" ::: "memory");

        // ---- O += P @ V -------------------------------------------------
        #pragma unroll
        for (int rb = 0; rb < 2; ++rb) {
            if (!rb_active[rb]) continue;
            const short* Prb = Pw + rb * 16 * VROW;
            // V B-fragments via hardware transpose read: each 16-lane
            // group cooperatively loads one 4(kv)x16(d) block — lane
            // l&15 receives its d-column's 4 kv values. Per fragment,
            // two tr16 reads cover the 8 kv rows of this lane group.
            const int p4 = lane & 15;     // piece index within the group
            #pragma unroll
            for (int ks = 0; ks < KVBLK / 32; ++ks) {   // 32-wide kv chunks
                bf16x8 p_frag = *reinterpret_cast<const bf16x8*>(
                    Prb + col16 * VROW + ks * 32 + k8 * 8);
                #pragma unroll
                for (int jd = 0; jd < djtiles; ++jd) {
                    bf16x8 v_frag;
                    #pragma unroll
                    for (int r = 0; r < 2; ++r) {
                        const int kvrow = ks * 32 + k8 * 8 + 4 * r + (p4 >> 2);
                        const int col = jd * 16 + (p4 & 3) * 4;
                        short4v t = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
                            (__attribute__((address_space(3))) short4v*)(
                                Vs + kvrow * KSLOT * 8 + col));
                        v_frag[r * 4 + 0] = t[0];
                        v_frag[r * 4 + 1] = t[1];
                        v_frag[r * 4 + 2] = t[2];
                        v_frag[r * 4 + 3] = t[3];
                    }
                    o_acc[rb][jd] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        p_frag, v_frag, o_acc[rb][jd], 0, 0, 0);
                }
            }
        }
        __syncthreads();   // K/Vt/P reused next tile
    }

    // ---- normalize + store ----------------------------------------------
    #pragma unroll
    for (int rb = 0; rb < 2; ++rb) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int qrow = qb + rbid[rb] * 16 + k8 * 4 + r;
            const float inv_l = 1.f / l_run[rb][r];
            #pragma unroll
            for (int jd = 0; jd < djtiles; ++jd)
                O[q_base + (long)qrow * D + jd * 16 + col16] =
                    __float2bfloat16(o_acc[rb][jd][r] * inv_l);
            if (col16 == 0)
                LSE[((long)batch * H + head) * S + qrow] =
                    m_run[rb][r] + __logf(l_run[rb][r]);
        }
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
    TORCH_CHECK(S % QBLK == 0, "sequence length must be a multiple of 128");
    TORCH_CHECK(H % Hkv == 0, "GQA requires H % Hkv == 0");
    auto qc = q.contiguous(), kc = k.contiguous(), vc = v.contiguous();

    auto o = torch::empty_like(qc);
    auto lse = torch::empty({B, H, S}, q.options().dtype(torch::kFloat32));

    const int grid = (int)(B * H * (S / QBLK));
    auto stream = c10::hip::getCurrentHIPStream().stream();
    #define LAUNCH_D(DD)                                                      \
        do {                                                                  \
            const int kvb = kvblk_for<DD>();                                  \
            const int lds = (2 * kvb * (DD / 8 + 1) * 8                       \
                             + 4 * 2 * 16 * (kvb + VPAD)) * 2;                \
            hipLaunchKernelGGL(attn_fwd_kernel<DD>, dim3(grid),               \
                dim3(THREADS), lds, stream,                                   \
                reinterpret_cast<const bf16*>(qc.data_ptr()),                 \
                reinterpret_cast<const bf16*>(kc.data_ptr()),                 \
                reinterpret_cast<const bf16*>(vc.data_ptr()),                 \
                reinterpret_cast<bf16*>(o.data_ptr()),                        \
                lse.data_ptr<float>(),                                        \
                (int)B, (int)H, (int)Hkv, (int)S, (float)scale);              \
        } while (0)
    switch ((int)D) {
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
