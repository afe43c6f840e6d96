// Flash attention backward for gfx950 (causal, GQA) — bf16 I/O, fp32
// accumulation, MFMA 16x16x32, LSE-based recompute.
//
// v2 structure (mirrors attention.hip's forward):
//   * split dQ / dKV kernels (deterministic, no atomics); GQA dK/dV come
//     back per q-head and the wrapper reduces the group;
//   * block = 256 threads = 4 waves; each wave owns TWO 16-row blocks
//     paired (w, 7-w) so causal work balances across waves;
//   * opposing-side tiles are 64 wide; natural-layout A/B fragments read
//     straight from global (the 16-32 KB tiles stay L2-hot), transposed
//     operands (K^T for dQ; Q^T / dO^T for dK, dV) staged into padded LDS
//     with the async T14 split: the next tile's chunks load to registers
//     under the current tile's compute;
//   * dS / P^T redistribute through per-wave LDS (raw bf16 bits as
//     shorts), published by a wave-local compiler fence — DS ops of one
//     wave complete in order;
//   * D in {64, 80, 96, 128} template instantiations (register arrays
//     compile-time indexed).
//
//   dQ kernel:  P = exp(scale*QK^T - lse); dP = dO V^T;
//               dS = scale * P o (dP - delta); dQ += dS @ K
//   dKV kernel: P^T = exp(scale*K Q^T - lse[col]); dP^T = V dO^T;
//               dV += P^T @ dO; dK += dS^T @ Q

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

typedef short short4v __attribute__((ext_vector_type(4)));

constexpr int THREADS = 256;
constexpr int RBLK = 128;    // own-side rows per block (2 x 16 per wave)
constexpr int CTILE = 64;    // opposing-side tile width
constexpr int VPAD = 8;      // transposed-tile row padding (bf16)
constexpr int DMAX = 128;

// B-fragment via ds_read_b64_tr_b16 from a NATURAL [row][KSLOT*8] LDS
// image: per 16-lane group, two reads cover the 8 opposing-index rows;
// lane l&15 receives its d-column (semantics pinned by ext.tr16_probe).
template <int KSLOT>
__device__ __forceinline__ bf16x8 tr16_frag(
    const short* img, int row0, int col0, int p4) {
    bf16x8 out;
    #pragma unroll
    for (int r = 0; r < 2; ++r) {
        const int row = row0 + 4 * r + (p4 >> 2);
        const int col = col0 + (p4 & 3) * 4;
        short4v t = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
            (__attribute__((address_space(3))) short4v*)(
                img + row * KSLOT * 8 + col));
        out[r * 4 + 0] = t[0];
        out[r * 4 + 1] = t[1];
        out[r * 4 + 2] = t[2];
        out[r * 4 + 3] = t[3];
    }
    return out;
}

template <int D>
__device__ __forceinline__ bf16x8 frag8(const bf16* base, long row, int d0) {
    if (d0 < D)
        return *reinterpret_cast<const bf16x8*>(base + row * D + d0);
    bf16x8 z;
    #pragma unroll
    for (int i = 0; i < 8; ++i) z[i] = 0;
    return z;
}

// ---------------------------------------------------------------- dQ ----
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

    constexpr int dchunks = (D + 31) / 32;
    constexpr int djtiles = D / 16;
    constexpr int VROW = CTILE + VPAD;

    const int qtile = blockIdx.x % (S / RBLK);
    const int head = (blockIdx.x / (S / RBLK)) % H;
    const int batch = blockIdx.x / (S / RBLK) / H;
    const int kv_head = head / (H / Hkv);
    const long q_base = (((long)batch * H + head) * S) * D;
    const long kv_base = (((long)batch * Hkv + kv_head) * S) * D;
    const long row_base = ((long)batch * H + head) * S;

    const int rbid[2] = {wave, 7 - wave};
    const int qb = qtile * RBLK;

    constexpr int KSLOT = D / 8 + 1;    // natural-tile slots per row (padded)
    extern __shared__ __attribute__((aligned(16))) char smem[];
    short* Ks = reinterpret_cast<short*>(smem);                    // [CTILE][KSLOT*8]
    short* Vs = Ks + CTILE * KSLOT * 8;                            // [CTILE][KSLOT*8]
    short* Sw = Vs + CTILE * KSLOT * 8 + wave * 2 * 16 * VROW;     // [2][16][VROW]

    // per-row state and Q/dO fragments for both row blocks
    float lse_r[2][4], delta_r[2][4];
    bf16x8 q_frag[2][dchunks], do_frag[2][dchunks];
    #pragma unroll
    for (int rb = 0; rb < 2; ++rb) {
        const int q0 = qb + rbid[rb] * 16;
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            lse_r[rb][r] = LSE[row_base + q0 + k8 * 4 + r];
            delta_r[rb][r] = Delta[row_base + q0 + k8 * 4 + r];
        }
        #pragma unroll
        for (int c = 0; c < dchunks; ++c) {
            q_frag[rb][c] = frag8<D>(Q + q_base, q0 + col16, c * 32 + k8 * 8);
            do_frag[rb][c] = frag8<D>(dO + q_base, q0 + col16, c * 32 + k8 * 8);
        }
    }

    floatx4 dq_acc[2][djtiles];
    #pragma unroll
    for (int rb = 0; rb < 2; ++rb)
        #pragma unroll
        for (int jd = 0; jd < djtiles; ++jd)
            dq_acc[rb][jd] = floatx4{0.f, 0.f, 0.f, 0.f};

    // T14 staged K (reused for the natural tile AND the transpose) plus
    // a direct-staged natural V tile; kv-row-major lane walk keeps the
    // transpose writes conflict-free
    constexpr int chunks = CTILE * D / 8;
    constexpr int per_thread = (chunks + THREADS - 1) / THREADS;
    bf16x8 k_stage[per_thread];
    auto issue_loads = [&](int kv0) {
        #pragma unroll
        for (int u = 0; u < per_thread; ++u) {
            const int c = threadIdx.x + u * THREADS;
            if (c >= chunks) break;
            k_stage[u] = *reinterpret_cast<const bf16x8*>(
                K + kv_base + (long)(kv0 + c % CTILE) * D + (c / CTILE) * 8);
        }
    };
    auto write_stage = [&](int kv0) {
        #pragma unroll
        for (int u = 0; u < per_thread; ++u) {
            const int c = threadIdx.x + u * THREADS;
            if (c >= chunks) break;
            const int row = c % CTILE;
            const int d0 = (c / CTILE) * 8;
            *reinterpret_cast<bf16x8*>(Ks + row * KSLOT * 8 + d0) = k_stage[u];
            bf16x8 vv = *reinterpret_cast<const bf16x8*>(
                V + kv_base + (long)(kv0 + row) * D + d0);
            *reinterpret_cast<bf16x8*>(Vs + row * KSLOT * 8 + d0) = vv;
        }
    };

    const int kv_end = qb + RBLK;
    issue_loads(0);
    for (int kv0 = 0; kv0 < kv_end; kv0 += CTILE) {
        write_stage(kv0);
        if (kv0 + CTILE < kv_end) issue_loads(kv0 + CTILE);
        __syncthreads();

        bool rb_active[2];
        #pragma unroll
        for (int rb = 0; rb < 2; ++rb) {
            const int q0 = qb + rbid[rb] * 16;
            rb_active[rb] = kv0 <= q0 + 15;
            if (!rb_active[rb]) continue;

            __builtin_amdgcn_s_setprio(1);
            floatx4 s_acc[4], dp_acc[4];
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                s_acc[j] = floatx4{0.f, 0.f, 0.f, 0.f};
                dp_acc[j] = floatx4{0.f, 0.f, 0.f, 0.f};
                const int kvrow = j * 16 + col16;
                #pragma unroll
                for (int c = 0; c < dchunks; ++c) {
                    const int d0 = c * 32 + k8 * 8;
                    bf16x8 kf, vf;
                    if (d0 < D) {
                        kf = *reinterpret_cast<const bf16x8*>(
                            Ks + kvrow * KSLOT * 8 + d0);
                        vf = *reinterpret_cast<const bf16x8*>(
                            Vs + kvrow * KSLOT * 8 + d0);
                    } else {
                        #pragma unroll
                        for (int i = 0; i < 8; ++i) { kf[i] = 0; vf[i] = 0; }
                    }
                    s_acc[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        q_frag[rb][c], kf, s_acc[j], 0, 0, 0);
                    dp_acc[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        do_frag[rb][c], vf, dp_acc[j], 0, 0, 0);
                }
            }
            __builtin_amdgcn_s_setprio(0);

            short* Srb = Sw + rb * 16 * VROW;
            #pragma unroll
            for (int j = 0; j < 4; ++j)
                #pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int qrow = q0 + k8 * 4 + r;
                    const int kvcol = kv0 + j * 16 + col16;
                    float p = (kvcol > qrow)
                                  ? 0.f
                                  : __expf(s_acc[j][r] * scale - lse_r[rb][r]);
                    float ds = scale * p * (dp_acc[j][r] - delta_r[rb][r]);
                    Srb[(k8 * 4 + r) * VROW + j * 16 + col16] =
                        float_to_bf16_bits(ds);
                }
        }

        // wave-local publish of dS (DS ops of one wave are in order)
        asm volatile("" ::: "memory");

        #pragma unroll
        for (int rb = 0; rb < 2; ++rb) {
            if (!rb_active[rb]) continue;
            const short* Srb = Sw + rb * 16 * VROW;
            const int p4 = lane & 15;
            #pragma unroll
            for (int ks = 0; ks < 2; ++ks) {
                bf16x8 ds_frag = *reinterpret_cast<const bf16x8*>(
                    Srb + col16 * VROW + ks * 32 + k8 * 8);
                #pragma unroll
                for (int jd = 0; jd < djtiles; ++jd) {
                    bf16x8 kt_frag = tr16_frag<KSLOT>(
                        Ks, ks * 32 + k8 * 8, jd * 16, p4);
                    dq_acc[rb][jd] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        ds_frag, kt_frag, dq_acc[rb][jd], 0, 0, 0);
                }
            }
        }
        __syncthreads();
    }

    #pragma unroll
    for (int rb = 0; rb < 2; ++rb)
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            const long qrow = qb + rbid[rb] * 16 + k8 * 4 + r;
            #pragma unroll
            for (int jd = 0; jd < djtiles; ++jd)
                dQ[q_base + qrow * D + jd * 16 + col16] =
                    __float2bfloat16(dq_acc[rb][jd][r]);
        }
}

// --------------------------------------------------------------- dKV ----
// One 16-row kv block per wave. At D=64 the block is EIGHT waves (128
// kv rows): all waves share one staged Q/dO stream, halving the number
// of Q/dO sweeps over the sequence (the kernel is wait-bound - 57-64%
// SQ_WAIT_ANY in the r2 PMC run). Measured: +15% at D=64; at D=80 the
// same change was -8.5% and at D=128/S=4096 -9.5% (deeper per-wave
// VGPR footprints lose the second independent block's latency
// cover), so those stay 4-wave x 64 rows.
template <int D>
__global__ __launch_bounds__(D == 64 ? 2 * THREADS : THREADS, 2)
void attn_bwd_dkv_kernel(
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

    constexpr int dchunks = (D + 31) / 32;
    constexpr int djtiles = D / 16;
    constexpr int VROW = CTILE + VPAD;
    constexpr int KSLOT = D / 8 + 1;
    constexpr int KVB = D == 64 ? 128 : 64;  // kv rows per block
    constexpr int NW = KVB / 16;             // waves per block

    const int kvtile = blockIdx.x % (S / KVB);
    const int head = (blockIdx.x / (S / KVB)) % H;
    const int batch = blockIdx.x / (S / KVB) / H;
    const int kv_head = head / (H / Hkv);
    const long q_base = (((long)batch * H + head) * S) * D;
    const long kv_base = (((long)batch * Hkv + kv_head) * S) * D;
    const long out_base = (((long)batch * H + head) * S) * D;
    const long row_base = ((long)batch * H + head) * S;

    const int kv0 = kvtile * KVB + wave * 16;   // this wave's kv rows

    // Natural Q/dO LDS images only: the dK/dV B-fragments come from
    // ds_read_b64_tr_b16 straight off these, so the scalar-transposed
    // Qt/dOt copies (and their LDS staging writes) are gone; the freed
    // LDS restores 2 blocks/CU at every D.
    extern __shared__ __attribute__((aligned(16))) char smem[];
    short* Qs = reinterpret_cast<short*>(smem);                    // [CTILE][KSLOT*8]
    short* dOs = Qs + CTILE * KSLOT * 8;                           // [CTILE][KSLOT*8]
    short* Sw = dOs + CTILE * KSLOT * 8 + wave * 16 * VROW;        // [16][VROW]
    short* Pw = dOs + CTILE * KSLOT * 8 + NW * 16 * VROW
                + wave * 16 * VROW;                                // [16][VROW]

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

    // T14 staged Q and dO (each register set feeds both the natural and
    // transposed LDS image); at D=128 the dO set would spill past 256
    // VGPRs on top of the dK+dV accumulators, so dO stages direct there.
    constexpr int NT = NW * 64;         // staging threads
    constexpr int chunks = CTILE * D / 8;
    constexpr int per_thread = (chunks + NT - 1) / NT;
    constexpr bool STAGE_DO = (D <= 96);
    bf16x8 q_stage[per_thread];
    bf16x8 do_stage[STAGE_DO ? per_thread : 1];
    auto issue_loads = [&](int q0) {
        #pragma unroll
        for (int u = 0; u < per_thread; ++u) {
            const int c = threadIdx.x + u * NT;
            if (c >= chunks) break;
            q_stage[u] = *reinterpret_cast<const bf16x8*>(
                Q + q_base + (long)(q0 + c % CTILE) * D + (c / CTILE) * 8);
            if constexpr (STAGE_DO)
                do_stage[u] = *reinterpret_cast<const bf16x8*>(
                    dO + q_base + (long)(q0 + c % CTILE) * D + (c / CTILE) * 8);
        }
    };
    auto write_stage = [&](int q0) {
        #pragma unroll
        for (int u = 0; u < per_thread; ++u) {
            const int c = threadIdx.x + u * NT;
            if (c >= chunks) break;
            const int row = c % CTILE;
            const int d0 = (c / CTILE) * 8;
            *reinterpret_cast<bf16x8*>(Qs + row * KSLOT * 8 + d0) = q_stage[u];
            bf16x8 dov;
            if constexpr (STAGE_DO)
                dov = do_stage[u];
            else
                dov = *reinterpret_cast<const bf16x8*>(
                    dO + q_base + (long)(q0 + row) * D + d0);
            *reinterpret_cast<bf16x8*>(dOs + row * KSLOT * 8 + d0) = dov;
        }
    };

    const int q_start = kvtile * KVB;   // causal: from the block's kv start
    issue_loads(q_start);
    for (int q0 = q_start; q0 < S; q0 += CTILE) {
        write_stage(q0);
        if (q0 + CTILE < S) issue_loads(q0 + CTILE);
        __syncthreads();

        const bool active = q0 + CTILE - 1 >= kv0;
        if (active) {
            __builtin_amdgcn_s_setprio(1);
            floatx4 st_acc[4], dpt_acc[4];
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                st_acc[j] = floatx4{0.f, 0.f, 0.f, 0.f};
                dpt_acc[j] = floatx4{0.f, 0.f, 0.f, 0.f};
                const int qrow = j * 16 + col16;
                #pragma unroll
                for (int c = 0; c < dchunks; ++c) {
                    const int d0 = c * 32 + k8 * 8;
                    bf16x8 qf, dof;
                    if (d0 < D) {
                        qf = *reinterpret_cast<const bf16x8*>(
                            Qs + qrow * KSLOT * 8 + d0);
                        dof = *reinterpret_cast<const bf16x8*>(
                            dOs + qrow * KSLOT * 8 + d0);
                    } else {
                        #pragma unroll
                        for (int i = 0; i < 8; ++i) { qf[i] = 0; dof[i] = 0; }
                    }
                    st_acc[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        k_frag[c], qf, st_acc[j], 0, 0, 0);
                    dpt_acc[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        v_frag[c], dof, dpt_acc[j], 0, 0, 0);
                }
            }
            __builtin_amdgcn_s_setprio(0);

            #pragma unroll
            for (int j = 0; j < 4; ++j)
                #pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int kvrow = kv0 + k8 * 4 + r;
                    const int qcol = q0 + j * 16 + col16;
                    float p = (qcol < kvrow)
                                  ? 0.f
                                  : __expf(st_acc[j][r] * scale
                                           - LSE[row_base + qcol]);
                    float ds = scale * p
                               * (dpt_acc[j][r] - Delta[row_base + qcol]);
                    Sw[(k8 * 4 + r) * VROW + j * 16 + col16] =
                        float_to_bf16_bits(ds);
                    Pw[(k8 * 4 + r) * VROW + j * 16 + col16] =
                        float_to_bf16_bits(p);
                }

            asm volatile("" ::: "memory");   // wave-local publish

            const int p4 = lane & 15;
            #pragma unroll
            for (int ks = 0; ks < 2; ++ks) {
                bf16x8 dst_frag = *reinterpret_cast<const bf16x8*>(
                    Sw + col16 * VROW + ks * 32 + k8 * 8);
                bf16x8 pt_frag = *reinterpret_cast<const bf16x8*>(
                    Pw + col16 * VROW + ks * 32 + k8 * 8);
                #pragma unroll
                for (int jd = 0; jd < djtiles; ++jd) {
                    bf16x8 qt_frag = tr16_frag<KSLOT>(
                        Qs, ks * 32 + k8 * 8, jd * 16, p4);
                    dk_acc[jd] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        dst_frag, qt_frag, dk_acc[jd], 0, 0, 0);
                    bf16x8 dot_frag = tr16_frag<KSLOT>(
                        dOs, ks * 32 + k8 * 8, jd * 16, p4);
                    dv_acc[jd] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        pt_frag, dot_frag, dv_acc[jd], 0, 0, 0);
                }
            }
        }
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
    TORCH_CHECK(S % RBLK == 0, "sequence length must be a multiple of 128");
    auto qc = q.contiguous(), kc = k.contiguous(), vc = v.contiguous();
    auto doc = d_o.contiguous();
    auto lsec = lse.contiguous(), dc = delta.contiguous();

    auto dq = torch::empty_like(qc);
    auto dk = torch::empty({B, H, S, D}, q.options());
    auto dv = torch::empty({B, H, S, D}, q.options());

    auto stream = c10::hip::getCurrentHIPStream().stream();
    const int grid = (int)(B * H * (S / RBLK));
    // dkv grid: one block per KVB kv rows (KVB is 128 at D=64, 64 else)

    #define LAUNCH_BWD(DD)                                                    \
        do {                                                                  \
            const int vrow = CTILE + VPAD;                                    \
            const int lds_dq = (2 * CTILE * (DD / 8 + 1) * 8                  \
                                + DD * vrow + 4 * 2 * 16 * vrow) * 2;         \
            const int nw_dkv = (DD == 64) ? 8 : 4;                            \
            const int lds_dkv = (2 * CTILE * (DD / 8 + 1) * 8                 \
                                 + 2 * nw_dkv * 16 * vrow) * 2;               \
            hipLaunchKernelGGL(attn_bwd_dq_kernel<DD>, dim3(grid),            \
                dim3(THREADS), lds_dq, stream,                                \
                reinterpret_cast<const bf16*>(qc.data_ptr()),                 \
                reinterpret_cast<const bf16*>(kc.data_ptr()),                 \
                reinterpret_cast<const bf16*>(vc.data_ptr()),                 \
                reinterpret_cast<const bf16*>(doc.data_ptr()),                \
                lsec.data_ptr<float>(), dc.data_ptr<float>(),                 \
                reinterpret_cast<bf16*>(dq.data_ptr()),                       \
                (int)B, (int)H, (int)Hkv, (int)S, (float)scale);              \
            const int kvb_dkv = (DD == 64) ? 128 : 64;                        \
            const int grid_dkv = (int)(B * H * (S / kvb_dkv));                \
            hipLaunchKernelGGL(attn_bwd_dkv_kernel<DD>, dim3(grid_dkv),       \
                dim3((DD == 64) ? 2 * THREADS : THREADS), lds_dkv, stream,    \
                reinterpret_cast<const bf16*>(qc.data_ptr()),                 \
                reinterpret_cast<const bf16*>(kc.data_ptr()),                 \
                reinterpret_cast<const bf16*>(vc.data_ptr()),                 \
                reinterpret_cast<const bf16*>(doc.data_ptr()),                \
                lsec.data_ptr<float>(), dc.data_ptr<float>(),                 \
                reinterpret_cast<bf16*>(dk.data_ptr()),                       \
                reinterpret_cast<bf16*>(dv.data_ptr()),                       \
                (int)B, (int)H, (int)Hkv, (int)S, (float)scale);              \
        } while (0)
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
