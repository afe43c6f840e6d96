// Single-query (decode) attention for gfx950 — the serving hot loop.
//
// Shape: q [B, H, 1, D] against cached K/V [B, Hkv, S, D] (GQA maps h ->
// h / (H/Hkv)). Matrix-vector work: no MFMA — the bound is streaming
// S*D*2 bytes of K and V per (b, h) from HBM, so the kernel keeps q in
// registers, walks rows with wave-wide coalesced loads, and carries an
// online-softmax accumulator; the four waves of a workgroup split the
// rows and merge (m, l, acc) triples through LDS at the end.
//
// One workgroup per (b, h): a decode batch of B*H >= ~2k workgroups
// fills the chip; a split-S two-pass variant for tiny batches is a
// follow-up (TODO.md).

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int THREADS = 256;
constexpr int NWAVE = THREADS / WAVE_SIZE;

// VPL = elements of a D-length vector each lane owns (lane l holds
// elements l, l+64, ...)
template <int D>
__global__ __launch_bounds__(THREADS) void attn_decode_kernel(
    const bf16* __restrict__ Q,   // [B, H, D]
    const bf16* __restrict__ K,   // [B, Hkv, S, D]
    const bf16* __restrict__ V,
    bf16* __restrict__ Out,       // [B, H, D]
    int B, int H, int Hkv, int S, float scale) {
    constexpr int VPL = (D + WAVE_SIZE - 1) / WAVE_SIZE;
    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
    const int h = blockIdx.x % H;
    const int b = blockIdx.x / H;
    const int kvh = h / (H / Hkv);
    const long q_base = ((long)b * H + h) * D;
    const long kv_base = (((long)b * Hkv + kvh) * S) * D;

    // q in registers (each lane holds its D-strided elements)
    float qv[VPL];
    #pragma unroll
    for (int i = 0; i < VPL; ++i) {
        const int d = lane + i * WAVE_SIZE;
        qv[i] = d < D ? bf16_bits_to_float(
            reinterpret_cast<const short*>(Q)[q_base + d]) : 0.f;
    }

    float m = -1e30f, l = 0.f;
    float acc[VPL];
    #pragma unroll
    for (int i = 0; i < VPL; ++i) acc[i] = 0.f;

    for (int r = wave; r < S; r += NWAVE) {
        const short* krow = reinterpret_cast<const short*>(K) + kv_base + (long)r * D;
        const short* vrow = reinterpret_cast<const short*>(V) + kv_base + (long)r * D;
        float dot = 0.f;
        #pragma unroll
        for (int i = 0; i < VPL; ++i) {
            const int d = lane + i * WAVE_SIZE;
            if (d < D) dot += qv[i] * bf16_bits_to_float(krow[d]);
        }
        dot = wave_reduce_sum(dot) * scale;   // broadcast to all lanes

        const float m_new = fmaxf(m, dot);
        const float alpha = __expf(m - m_new);
        const float p = __expf(dot - m_new);
        l = l * alpha + p;
        #pragma unroll
        for (int i = 0; i < VPL; ++i) {
            const int d = lane + i * WAVE_SIZE;
            acc[i] = acc[i] * alpha
                     + (d < D ? p * bf16_bits_to_float(vrow[d]) : 0.f);
        }
        m = m_new;
    }

    // merge the 4 wave-partials through LDS: (m_w, l_w, acc_w)
    __shared__ float sm[NWAVE], sl[NWAVE];
    __shared__ float sacc[NWAVE][VPL * WAVE_SIZE];
    if (lane == 0) { sm[wave] = m; sl[wave] = l; }
    #pragma unroll
    for (int i = 0; i < VPL; ++i)
        sacc[wave][lane + i * WAVE_SIZE] = acc[i];
    __syncthreads();

    if (wave == 0) {
        float gm = -1e30f;
        #pragma unroll
        for (int w = 0; w < NWAVE; ++w) gm = fmaxf(gm, sm[w]);
        float gl = 0.f;
        float gacc[VPL];
        #pragma unroll
        for (int i = 0; i < VPL; ++i) gacc[i] = 0.f;
        #pragma unroll
        for (int w = 0; w < NWAVE; ++w) {
            const float aw = __expf(sm[w] - gm);
            gl += sl[w] * aw;
            #pragma unroll
            for (int i = 0; i < VPL; ++i)
                gacc[i] += sacc[w][lane + i * WAVE_SIZE] * aw;
        }
        const float inv = 1.0f / gl;
        #pragma unroll
        for (int i = 0; i < VPL; ++i) {
            const int d = lane + i * WAVE_SIZE;
            if (d < D)
                reinterpret_cast<short*>(Out)[q_base + d] =
                    float_to_bf16_bits(gacc[i] * inv);
        }
    }
}

}  // namespace

torch::Tensor attn_decode(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                          double scale) {
    TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16);
    TORCH_CHECK(q.dim() == 4 && q.size(2) == 1, "q must be [B, H, 1, D]");
    const long B = q.size(0), H = q.size(1), D = q.size(3);
    const long Hkv = k.size(1), S = k.size(2);
    TORCH_CHECK(k.size(3) == D && v.size(2) == S);
    auto qc = q.contiguous(), kc = k.contiguous(), vc = v.contiguous();
    auto out = torch::empty_like(qc);
    auto stream = c10::hip::getCurrentHIPStream().stream();
    const int grid = (int)(B * H);

    #define LAUNCH_DEC(DD)                                                \
        hipLaunchKernelGGL(attn_decode_kernel<DD>, dim3(grid),            \
            dim3(THREADS), 0, stream,                                     \
            reinterpret_cast<const bf16*>(qc.data_ptr()),                 \
            reinterpret_cast<const bf16*>(kc.data_ptr()),                 \
            reinterpret_cast<const bf16*>(vc.data_ptr()),                 \
            reinterpret_cast<bf16*>(out.data_ptr()),                      \
            (int)B, (int)H, (int)Hkv, (int)S, (float)scale)

    switch (D) {
        case 64: LAUNCH_DEC(64); break;
        case 80: LAUNCH_DEC(80); break;
        case 96: LAUNCH_DEC(96); break;
        case 128: LAUNCH_DEC(128); break;
        default: TORCH_CHECK(false, "unsupported head dim ", D);
    }
    #undef LAUNCH_DEC
    HIP_CHECK_LAST();
    return out;
}
