// Fused QKV relayout + RoPE for gfx950 (Llama path).
//
// The eager path costs three passes over the QKV tensor per block each
// direction: qkv_split_transpose, rope(q), rope(k) (relayout.hip,
// rope.hip). This kernel does the [B, S, (nq+2nkv)D] -> 3x[B, h, S, D]
// relayout AND the neox half-rotation in ONE pass: rotation pairs
// (d, d+D/2) are both touched while the element is in registers, v heads
// pass through unrotated. Backward is the same walk with the inverse
// rotation (sign-flipped sin), gathering dq/dk/dv into dqkv.
//
// Memory-bound by design: 16-B chunks, writes coalesced in the output
// (bhsd) layout, fp32 cos/sin table read once per (s, d) pair.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;

__global__ void qkv_rope_split_kernel(
    const bf16x8* __restrict__ qkv,   // [B, S, HT*D/8]
    const float* __restrict__ cos_t,  // [S, D/2]
    const float* __restrict__ sin_t,
    bf16x8* __restrict__ q,           // [B, nq, S, D/8]
    bf16x8* __restrict__ k,           // [B, nkv, S, D/8]
    bf16x8* __restrict__ v,
    int B, int S, int nq, int nkv, int D) {
    const int HT = nq + 2 * nkv;
    const int half8 = D / 16;              // 8-chunks per half
    const long total = (long)B * HT * S * half8;
    const long stride = (long)gridDim.x * BLOCK;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < total;
         i += stride) {
        const int c = (int)(i % half8);    // chunk within the FIRST half
        const int s = (int)((i / half8) % S);
        const int h = (int)((i / half8 / S) % HT);
        const int b = (int)(i / half8 / S / HT);

        const long in_row = ((long)b * S + s) * (HT * (D / 8)) + h * (D / 8);
        bf16x8 lo = qkv[in_row + c];
        bf16x8 hi = qkv[in_row + half8 + c];

        if (h < nq + nkv) {                // q and k heads rotate
            const float* ct = cos_t + (long)s * (D / 2) + c * 8;
            const float* st = sin_t + (long)s * (D / 2) + c * 8;
            bf16x8 olo, ohi;
            #pragma unroll
            for (int e = 0; e < 8; ++e) {
                const float xl = bf16_bits_to_float(lo[e]);
                const float xh = bf16_bits_to_float(hi[e]);
                olo[e] = float_to_bf16_bits(xl * ct[e] - xh * st[e]);
                ohi[e] = float_to_bf16_bits(xh * ct[e] + xl * st[e]);
            }
            lo = olo;
            hi = ohi;
        }

        bf16x8* out;
        long out_row;
        if (h < nq) {
            out = q;
            out_row = (((long)b * nq + h) * S + s) * (D / 8);
        } else if (h < nq + nkv) {
            out = k;
            out_row = (((long)b * nkv + (h - nq)) * S + s) * (D / 8);
        } else {
            out = v;
            out_row = (((long)b * nkv + (h - nq - nkv)) * S + s) * (D / 8);
        }
        out[out_row + c] = lo;
        out[out_row + half8 + c] = hi;
    }
}

// backward: dqkv[b, s, h, d] gathered from dq/dk/dv with the transposed
// rotation applied to q/k head grads
__global__ void qkv_rope_split_bwd_kernel(
    const bf16x8* __restrict__ dq,
    const bf16x8* __restrict__ dk,
    const bf16x8* __restrict__ dv_,
    const float* __restrict__ cos_t,
    const float* __restrict__ sin_t,
    bf16x8* __restrict__ dqkv,
    int B, int S, int nq, int nkv, int D) {
    const int HT = nq + 2 * nkv;
    const int half8 = D / 16;
    const long total = (long)B * S * HT * half8;
    const long stride = (long)gridDim.x * BLOCK;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < total;
         i += stride) {
        const int c = (int)(i % half8);
        const int h = (int)((i / half8) % HT);
        const int s = (int)((i / half8 / HT) % S);
        const int b = (int)(i / half8 / HT / S);

        const bf16x8* src;
        long src_row;
        if (h < nq) {
            src = dq;
            src_row = (((long)b * nq + h) * S + s) * (D / 8);
        } else if (h < nq + nkv) {
            src = dk;
            src_row = (((long)b * nkv + (h - nq)) * S + s) * (D / 8);
        } else {
            src = dv_;
            src_row = (((long)b * nkv + (h - nq - nkv)) * S + s) * (D / 8);
        }
        bf16x8 lo = src[src_row + c];
        bf16x8 hi = src[src_row + half8 + c];

        if (h < nq + nkv) {
            const float* ct = cos_t + (long)s * (D / 2) + c * 8;
            const float* st = sin_t + (long)s * (D / 2) + c * 8;
            bf16x8 olo, ohi;
            #pragma unroll
            for (int e = 0; e < 8; ++e) {
                const float xl = bf16_bits_to_float(lo[e]);
                const float xh = bf16_bits_to_float(hi[e]);
                olo[e] = float_to_bf16_bits(xl * ct[e] + xh * st[e]);
                ohi[e] = float_to_bf16_bits(xh * ct[e] - xl * st[e]);
            }
            lo = olo;
            hi = ohi;
        }

        const long out_row = ((long)b * S + s) * (HT * (D / 8)) + h * (D / 8);
        dqkv[out_row + c] = lo;
        dqkv[out_row + half8 + c] = hi;
    }
}

int grid_for(long items) {
    return (int)std::min<long>((items + BLOCK - 1) / BLOCK, 2048);
}

}  // namespace

std::vector<torch::Tensor> qkv_rope_split(
    torch::Tensor qkv, long nq, long nkv, long head_dim,
    torch::Tensor cos_t, torch::Tensor sin_t) {
    TORCH_CHECK(qkv.is_cuda() && qkv.dtype() == torch::kBFloat16);
    TORCH_CHECK(qkv.dim() == 3 && head_dim % 16 == 0);
    const long B = qkv.size(0), S = qkv.size(1);
    TORCH_CHECK(qkv.size(2) == (nq + 2 * nkv) * head_dim, "qkv width mismatch");
    TORCH_CHECK(cos_t.size(0) >= S && cos_t.size(1) == head_dim / 2);
    auto qc = qkv.contiguous();

    auto q = torch::empty({B, nq, S, head_dim}, qkv.options());
    auto k = torch::empty({B, nkv, S, head_dim}, qkv.options());
    auto v = torch::empty({B, nkv, S, head_dim}, qkv.options());
    const long items = (long)B * (nq + 2 * nkv) * S * (head_dim / 16);
    hipLaunchKernelGGL(qkv_rope_split_kernel, dim3(grid_for(items)),
        dim3(BLOCK), 0, c10::hip::getCurrentHIPStream().stream(),
        reinterpret_cast<const bf16x8*>(qc.data_ptr()),
        cos_t.data_ptr<float>(), sin_t.data_ptr<float>(),
        reinterpret_cast<bf16x8*>(q.data_ptr()),
        reinterpret_cast<bf16x8*>(k.data_ptr()),
        reinterpret_cast<bf16x8*>(v.data_ptr()),
        (int)B, (int)S, (int)nq, (int)nkv, (int)head_dim);
    HIP_CHECK_LAST();
    return {q, k, v};
}

torch::Tensor qkv_rope_split_bwd(
    torch::Tensor dq, torch::Tensor dk, torch::Tensor dv, long head_dim,
    torch::Tensor cos_t, torch::Tensor sin_t) {
    const long B = dq.size(0), nq = dq.size(1), S = dq.size(2);
    const long nkv = dk.size(1);
    auto dqkv = torch::empty({B, S, (nq + 2 * nkv) * head_dim}, dq.options());
    const long items = dqkv.numel() / 8 / 2;
    hipLaunchKernelGGL(qkv_rope_split_bwd_kernel, dim3(grid_for(items)),
        dim3(BLOCK), 0, c10::hip::getCurrentHIPStream().stream(),
        reinterpret_cast<const bf16x8*>(dq.contiguous().data_ptr()),
        reinterpret_cast<const bf16x8*>(dk.contiguous().data_ptr()),
        reinterpret_cast<const bf16x8*>(dv.contiguous().data_ptr()),
        cos_t.data_ptr<float>(), sin_t.data_ptr<float>(),
        reinterpret_cast<bf16x8*>(dqkv.data_ptr()),
        (int)B, (int)S, (int)nq, (int)nkv, (int)head_dim);
    HIP_CHECK_LAST();
    return dqkv;
}
