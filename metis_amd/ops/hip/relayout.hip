// Head relayout kernels for gfx950: the [B, S, H*D] <-> [B, H, S, D]
// transposes around attention, fused so each direction is ONE pass.
//
// Eager PyTorch spends six passes per block on this (three .contiguous()
// after chunk+transpose on the way in, one on the way out, and their
// backwards) — ~6% of the flagship step's non-GEMM kernel time.
//
// qkv_split_transpose: qkv [B, S, (nq+2*nkv)*D] -> q [B,nq,S,D],
//                      k/v [B,nkv,S,D]  (and the reverse for backward)
// heads_merge:         x [B, H, S, D] -> [B, S, H*D]  (and reverse)
//
// All element moves are 16-B (8 bf16) chunks along D; writes are fully
// coalesced in the output layout, reads are 16-B granules the L2 absorbs.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;

// one 16-B chunk per thread iteration, indexed in OUTPUT (bhsd) order
__global__ void split_transpose_kernel(
    const bf16x8* __restrict__ qkv,   // [B, S, HT*D/8]
    bf16x8* __restrict__ q,           // [B, nq, S, D/8]
    bf16x8* __restrict__ k,           // [B, nkv, S, D/8]
    bf16x8* __restrict__ v,
    int B, int S, int nq, int nkv, int dv) {   // dv = D / 8
    const int HT = nq + 2 * nkv;
    const long total = (long)B * HT * S * dv;
    const long stride = (long)gridDim.x * BLOCK;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < total; i += stride) {
        const int c = (int)(i % dv);
        const int s = (int)((i / dv) % S);
        const int h = (int)((i / dv / S) % HT);
        const int b = (int)(i / dv / S / HT);
        const bf16x8 val = qkv[((long)b * S + s) * (HT * dv) + h * dv + c];
        if (h < nq) {
            q[(((long)b * nq + h) * S + s) * dv + c] = val;
        } else if (h < nq + nkv) {
            k[(((long)b * nkv + (h - nq)) * S + s) * dv + c] = val;
        } else {
            v[(((long)b * nkv + (h - nq - nkv)) * S + s) * dv + c] = val;
        }
    }
}

// reverse: gather dq/dk/dv back into the fused [B, S, HT*D] layout
__global__ void split_transpose_bwd_kernel(
    const bf16x8* __restrict__ dq,
    const bf16x8* __restrict__ dk,
    const bf16x8* __restrict__ dv_,
    bf16x8* __restrict__ dqkv,
    int B, int S, int nq, int nkv, int dv) {
    const int HT = nq + 2 * nkv;
    const long total = (long)B * S * HT * dv;
    const long stride = (long)gridDim.x * BLOCK;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < total; i += stride) {
        const int c = (int)(i % dv);
        const int h = (int)((i / dv) % HT);
        const int s = (int)((i / dv / HT) % S);
        const int b = (int)(i / dv / HT / S);
        bf16x8 val;
        if (h < nq) {
            val = dq[(((long)b * nq + h) * S + s) * dv + c];
        } else if (h < nq + nkv) {
            val = dk[(((long)b * nkv + (h - nq)) * S + s) * dv + c];
        } else {
            val = dv_[(((long)b * nkv + (h - nq - nkv)) * S + s) * dv + c];
        }
        dqkv[i] = val;
    }
}

// [B, H, S, D] -> [B, S, H*D] (output-indexed, coalesced writes)
__global__ void heads_merge_kernel(
    const bf16x8* __restrict__ x,
    bf16x8* __restrict__ y,
    int B, int H, int S, int dv) {
    const long total = (long)B * S * H * dv;
    const long stride = (long)gridDim.x * BLOCK;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < total; i += stride) {
        const int c = (int)(i % dv);
        const int h = (int)((i / dv) % H);
        const int s = (int)((i / dv / H) % S);
        const int b = (int)(i / dv / H / S);
        y[i] = x[(((long)b * H + h) * S + s) * dv + c];
    }
}

// [B, S, H*D] -> [B, H, S, D]
__global__ void heads_unmerge_kernel(
    const bf16x8* __restrict__ y,
    bf16x8* __restrict__ x,
    int B, int H, int S, int dv) {
    const long total = (long)B * H * S * dv;
    const long stride = (long)gridDim.x * BLOCK;
    for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < total; i += stride) {
        const int c = (int)(i % dv);
        const int s = (int)((i / dv) % S);
        const int h = (int)((i / dv / S) % H);
        const int b = (int)(i / dv / S / H);
        x[i] = y[((long)b * S + s) * (H * dv) + h * dv + c];
    }
}

int grid_for(long chunks) {
    return (int)std::min<long>((chunks + BLOCK - 1) / BLOCK, 2048);
}

}  // namespace

std::vector<torch::Tensor> qkv_split_transpose(
    torch::Tensor qkv, long nq, long nkv, long head_dim) {
    TORCH_CHECK(qkv.is_cuda() && qkv.dtype() == torch::kBFloat16);
    TORCH_CHECK(qkv.dim() == 3 && head_dim % 8 == 0);
    const long B = qkv.size(0), S = qkv.size(1);
    TORCH_CHECK(qkv.size(2) == (nq + 2 * nkv) * head_dim, "qkv width mismatch");
    auto qc = qkv.contiguous();

    auto q = torch::empty({B, nq, S, head_dim}, qkv.options());
    auto k = torch::empty({B, nkv, S, head_dim}, qkv.options());
    auto v = torch::empty({B, nkv, S, head_dim}, qkv.options());
    const long chunks = qkv.numel() / 8;
    hipLaunchKernelGGL(split_transpose_kernel, dim3(grid_for(chunks)),
        dim3(BLOCK), 0, c10::hip::getCurrentHIPStream().stream(),
        reinterpret_cast<const bf16x8*>(qc.data_ptr()),
        reinterpret_cast<bf16x8*>(q.data_ptr()),
        reinterpret_cast<bf16x8*>(k.data_ptr()),
        reinterpret_cast<bf16x8*>(v.data_ptr()),
        (int)B, (int)S, (int)nq, (int)nkv, (int)(head_dim / 8));
    HIP_CHECK_LAST();
    return {q, k, v};
}

torch::Tensor qkv_split_transpose_bwd(
    torch::Tensor dq, torch::Tensor dk, torch::Tensor dv, long head_dim) {
    const long B = dq.size(0), nq = dq.size(1), S = dq.size(2);
    const long nkv = dk.size(1);
    auto dqkv = torch::empty({B, S, (nq + 2 * nkv) * head_dim}, dq.options());
    const long chunks = dqkv.numel() / 8;
    hipLaunchKernelGGL(split_transpose_bwd_kernel, dim3(grid_for(chunks)),
        dim3(BLOCK), 0, c10::hip::getCurrentHIPStream().stream(),
        reinterpret_cast<const bf16x8*>(dq.contiguous().data_ptr()),
        reinterpret_cast<const bf16x8*>(dk.contiguous().data_ptr()),
        reinterpret_cast<const bf16x8*>(dv.contiguous().data_ptr()),
        reinterpret_cast<bf16x8*>(dqkv.data_ptr()),
        (int)B, (int)S, (int)nq, (int)nkv, (int)(head_dim / 8));
    HIP_CHECK_LAST();
    return dqkv;
}

torch::Tensor heads_merge(torch::Tensor x) {
    // [B, H, S, D] -> [B, S, H*D]
    TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16 && x.dim() == 4);
    const long B = x.size(0), H = x.size(1), S = x.size(2), D = x.size(3);
    TORCH_CHECK(D % 8 == 0);
    auto y = torch::empty({B, S, H * D}, x.options());
    const long chunks = x.numel() / 8;
    hipLaunchKernelGGL(heads_merge_kernel, dim3(grid_for(chunks)), dim3(BLOCK),
        0, c10::hip::getCurrentHIPStream().stream(),
        reinterpret_cast<const bf16x8*>(x.contiguous().data_ptr()),
        reinterpret_cast<bf16x8*>(y.data_ptr()),
        (int)B, (int)H, (int)S, (int)(D / 8));
    HIP_CHECK_LAST();
    return y;
}

torch::Tensor heads_unmerge(torch::Tensor y, long H) {
    // [B, S, H*D] -> [B, H, S, D]
    const long B = y.size(0), S = y.size(1), D = y.size(2) / H;
    auto x = torch::empty({B, H, S, D}, y.options());
    const long chunks = y.numel() / 8;
    hipLaunchKernelGGL(heads_unmerge_kernel, dim3(grid_for(chunks)),
        dim3(BLOCK), 0, c10::hip::getCurrentHIPStream().stream(),
        reinterpret_cast<const bf16x8*>(y.contiguous().data_ptr()),
        reinterpret_cast<bf16x8*>(x.data_ptr()),
        (int)B, (int)H, (int)S, (int)(D / 8));
    HIP_CHECK_LAST();
    return x;
}
