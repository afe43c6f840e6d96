// Rotary position embedding (RoPE) for gfx950 — bf16 I/O, neox/llama
// half-rotation style, cos/sin table precomputed on device once
// (guide Appendix B: on-device trig per element turns a memory-bound op
// VALU-bound; a [S, D/2] fp32 table is the fix).
//
// x [B, H, S, D]: y[..., d]       = x[d] * cos[m] - x[d + D/2] * sin[m]
//                 y[..., d + D/2] = x[d + D/2] * cos[m] + x[d] * sin[m]
// with m = d (table column), per position s. Backward = forward with the
// sin sign flipped (rotation transpose), handled by a flag.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;

typedef short bf16x4 __attribute__((ext_vector_type(4)));

__global__ void rope_kernel(
    const bf16x4* __restrict__ x,     // [rows, D] as 4-chunks, rows = B*H*S
    const float* __restrict__ cos_t,  // [S, D/2]
    const float* __restrict__ sin_t,  // [S, D/2]
    bf16x4* __restrict__ y,
    int rows, int S, int D, int sign) {
    const int half = D / 2;
    const long total = (long)rows * half / 4;   // 4 rotation pairs per item
    const long i0 = (long)blockIdx.x * BLOCK + threadIdx.x;
    const long stride = (long)gridDim.x * BLOCK;

    for (long i = i0; i < total; i += stride) {
        const long row = i / (half / 4);
        const int d4 = (int)(i % (half / 4));   // 4-chunk within first half
        const int s = (int)(row % S);

        const long base = row * (D / 4);
        bf16x4 lo = x[base + d4];
        bf16x4 hi = x[base + half / 4 + d4];
        const float* c = cos_t + (long)s * half + d4 * 4;
        const float* sn = sin_t + (long)s * half + d4 * 4;
        bf16x4 olo, ohi;
        #pragma unroll
        for (int k = 0; k < 4; ++k) {
            float xl = bf16_bits_to_float(lo[k]);
            float xh = bf16_bits_to_float(hi[k]);
            float sv = sign * sn[k];
            olo[k] = float_to_bf16_bits(xl * c[k] - xh * sv);
            ohi[k] = float_to_bf16_bits(xh * c[k] + xl * sv);
        }
        y[base + d4] = olo;
        y[base + half / 4 + d4] = ohi;
    }
}

}  // namespace

torch::Tensor rope_apply(
    torch::Tensor x, torch::Tensor cos_t, torch::Tensor sin_t, bool backward) {
    TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16);
    TORCH_CHECK(x.dim() == 4, "x must be [B, H, S, D]");
    const long D = x.size(3), S = x.size(2);
    TORCH_CHECK(D % 8 == 0, "head dim must be a multiple of 8");
    TORCH_CHECK(cos_t.size(0) >= S && cos_t.size(1) == D / 2, "table mismatch");
    auto xc = x.contiguous();
    auto y = torch::empty_like(xc);
    const long rows = x.numel() / D;

    const long work = rows * (D / 2) / 4;
    const int grid = (int)std::min<long>((work + BLOCK - 1) / BLOCK, 2048);
    hipLaunchKernelGGL(rope_kernel, dim3(grid), dim3(BLOCK), 0,
        c10::hip::getCurrentHIPStream().stream(),
        reinterpret_cast<const bf16x4*>(xc.data_ptr()),
        cos_t.data_ptr<float>(), sin_t.data_ptr<float>(),
        reinterpret_cast<bf16x4*>(y.data_ptr()),
        (int)rows, (int)S, (int)D, backward ? -1 : 1);
    HIP_CHECK_LAST();
    return y;
}
