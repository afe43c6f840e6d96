// Fused RMSNorm forward/backward for gfx950 (bf16 I/O, fp32 statistics).
// Llama-family norm: y = x * rstd * gamma, rstd = 1/sqrt(mean(x^2) + eps).
// Memory-bound; same vectorized structure as layernorm.hip.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;
constexpr int VEC = 8;

__global__ void rms_fwd_kernel(
    const bf16x8* __restrict__ x,
    const floatx4* __restrict__ gamma,
    bf16x8* __restrict__ y,
    float* __restrict__ rstd_out,
    int rows, int hv, float eps) {
    __shared__ float scratch[BLOCK / WAVE_SIZE];

    for (int row = blockIdx.x; row < rows; row += gridDim.x) {
        const bf16x8* xrow = x + (long)row * hv;
        float sumsq = 0.f;
        for (int i = threadIdx.x; i < hv; i += BLOCK) {
            bf16x8 v = xrow[i];
            #pragma unroll
            for (int k = 0; k < VEC; ++k) {
                float f = bf16_bits_to_float(v[k]);
                sumsq += f * f;
            }
        }
        sumsq = block_reduce_sum(sumsq, scratch);
        const float rstd = rsqrtf(sumsq / (hv * VEC) + eps);
        if (threadIdx.x == 0) rstd_out[row] = rstd;

        bf16x8* yrow = y + (long)row * hv;
        for (int i = threadIdx.x; i < hv; i += BLOCK) {
            bf16x8 v = xrow[i];
            floatx4 g0 = gamma[i * 2], g1 = gamma[i * 2 + 1];
            bf16x8 o;
            #pragma unroll
            for (int k = 0; k < VEC; ++k) {
                float g = (k < 4) ? g0[k] : g1[k - 4];
                o[k] = float_to_bf16_bits(bf16_bits_to_float(v[k]) * rstd * g);
            }
            yrow[i] = o;
        }
        __syncthreads();
    }
}

// dx = rstd * (dy*g - xhat * mean(dy*g*xhat)), xhat = x * rstd
__global__ void rms_bwd_kernel(
    const bf16x8* __restrict__ dy,
    const bf16x8* __restrict__ x,
    const floatx4* __restrict__ gamma,
    const float* __restrict__ rstd_in,
    bf16x8* __restrict__ dx,
    float* __restrict__ dgamma_part,   // [gridDim.x, H]
    int rows, int hv) {
    __shared__ float scratch[BLOCK / WAVE_SIZE];
    extern __shared__ __attribute__((aligned(16))) float acc_lds[];  // [H]

    const int H = hv * VEC;
    float* dg = acc_lds;   // per-block LDS accumulator (global write once)
    for (int i = threadIdx.x; i < H; i += BLOCK) dg[i] = 0.f;

    for (int row = blockIdx.x; row < rows; row += gridDim.x) {
        const bf16x8* dyrow = dy + (long)row * hv;
        const bf16x8* xrow = x + (long)row * hv;
        const float rstd = rstd_in[row];

        float s2 = 0.f;
        for (int i = threadIdx.x; i < hv; i += BLOCK) {
            bf16x8 dv = dyrow[i], xv = xrow[i];
            floatx4 g0 = gamma[i * 2], g1 = gamma[i * 2 + 1];
            #pragma unroll
            for (int k = 0; k < VEC; ++k) {
                float g = (k < 4) ? g0[k] : g1[k - 4];
                s2 += bf16_bits_to_float(dv[k]) * g
                      * bf16_bits_to_float(xv[k]) * rstd;
            }
        }
        s2 = block_reduce_sum(s2, scratch) / (hv * VEC);

        bf16x8* dxrow = dx + (long)row * hv;
        for (int i = threadIdx.x; i < hv; i += BLOCK) {
            bf16x8 dv = dyrow[i], xv = xrow[i];
            floatx4 g0 = gamma[i * 2], g1 = gamma[i * 2 + 1];
            bf16x8 o;
            #pragma unroll
            for (int k = 0; k < VEC; ++k) {
                float g = (k < 4) ? g0[k] : g1[k - 4];
                float dyf = bf16_bits_to_float(dv[k]);
                float xhat = bf16_bits_to_float(xv[k]) * rstd;
                o[k] = float_to_bf16_bits(rstd * (dyf * g) - rstd * xhat * s2);
                dg[k * hv + i] += dyf * xhat;   // [k][i]: bank-friendly
            }
            dxrow[i] = o;
        }
        __syncthreads();
    }

    for (int i = threadIdx.x; i < H; i += BLOCK)
        dgamma_part[(long)blockIdx.x * H + i] = dg[(i % VEC) * hv + i / VEC];
}

__global__ void rms_bwd_reduce_kernel(
    const float* __restrict__ dgamma_part,
    float* __restrict__ dgamma,
    int nslabs, int H) {
    const int col = blockIdx.x * blockDim.x + threadIdx.x;
    if (col >= H) return;
    float g = 0.f;
    for (int s = 0; s < nslabs; ++s) g += dgamma_part[(long)s * H + col];
    dgamma[col] = g;
}

}  // namespace

std::vector<torch::Tensor> rmsnorm_fwd(
    torch::Tensor x, torch::Tensor gamma, double eps) {
    TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16, "x must be CUDA bf16");
    TORCH_CHECK(x.is_contiguous());
    const long H = x.size(-1);
    TORCH_CHECK(H % 8 == 0);
    const long rows = x.numel() / H;

    auto y = torch::empty_like(x);
    auto rstd = torch::empty({rows}, x.options().dtype(torch::kFloat32));
    auto gamma_f = gamma.to(torch::kFloat32).contiguous();

    const int grid = (int)std::min<long>(rows, 2048);
    hipLaunchKernelGGL(rms_fwd_kernel, dim3(grid), dim3(BLOCK), 0,
        c10::hip::getCurrentHIPStream().stream(),
        reinterpret_cast<const bf16x8*>(x.data_ptr()),
        reinterpret_cast<const floatx4*>(gamma_f.data_ptr()),
        reinterpret_cast<bf16x8*>(y.data_ptr()),
        rstd.data_ptr<float>(), (int)rows, (int)(H / 8), (float)eps);
    HIP_CHECK_LAST();
    return {y, rstd};
}

std::vector<torch::Tensor> rmsnorm_bwd(
    torch::Tensor dy, torch::Tensor x, torch::Tensor gamma, torch::Tensor rstd) {
    auto dyc = dy.contiguous();
    const long H = x.size(-1);
    const long rows = x.numel() / H;

    auto dx = torch::empty_like(x);
    const int grid = (int)std::min<long>(rows, 128);
    auto dgamma_part = torch::empty({grid, H}, x.options().dtype(torch::kFloat32));
    auto gamma_f = gamma.to(torch::kFloat32).contiguous();

    TORCH_CHECK(H <= 16384, "rmsnorm bwd supports hidden size <= 16384");
    hipLaunchKernelGGL(rms_bwd_kernel, dim3(grid), dim3(BLOCK), H * (int)sizeof(float),
        c10::hip::getCurrentHIPStream().stream(),
        reinterpret_cast<const bf16x8*>(dyc.data_ptr()),
        reinterpret_cast<const bf16x8*>(x.data_ptr()),
        reinterpret_cast<const floatx4*>(gamma_f.data_ptr()),
        rstd.data_ptr<float>(),
        reinterpret_cast<bf16x8*>(dx.data_ptr()),
        dgamma_part.data_ptr<float>(), (int)rows, (int)(H / 8));
    HIP_CHECK_LAST();

    auto dgamma = torch::empty({H}, x.options().dtype(torch::kFloat32));
    hipLaunchKernelGGL(rms_bwd_reduce_kernel,
        dim3((H + 255) / 256), dim3(256), 0,
        c10::hip::getCurrentHIPStream().stream(),
        dgamma_part.data_ptr<float>(), dgamma.data_ptr<float>(),
        grid, (int)H);
    HIP_CHECK_LAST();
    return {dx, dgamma};
}
