// Fused LayerNorm forward/backward for gfx950 (bf16 I/O, fp32 statistics).
//
// The transformer hot-op the Metis profiler times per layer (reference
// README.md:142-186 prescribes the per-layer profiler; the kernel itself is
// new MI355X work). Memory-bound: the design target is the HBM roofline
// (~6.3 TB/s achievable), reached by 16 B/lane vectorized bf16 access and
// one-pass row statistics (guide G13; scalar bf16 is ~2x slower).
//
// Layout: x [rows, H] row-major bf16; gamma/beta fp32[H]; y bf16;
// mean/rstd fp32[rows] saved for backward. One 256-thread block per row
// (grid-stride over rows), H padded to a multiple of 8 by the wrapper.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;
constexpr int VEC = 8;   // bf16 per lane per step (16 B)

__global__ void ln_fwd_kernel(
    const bf16x8* __restrict__ x,
    const floatx4* __restrict__ gamma,   // read as float4 pairs
    const floatx4* __restrict__ beta,
    bf16x8* __restrict__ y,
    float* __restrict__ mean_out,
    float* __restrict__ rstd_out,
    int rows,
    int hv,            // H / VEC
    float eps) {
    __shared__ float scratch[BLOCK / WAVE_SIZE];

    for (int row = blockIdx.x; row < rows; row += gridDim.x) {
        const bf16x8* xrow = x + (long)row * hv;

        float sum = 0.f, sumsq = 0.f;
        for (int i = threadIdx.x; i < hv; i += BLOCK) {
            bf16x8 v = xrow[i];
            #pragma unroll
            for (int k = 0; k < VEC; ++k) {
                float f = bf16_bits_to_float(v[k]);
                sum += f;
                sumsq += f * f;
            }
        }
        sum = block_reduce_sum(sum, scratch);
        __syncthreads();
        sumsq = block_reduce_sum(sumsq, scratch);

        const float inv_n = 1.f / (hv * VEC);
        const float mean = sum * inv_n;
        const float var = sumsq * inv_n - mean * mean;
        const float rstd = rsqrtf(var + eps);
        if (threadIdx.x == 0) {
            mean_out[row] = mean;
            rstd_out[row] = rstd;
        }

        bf16x8* yrow = y + (long)row * hv;
        for (int i = threadIdx.x; i < hv; i += BLOCK) {
            bf16x8 v = xrow[i];
            floatx4 g0 = gamma[i * 2], g1 = gamma[i * 2 + 1];
            floatx4 b0 = beta[i * 2], b1 = beta[i * 2 + 1];
            bf16x8 o;
            #pragma unroll
            for (int k = 0; k < 4; ++k) {
                float xn = (bf16_bits_to_float(v[k]) - mean) * rstd;
                o[k] = float_to_bf16_bits(xn * g0[k] + b0[k]);
            }
            #pragma unroll
            for (int k = 0; k < 4; ++k) {
                float xn = (bf16_bits_to_float(v[4 + k]) - mean) * rstd;
                o[4 + k] = float_to_bf16_bits(xn * g1[k] + b1[k]);
            }
            yrow[i] = o;
        }
        __syncthreads();
    }
}

// Backward pass 1: dx per row + per-block partial dgamma/dbeta slabs.
// dx = rstd * (dy*g - mean(dy*g) - xhat * mean(dy*g*xhat))
// dgamma/dbeta partials accumulate in LDS across the block's rows (each
// thread owns fixed columns) and hit global memory once per block — the
// per-row global read-modify-write was ~2x this kernel's traffic.
__global__ void ln_bwd_kernel(
    const bf16x8* __restrict__ dy,
    const bf16x8* __restrict__ x,
    const floatx4* __restrict__ gamma,
    const float* __restrict__ mean_in,
    const float* __restrict__ rstd_in,
    bf16x8* __restrict__ dx,
    float* __restrict__ dgamma_part,   // [gridDim.x, H]
    float* __restrict__ dbeta_part,    // [gridDim.x, H]
    int rows,
    int hv) {
    __shared__ float scratch[BLOCK / WAVE_SIZE];
    extern __shared__ __attribute__((aligned(16))) float acc_lds[];  // [2][H]

    const int H = hv * VEC;
    float* dg = acc_lds;
    float* db = acc_lds + H;
    for (int i = threadIdx.x; i < H; i += BLOCK) {
        dg[i] = 0.f;
        db[i] = 0.f;
    }

    for (int row = blockIdx.x; row < rows; row += gridDim.x) {
        const bf16x8* dyrow = dy + (long)row * hv;
        const bf16x8* xrow = x + (long)row * hv;
        const float mean = mean_in[row];
        const float rstd = rstd_in[row];

        float s1 = 0.f, s2 = 0.f;   // mean(dy*g), mean(dy*g*xhat)
        for (int i = threadIdx.x; i < hv; i += BLOCK) {
            bf16x8 dv = dyrow[i];
            bf16x8 xv = xrow[i];
            floatx4 g0 = gamma[i * 2], g1 = gamma[i * 2 + 1];
            #pragma unroll
            for (int k = 0; k < VEC; ++k) {
                float g = (k < 4) ? g0[k] : g1[k - 4];
                float dyg = bf16_bits_to_float(dv[k]) * g;
                float xhat = (bf16_bits_to_float(xv[k]) - mean) * rstd;
                s1 += dyg;
                s2 += dyg * xhat;
            }
        }
        s1 = block_reduce_sum(s1, scratch);
        __syncthreads();
        s2 = block_reduce_sum(s2, scratch);
        const float inv_n = 1.f / (hv * VEC);
        s1 *= inv_n;
        s2 *= inv_n;

        bf16x8* dxrow = dx + (long)row * hv;
        for (int i = threadIdx.x; i < hv; i += BLOCK) {
            bf16x8 dv = dyrow[i];
            bf16x8 xv = xrow[i];
            floatx4 g0 = gamma[i * 2], g1 = gamma[i * 2 + 1];
            bf16x8 o;
            #pragma unroll
            for (int k = 0; k < VEC; ++k) {
                float g = (k < 4) ? g0[k] : g1[k - 4];
                float dyf = bf16_bits_to_float(dv[k]);
                float xhat = (bf16_bits_to_float(xv[k]) - mean) * rstd;
                o[k] = float_to_bf16_bits(rstd * (dyf * g - s1 - xhat * s2));
                // [k][i] layout: consecutive threads hit consecutive banks
                dg[k * hv + i] += dyf * xhat;
                db[k * hv + i] += dyf;
            }
            dxrow[i] = o;
        }
        __syncthreads();
    }

    // one global write of this block's partials ([k][i] -> column order)
    for (int i = threadIdx.x; i < H; i += BLOCK) {
        const int col_i = i / VEC, col_k = i % VEC;
        dgamma_part[(long)blockIdx.x * H + i] = dg[col_k * hv + col_i];
        dbeta_part[(long)blockIdx.x * H + i] = db[col_k * hv + col_i];
    }
}

// Backward pass 2: reduce the per-block slabs into dgamma/dbeta.
__global__ void ln_bwd_reduce_kernel(
    const float* __restrict__ dgamma_part,
    const float* __restrict__ dbeta_part,
    float* __restrict__ dgamma,
    float* __restrict__ dbeta,
    int nslabs,
    int H) {
    const int col = blockIdx.x * blockDim.x + threadIdx.x;
    if (col >= H) return;
    float g = 0.f, b = 0.f;
    for (int s = 0; s < nslabs; ++s) {
        g += dgamma_part[(long)s * H + col];
        b += dbeta_part[(long)s * H + col];
    }
    dgamma[col] = g;
    dbeta[col] = b;
}

}  // namespace

std::vector<torch::Tensor> layernorm_fwd(
    torch::Tensor x, torch::Tensor gamma, torch::Tensor beta, double eps) {
    TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16, "x must be CUDA bf16");
    TORCH_CHECK(x.is_contiguous(), "x must be contiguous");
    const long H = x.size(-1);
    TORCH_CHECK(H % 8 == 0, "hidden size must be a multiple of 8");
    const long rows = x.numel() / H;

    auto y = torch::empty_like(x);
    auto f32 = x.options().dtype(torch::kFloat32);
    auto mean = torch::empty({rows}, f32);
    auto rstd = torch::empty({rows}, f32);
    auto gamma_f = gamma.to(torch::kFloat32).contiguous();
    auto beta_f = beta.to(torch::kFloat32).contiguous();

    const int grid = (int)std::min<long>(rows, 2048);
    hipLaunchKernelGGL(
        ln_fwd_kernel, dim3(grid), dim3(BLOCK), 0,
        c10::hip::getCurrentHIPStream().stream(),
        reinterpret_cast<const bf16x8*>(x.data_ptr()),
        reinterpret_cast<const floatx4*>(gamma_f.data_ptr()),
        reinterpret_cast<const floatx4*>(beta_f.data_ptr()),
        reinterpret_cast<bf16x8*>(y.data_ptr()),
        mean.data_ptr<float>(), rstd.data_ptr<float>(),
        (int)rows, (int)(H / 8), (float)eps);
    HIP_CHECK_LAST();
    return {y, mean, rstd};
}

std::vector<torch::Tensor> layernorm_bwd(
    torch::Tensor dy, torch::Tensor x, torch::Tensor gamma,
    torch::Tensor mean, torch::Tensor rstd) {
    TORCH_CHECK(dy.is_cuda() && dy.dtype() == torch::kBFloat16, "dy must be CUDA bf16");
    auto dyc = dy.contiguous();
    const long H = x.size(-1);
    const long rows = x.numel() / H;

    auto dx = torch::empty_like(x);
    auto f32 = x.options().dtype(torch::kFloat32);
    const int grid = (int)std::min<long>(rows, 128);
    auto dgamma_part = torch::empty({grid, H}, f32);
    auto dbeta_part = torch::empty({grid, H}, f32);
    auto gamma_f = gamma.to(torch::kFloat32).contiguous();

    TORCH_CHECK(H <= 8192, "layernorm bwd supports hidden size <= 8192");
    hipLaunchKernelGGL(
        ln_bwd_kernel, dim3(grid), dim3(BLOCK), 2 * H * sizeof(float),
        c10::hip::getCurrentHIPStream().stream(),
        reinterpret_cast<const bf16x8*>(dyc.data_ptr()),
        reinterpret_cast<const bf16x8*>(x.data_ptr()),
        reinterpret_cast<const floatx4*>(gamma_f.data_ptr()),
        mean.data_ptr<float>(), rstd.data_ptr<float>(),
        reinterpret_cast<bf16x8*>(dx.data_ptr()),
        dgamma_part.data_ptr<float>(), dbeta_part.data_ptr<float>(),
        (int)rows, (int)(H / 8));
    HIP_CHECK_LAST();

    auto dgamma = torch::empty({H}, f32);
    auto dbeta = torch::empty({H}, f32);
    const int rblock = 256;
    hipLaunchKernelGGL(
        ln_bwd_reduce_kernel, dim3((H + rblock - 1) / rblock), dim3(rblock), 0,
        c10::hip::getCurrentHIPStream().stream(),
        dgamma_part.data_ptr<float>(), dbeta_part.data_ptr<float>(),
        dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
        grid, (int)H);
    HIP_CHECK_LAST();
    return {dx, dgamma, dbeta};
}
