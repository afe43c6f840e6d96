// Fused AdamW optimizer step for gfx950.
//
// The `optimizer_time_ms` hot-op of the Metis profile schema (reference
// README.md:82; data_loader.py:19 consumes it). One kernel updates the
// fp32 master weights, fp32 m/v moments and the bf16 working copy in a
// single pass — 5 tensors streamed once, memory-bound, vectorized 4
// fp32 / lane (16 B).
//
// update (AdamW, decoupled weight decay):
//   m = b1*m + (1-b1)*g;  v = b2*v + (1-b2)*g^2
//   mhat = m / (1-b1^t); vhat = v / (1-b2^t)
//   p -= lr * (mhat / (sqrt(vhat) + eps) + wd * p)

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;

typedef short bf16x4 __attribute__((ext_vector_type(4)));

__global__ void adamw_kernel(
    float* __restrict__ master,      // fp32 master params
    bf16x4* __restrict__ model,      // bf16 working copy (may be null)
    const bf16x4* __restrict__ grad_bf16,  // one of grad_* is non-null
    const float* __restrict__ grad_f32,
    float* __restrict__ m,
    float* __restrict__ v,
    long n,
    float lr, float beta1, float beta2, float eps,
    float weight_decay, float bias_c1, float bias_c2,
    float grad_scale) {
    const long i0 = (long)(blockIdx.x * (long)BLOCK + threadIdx.x) * 4;
    const long stride = (long)gridDim.x * BLOCK * 4;

    for (long i = i0; i < n; i += stride) {
        float4 g;
        if (grad_bf16 != nullptr) {
            bf16x4 gv = grad_bf16[i / 4];
            g = make_float4(bf16_bits_to_float(gv[0]), bf16_bits_to_float(gv[1]),
                            bf16_bits_to_float(gv[2]), bf16_bits_to_float(gv[3]));
        } else {
            const float4* gp = reinterpret_cast<const float4*>(grad_f32 + i);
            g = *gp;
        }
        float4 p = *reinterpret_cast<float4*>(master + i);
        float4 mo = *reinterpret_cast<float4*>(m + i);
        float4 vo = *reinterpret_cast<float4*>(v + i);

        float gs[4] = {g.x * grad_scale, g.y * grad_scale, g.z * grad_scale, g.w * grad_scale};
        float ps[4] = {p.x, p.y, p.z, p.w};
        float ms[4] = {mo.x, mo.y, mo.z, mo.w};
        float vs[4] = {vo.x, vo.y, vo.z, vo.w};
        bf16x4 out_bf;
        #pragma unroll
        for (int k = 0; k < 4; ++k) {
            ms[k] = beta1 * ms[k] + (1.f - beta1) * gs[k];
            vs[k] = beta2 * vs[k] + (1.f - beta2) * gs[k] * gs[k];
            float mhat = ms[k] * bias_c1;
            float vhat = vs[k] * bias_c2;
            ps[k] -= lr * (mhat / (sqrtf(vhat) + eps) + weight_decay * ps[k]);
            out_bf[k] = float_to_bf16_bits(ps[k]);
        }

        *reinterpret_cast<float4*>(master + i) = make_float4(ps[0], ps[1], ps[2], ps[3]);
        *reinterpret_cast<float4*>(m + i) = make_float4(ms[0], ms[1], ms[2], ms[3]);
        *reinterpret_cast<float4*>(v + i) = make_float4(vs[0], vs[1], vs[2], vs[3]);
        if (model != nullptr) model[i / 4] = out_bf;
    }
}

}  // namespace

void adamw_step(
    torch::Tensor master, torch::Tensor model, torch::Tensor grad,
    torch::Tensor m, torch::Tensor v,
    double lr, double beta1, double beta2, double eps,
    double weight_decay, long step, double grad_scale) {
    TORCH_CHECK(master.is_cuda() && master.dtype() == torch::kFloat32,
                "master must be CUDA fp32");
    const long n = master.numel();
    TORCH_CHECK(n % 4 == 0, "parameter count must be a multiple of 4 "
                "(pad the flat buffer)");
    TORCH_CHECK(m.numel() == n && v.numel() == n && grad.numel() == n,
                "m/v/grad size mismatch");

    const bool grad_is_bf16 = grad.dtype() == torch::kBFloat16;
    const bool has_model = model.defined() && model.numel() == n;

    const float bias_c1 = 1.f / (1.f - powf((float)beta1, (float)step));
    const float bias_c2 = 1.f / (1.f - powf((float)beta2, (float)step));

    const long work = n / 4;
    const int grid = (int)std::min<long>((work + BLOCK - 1) / BLOCK, 2048);
    hipLaunchKernelGGL(
        adamw_kernel, dim3(grid), dim3(BLOCK), 0,
        c10::hip::getCurrentHIPStream().stream(),
        master.data_ptr<float>(),
        has_model ? reinterpret_cast<bf16x4*>(model.data_ptr()) : nullptr,
        grad_is_bf16 ? reinterpret_cast<const bf16x4*>(grad.data_ptr()) : nullptr,
        grad_is_bf16 ? nullptr : grad.data_ptr<float>(),
        m.data_ptr<float>(), v.data_ptr<float>(),
        n, (float)lr, (float)beta1, (float)beta2, (float)eps,
        (float)weight_decay, bias_c1, bias_c2, (float)grad_scale);
    HIP_CHECK_LAST();
}
