// Common device helpers for metis_amd CDNA4 (gfx950) kernels.
//
// Conventions (cdna_hip_programming.md):
//  * wavefront = 64 lanes; block sizes are multiples of 64
//  * bf16 memory traffic is vectorized (short4/short8 reinterpret, 8-16 B
//    per lane) — hipcc does not auto-vectorize scalar bf16 loads (G13)
//  * reductions: wave shuffle first, then one LDS slot per wave
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE_SIZE 64

using bf16 = __hip_bfloat16;

// 8 bf16 = 16 bytes, one dwordx4 load per lane.
typedef short bf16x8 __attribute__((ext_vector_type(8)));
typedef float floatx4 __attribute__((ext_vector_type(4)));

__device__ __forceinline__ float bf16_bits_to_float(short u) {
    union { unsigned int i; float f; } cvt;
    cvt.i = ((unsigned int)(unsigned short)u) << 16;
    return cvt.f;
}

__device__ __forceinline__ short float_to_bf16_bits(float f) {
    // round-to-nearest-even via __float2bfloat16 (hardware conversion)
    __hip_bfloat16 h = __float2bfloat16(f);
    return *reinterpret_cast<short*>(&h);
}

// Wave-wide sum over all 64 lanes (butterfly shuffle).
__device__ __forceinline__ float wave_reduce_sum(float v) {
    #pragma unroll
    for (int off = WAVE_SIZE / 2; off > 0; off >>= 1)
        v += __shfl_xor(v, off, WAVE_SIZE);
    return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
    #pragma unroll
    for (int off = WAVE_SIZE / 2; off > 0; off >>= 1)
        v = fmaxf(v, __shfl_xor(v, off, WAVE_SIZE));
    return v;
}

// Block-wide sum. `scratch` is LDS with >= blockDim.x/64 floats.
// Every thread returns the total.
__device__ __forceinline__ float block_reduce_sum(float v, float* scratch) {
    const int lane = threadIdx.x & (WAVE_SIZE - 1);
    const int wave = threadIdx.x / WAVE_SIZE;
    const int num_waves = blockDim.x / WAVE_SIZE;
    v = wave_reduce_sum(v);
    if (lane == 0) scratch[wave] = v;
    __syncthreads();
    float total = (threadIdx.x < num_waves) ? scratch[threadIdx.x] : 0.f;
    #pragma unroll
    for (int off = WAVE_SIZE / 2; off > 0; off >>= 1)
        total += __shfl_xor(total, off, WAVE_SIZE);
    total = __shfl(total, 0, WAVE_SIZE);
    if (num_waves > 1) {
        // broadcast wave 0's total to every wave through LDS
        if (threadIdx.x == 0) scratch[0] = total;
        __syncthreads();
        total = scratch[0];
        __syncthreads();
    }
    return total;
}

#define HIP_CHECK_LAST()                                              \
    do {                                                              \
        hipError_t _e = hipGetLastError();                            \
        if (_e != hipSuccess)                                         \
            TORCH_CHECK(false, "HIP kernel launch failed: ",          \
                        hipGetErrorString(_e));                       \
    } while (0)
