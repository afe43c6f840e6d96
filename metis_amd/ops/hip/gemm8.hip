// 256x256x64-tile bf16 GEMM for gfx950 — the cdna guide's top-tier
// "glds + 2 LDS buffers + BK=64" structure (cdna_hip_programming.md §5
// glds table): C[M,N] = A[M,K] @ B^T where B is stored [N][K]
// (torch F.linear weight layout).
//
//   * 512 threads = 8 waves as 2(M) x 4(N); per-wave output 128 x 64
//     = 8x4 fragments of v_mfma_f32_16x16x32_bf16 (128 accumulator
//     VGPRs), K consumed 64 per tile;
//   * both operands stage global -> LDS with global_load_lds dwordx4
//     (lane-linear destination), st_16x32 XOR swizzle applied on the
//     SOURCE address and re-applied on every ds_read_b128 (guide rule
//     21: both-sides-or-neither) — 4-way banks instead of 16-way;
//   * two LDS buffers (128 KiB total = the whole per-CU budget at one
//     512-thread block, 8 waves = 2/SIMD): tile t+1's 16 glds issue
//     before tile t's compute, land during it, drained by a counted
//     s_waitcnt at the loop bottom;
//   * s_setprio(1) around each 16-MFMA quadrant cluster (T5);
//     bijective XCD-aware block remap (T1) for L2 locality.
//
// Index math (staging, swizzle, fragment gathers, MFMA lane semantics,
// C mapping) is verified lane-for-lane against numpy by
// scripts/gemm8_sim.py before this kernel is trusted.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int BM = 256, BN = 256, BK = 64;
constexpr int THREADS = 512;
constexpr int HALF_E = 128 * 64;          // elements per half image
constexpr int ROW_B = BK * 2;             // 128 B per LDS row

__device__ __forceinline__ int swz(int byte_off) {
    // st_16x32: XOR byte bit5 with bit9 within each 1024-B subtile
    return byte_off ^ (((byte_off >> 9) & 1) << 5);
}

template <bool GUARD>
__global__ __launch_bounds__(THREADS, 2) void gemm8_kernel(
    const bf16* __restrict__ A,   // [M, K] row-major
    const bf16* __restrict__ B,   // [N, K] row-major (B^T)
    bf16* __restrict__ C,         // [M, N]
    int M, int N, int K) {
    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
    const int col16 = lane & 15;
    const int k8 = lane >> 4;

    extern __shared__ __attribute__((aligned(16))) char smem[];
    short* lds = reinterpret_cast<short*>(smem);
    // buffer b, operand o (0 = A, 1 = B), half h
    auto img = [&](int b, int o, int h) {
        return lds + (((b * 2 + o) * 2 + h) * HALF_E);
    };

    // bijective XCD remap (T1): consecutive tiles share an XCD's L2
    const int nwg = gridDim.x;
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = blockIdx.x % 8, idx = blockIdx.x / 8;
    const int wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
                   + idx;
    const int tiles_n = N / BN;
    const long tm = (long)(wg / tiles_n) * BM;
    const long tn = (long)(wg % tiles_n) * BN;

    const int wm = wave >> 2, wn = wave & 3;      // 2 x 4 wave grid
    const int m0 = wm * 128, n0 = wn * 64;        // per-wave sub-tile

    // stage one K-tile (4 half-tiles x 16 KiB) into buffer b:
    // 2 glds per thread per half-tile, lane-linear destination, source
    // pre-swizzled so the swizzled ds_read finds natural data
    auto stage = [&](int b, int k0) {
        const bf16* src[2] = {A, B};
        const long base_row[2] = {tm, tn};
        #pragma unroll
        for (int o = 0; o < 2; ++o)
            #pragma unroll
            for (int h = 0; h < 2; ++h) {
                short* dst_half = img(b, o, h);
                #pragma unroll
                for (int u = 0; u < 2; ++u) {
                    const int dst_byte = (int)(u * THREADS + threadIdx.x) * 16;
                    const int sb = swz(dst_byte);
                    const int row = sb / ROW_B;
                    const int col = (sb % ROW_B) / 2;
                    const bf16* sp = src[o]
                        + (base_row[o] + h * 128 + row) * (long)K + k0 + col;
                    typedef __attribute__((address_space(1)))
                        const unsigned int glds_src_t;
                    typedef __attribute__((address_space(3)))
                        unsigned int glds_dst_t;
                    __builtin_amdgcn_global_load_lds(
                        (glds_src_t*)(const void*)sp,
                        (glds_dst_t*)(void*)(
                            dst_half + (long)(u * THREADS + wave * 64) * 8),
                        16, 0, 0);
                }
            }
    };

    // fragment read: 16 B at the swizzled offset of (row_in_half, kchunk)
    auto frag = [&](const short* half_img, int hrow, int kbyte) {
        const int byte = hrow * ROW_B + kbyte;
        return *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const char*>(half_img) + swz(byte));
    };

    floatx4 acc[8][4];
    #pragma unroll
    for (int fm = 0; fm < 8; ++fm)
        #pragma unroll
        for (int fn = 0; fn < 4; ++fn)
            acc[fm][fn] = floatx4{0.f, 0.f, 0.f, 0.f};

    const short* Ah = nullptr;  // this wave's A half: rows m0..m0+127
    const short* Bh[2] = {nullptr, nullptr};

    stage(0, 0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

    const int ktiles = K / BK;
    for (int kt = 0; kt < ktiles; ++kt) {
        const int buf = kt & 1;
        if (kt + 1 < ktiles) stage(buf ^ 1, (kt + 1) * BK);

        // this wave's images: A half = wm; B halves for cols n0..n0+63
        Ah = img(buf, 0, wm);
        Bh[0] = img(buf, 1, (n0) / 128);
        // per-wave 64 B cols live inside ONE half (n0 % 128 in {0, 64})
        const int brow0 = n0 % 128;

        #pragma unroll
        for (int kc = 0; kc < 2; ++kc) {           // K chunks of 32
            const int kbyte = (kc * 32 + k8 * 8) * 2;
            // B fragments for the wave's 4 N-positions (reused over fm)
            bf16x8 bfr[4];
            #pragma unroll
            for (int fn = 0; fn < 4; ++fn)
                bfr[fn] = frag(Bh[0], brow0 + fn * 16 + col16, kbyte);
            #pragma unroll
            for (int fm = 0; fm < 8; ++fm) {
                const bf16x8 afr = frag(Ah, fm * 16 + col16, kbyte);
                __builtin_amdgcn_s_setprio(1);     // T5: MFMA cluster
                #pragma unroll
                for (int fn = 0; fn < 4; ++fn)
                    acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        afr, bfr[fn], acc[fm][fn], 0, 0, 0);
                __builtin_amdgcn_s_setprio(0);
            }
        }

        // drain the next tile's glds (issued above, landing under the
        // compute) and rendezvous before it is consumed / this buffer
        // is restaged
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
    }

    // epilogue: C fragment layout row = k8*4 + r_, col = col16
    #pragma unroll
    for (int fm = 0; fm < 8; ++fm)
        #pragma unroll
        for (int fn = 0; fn < 4; ++fn)
            #pragma unroll
            for (int r_ = 0; r_ < 4; ++r_) {
                const long row = tm + m0 + fm * 16 + k8 * 4 + r_;
                const long col = tn + n0 + fn * 16 + col16;
                if (!GUARD || (row < M && col < N))
                    C[row * N + col] = __float2bfloat16(acc[fm][fn][r_]);
            }
}

}  // namespace

torch::Tensor gemm8_bf16(torch::Tensor x, torch::Tensor w) {
    TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16);
    TORCH_CHECK(x.dim() == 2 && w.dim() == 2 && x.size(1) == w.size(1));
    const long M = x.size(0), K = x.size(1), N = w.size(0);
    TORCH_CHECK(M % BM == 0 && N % BN == 0 && K % BK == 0,
                "gemm8: M/N must be multiples of 256, K of 64");
    auto xc = x.contiguous(), wc = w.contiguous();
    auto out = torch::empty({M, N}, x.options());
    const int grid = (int)((M / BM) * (N / BN));
    const int lds_bytes = 2 * 2 * 2 * HALF_E * 2;   // 128 KiB
    static bool attr_set = false;
    if (!attr_set) {
        // dynamic LDS above the 64 KiB default needs the opt-in
        (void)hipFuncSetAttribute(
            reinterpret_cast<const void*>(&gemm8_kernel<false>),
            hipFuncAttributeMaxDynamicSharedMemorySize, lds_bytes);
        attr_set = true;
    }
    auto stream = c10::hip::getCurrentHIPStream().stream();
    hipLaunchKernelGGL(gemm8_kernel<false>, dim3(grid), dim3(THREADS),
                       lds_bytes, stream,
                       reinterpret_cast<const bf16*>(xc.data_ptr()),
                       reinterpret_cast<const bf16*>(wc.data_ptr()),
                       reinterpret_cast<bf16*>(out.data_ptr()),
                       (int)M, (int)N, (int)K);
    HIP_CHECK_LAST();
    return out;
}
