// bf16 MFMA GEMM with fused epilogue for gfx950.
//
// C[M,N] = A[M,K] @ B[N,K]^T (+bias, +GELU) — B is given N-major ("weight
// layout", torch Linear's [out, in]), so BOTH operands feed the MFMA with
// 8 contiguous bf16 per lane along K.
//
// Structure (cdna_hip_programming.md §5 anatomy, ladder steps 0-2):
// 128x128 block tile, BK=64, 4 waves in a 2x2 grid (64x64 per wave,
// 4x4 fragments of mfma_f32_16x16x32_bf16), double-buffered LDS with
// XOR-swizzled ds_write_b128/ds_read_b128 staging (T2: breaks the
// 16-way bank conflict of 128-B-stride rows).
//
// Fragment layout (mfma_f32_16x16x32_bf16, verified by the tile-probe op
// + asymmetric-input GPU tests):
//   A: lane l holds A[l & 15][8*(l>>4) + i], i = 0..7
//   B: lane l holds B^T[8*(l>>4) + i][l & 15] = Bn[l & 15][8*(l>>4) + i]
//   C/D: lane l, reg r -> row (l>>4)*4 + r, col l & 15
//
// A plain-GEMM fallback it is NOT: torch.matmul (hipBLASLt) serves plain
// GEMMs; this kernel exists for the fused epilogues (bias+GELU etc.).

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

typedef short bf16x4 __attribute__((ext_vector_type(4)));

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int THREADS = 256;

// XOR swizzle on the 16-byte-slot index within a [rows][BK] bf16 tile:
// row stride is BK*2 = 128 B = 8 slots; slot ^= (row & 7) spreads a
// 16-lane column-read group over 8 slots (guide §6 G4).
__device__ __forceinline__ int swz_slot(int row, int slot) {
    return (row * (BK / 8)) + (slot ^ (row & 7));
}

__device__ __forceinline__ float gelu_tanh(float x) {
    const float c = 0.7978845608028654f;  // sqrt(2/pi)
    float t = tanhf(c * (x + 0.044715f * x * x * x));
    return 0.5f * x * (1.f + t);
}

template <int EPILOGUE>  // 0 = none, 1 = bias, 2 = bias + gelu
__global__ __launch_bounds__(THREADS, 2) void gemm_bf16_kernel(
    const bf16x8* __restrict__ A,   // [M, K] row-major, viewed as 16B chunks
    const bf16x8* __restrict__ B,   // [N, K] row-major
    bf16* __restrict__ C,           // [M, N] row-major
    const float* __restrict__ bias, // [N] or null
    int M, int N, int K,
    int ntile_n) {
    // XCD-aware tile swizzle (T1): contiguous tiles per XCD for L2 reuse
    int nwg = gridDim.x;
    int wg = blockIdx.x;
    if (nwg % 8 == 0) {
        int cpx = nwg / 8;
        wg = (wg % 8) * cpx + wg / 8;
    }
    const int tile_m = wg / ntile_n;
    const int tile_n = wg % ntile_n;

    const int kv = K / 8;            // 16B chunks per row
    const int wave = threadIdx.x >> 6;
    const int wm = wave >> 1;        // wave row (0..1)
    const int wn = wave & 1;         // wave col (0..1)

    extern __shared__ __attribute__((aligned(16))) char smem[];
    // [2 buffers][A: BM x BK | B: BN x BK] bf16 (computed offsets: an LDS
    // pointer array in a local initializer fails to compile for gfx950)
    auto sA = [&](int buf) {
        return reinterpret_cast<bf16x8*>(smem + buf * 2 * (BM + BN) * BK);
    };
    auto sB = [&](int buf) {
        return reinterpret_cast<bf16x8*>(
            smem + buf * 2 * (BM + BN) * BK + 2 * BM * BK);
    };

    // staging: global_load_lds DMA, 16 B per lane (ladder step 3 — the
    // compiler never auto-emits it; width 4 -> 16 alone is +67% there).
    // The LDS image is lane-linear (chunk p = row*8 + slot), so the XOR
    // swizzle moves to the SOURCE address: position (row, s) receives
    // source slot s ^ (row & 7) — the same involution the reads apply
    // (guide rule 21: both sides or neither).
    typedef __attribute__((address_space(1))) const unsigned int glds_src_t;
    typedef __attribute__((address_space(3))) unsigned int glds_dst_t;
    const int lane = threadIdx.x & 63;
    const int wave4 = threadIdx.x >> 6;

    auto stage = [&](int buf, int k0) {
        const long a_base = (long)tile_m * BM;
        const long b_base = (long)tile_n * BN;
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
            const int chunkbase = (j * 4 + wave4) * 64;   // wave-uniform
            const int p = chunkbase + lane;
            const int row = p >> 3;
            const int src_slot = (p & 7) ^ (row & 7);
            const bf16x8* asrc = A + (a_base + row) * kv + k0 / 8 + src_slot;
            __builtin_amdgcn_global_load_lds(
                (glds_src_t*)asrc,
                (glds_dst_t*)(reinterpret_cast<char*>(sA(buf)) + chunkbase * 16),
                16, 0, 0);
            const bf16x8* bsrc = B + (b_base + row) * kv + k0 / 8 + src_slot;
            __builtin_amdgcn_global_load_lds(
                (glds_src_t*)bsrc,
                (glds_dst_t*)(reinterpret_cast<char*>(sB(buf)) + chunkbase * 16),
                16, 0, 0);
        }
    };

    floatx4 acc[4][4];
    #pragma unroll
    for (int i = 0; i < 4; ++i)
        #pragma unroll
        for (int j = 0; j < 4; ++j)
            acc[i][j] = floatx4{0.f, 0.f, 0.f, 0.f};

    stage(0, 0);
    __syncthreads();

    const int nk = K / BK;
    for (int kt = 0; kt < nk; ++kt) {
        if (kt + 1 < nk) stage((kt + 1) & 1, (kt + 1) * BK);
        const int buf = kt & 1;

        // per wave: rows wm*64 + i*16 + (lane&15), cols wn*64 + j*16 + ...
        #pragma unroll
        for (int ks = 0; ks < 2; ++ks) {       // two K=32 halves of BK
            bf16x8 a_frag[4], b_frag[4];
            #pragma unroll
            for (int i = 0; i < 4; ++i) {
                int row = wm * 64 + i * 16 + (lane & 15);
                int slot = ks * 4 + (lane >> 4);   // 8 slots per row: k chunk
                a_frag[i] = sA(buf)[swz_slot(row, slot)];
            }
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                int row = wn * 64 + j * 16 + (lane & 15);
                int slot = ks * 4 + (lane >> 4);
                b_frag[j] = sB(buf)[swz_slot(row, slot)];
            }
            #pragma unroll
            for (int i = 0; i < 4; ++i)
                #pragma unroll
                for (int j = 0; j < 4; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a_frag[i], b_frag[j], acc[i][j], 0, 0, 0);
        }
        __syncthreads();
    }

    // epilogue: C[row][col], row = tile_m*BM + wm*64 + i*16 + (lane>>4)*4 + r
    const long row0 = (long)tile_m * BM + wm * 64;
    const long col0 = (long)tile_n * BN + wn * 64;
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            long row = row0 + i * 16 + (lane >> 4) * 4 + r;
            if (row >= M) continue;
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                long col = col0 + j * 16 + (lane & 15);
                float v = acc[i][j][r];
                if (EPILOGUE >= 1) v += bias[col];
                if (EPILOGUE >= 2) v = gelu_tanh(v);
                C[row * N + col] = __float2bfloat16(v);
            }
        }
    }
}

// Single-wave probe: D[16,16] = A[16,32] @ B[32,16] with the assumed
// fragment layout — the layout-verification oracle for the GPU tests.
__global__ void mfma_tile_probe_kernel(
    const short* __restrict__ A,  // [16][32] bf16 bits, row-major
    const short* __restrict__ B,  // [32][16]
    float* __restrict__ D) {      // [16][16]
    const int lane = threadIdx.x;
    bf16x8 a, b;
    #pragma unroll
    for (int i = 0; i < 8; ++i) {
        a[i] = A[(lane & 15) * 32 + 8 * (lane >> 4) + i];
        b[i] = B[(8 * (lane >> 4) + i) * 16 + (lane & 15)];
    }
    floatx4 c = {0.f, 0.f, 0.f, 0.f};
    c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
    #pragma unroll
    for (int r = 0; r < 4; ++r)
        D[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = c[r];
}

}  // namespace

torch::Tensor gemm_bf16(
    torch::Tensor a, torch::Tensor b_nk, c10::optional<torch::Tensor> bias,
    long epilogue) {
    TORCH_CHECK(a.is_cuda() && a.dtype() == torch::kBFloat16, "a must be CUDA bf16");
    TORCH_CHECK(b_nk.dtype() == torch::kBFloat16, "b must be bf16 [N, K]");
    TORCH_CHECK(a.is_contiguous() && b_nk.is_contiguous());
    const long M = a.size(0), K = a.size(1), N = b_nk.size(0);
    TORCH_CHECK(b_nk.size(1) == K, "K mismatch");
    TORCH_CHECK(M % BM == 0 && N % BN == 0 && K % BK == 0,
                "shapes must tile by 128x128x64 (pad upstream)");
    TORCH_CHECK(epilogue == 0 || bias.has_value(), "epilogue needs bias");

    auto c = torch::empty({M, N}, a.options());
    torch::Tensor bias_f;
    const float* bias_ptr = nullptr;
    if (bias.has_value()) {
        bias_f = bias->to(torch::kFloat32).contiguous();
        bias_ptr = bias_f.data_ptr<float>();
    }

    const int ntile_n = (int)(N / BN);
    const int grid = (int)((M / BM) * ntile_n);
    const int lds_bytes = 2 * 2 * (BM + BN) * BK;  // double buffer
    auto stream = c10::hip::getCurrentHIPStream().stream();

    #define LAUNCH(E)                                                        \
        hipLaunchKernelGGL(gemm_bf16_kernel<E>, dim3(grid), dim3(THREADS),   \
            lds_bytes, stream,                                               \
            reinterpret_cast<const bf16x8*>(a.data_ptr()),                   \
            reinterpret_cast<const bf16x8*>(b_nk.data_ptr()),                \
            reinterpret_cast<bf16*>(c.data_ptr()), bias_ptr,                 \
            (int)M, (int)N, (int)K, ntile_n)
    if (epilogue == 0) LAUNCH(0);
    else if (epilogue == 1) LAUNCH(1);
    else LAUNCH(2);
    #undef LAUNCH
    HIP_CHECK_LAST();
    return c;
}

torch::Tensor mfma_tile_probe(torch::Tensor a, torch::Tensor b) {
    TORCH_CHECK(a.is_cuda() && a.sizes() == torch::IntArrayRef({16, 32}));
    TORCH_CHECK(b.sizes() == torch::IntArrayRef({32, 16}));
    auto d = torch::empty({16, 16}, a.options().dtype(torch::kFloat32));
    hipLaunchKernelGGL(mfma_tile_probe_kernel, dim3(1), dim3(64), 0,
                       c10::hip::getCurrentHIPStream().stream(),
                       reinterpret_cast<const short*>(a.contiguous().data_ptr()),
                       reinterpret_cast<const short*>(b.contiguous().data_ptr()),
                       d.data_ptr<float>());
    HIP_CHECK_LAST();
    return d;
}

// Debug probe: pin ds_read_b64_tr_b16 lane semantics empirically.
// Fills LDS[i] = i (256 shorts), every lane reads with addr = base +
// (lane * stride_elems) * 2 bytes, returns the 4 elems per lane.
namespace {
typedef short short4v __attribute__((ext_vector_type(4)));
__global__ void tr16_probe_kernel(short* __restrict__ out, int stride_elems) {
    __shared__ short lds[256];
    if (threadIdx.x < 256) lds[threadIdx.x] = (short)threadIdx.x;
    __syncthreads();
    const int lane = threadIdx.x & 63;
    short4v v = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
        (__attribute__((address_space(3))) short4v*)&lds[lane * stride_elems]);
    #pragma unroll
    for (int j = 0; j < 4; ++j) out[lane * 4 + j] = v[j];
}
}  // namespace

torch::Tensor tr16_probe(long stride_elems) {
    auto out = torch::empty({64, 4},
        torch::TensorOptions().dtype(torch::kInt16).device(torch::kCUDA));
    hipLaunchKernelGGL(tr16_probe_kernel, dim3(1), dim3(64), 0,
        c10::hip::getCurrentHIPStream().stream(),
        reinterpret_cast<short*>(out.data_ptr()), (int)stride_elems);
    HIP_CHECK_LAST();
    return out;
}
